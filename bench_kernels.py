#!/usr/bin/env python3
"""Microbenchmarks for the hand-written HIP kernels (profiling target for
rocprofv3 PMC counter collection — run each kernel in isolation).

    python bench_kernels.py [attention|layernorm|ce|adam|all] [iters]
"""
import sys
import time

import torch

sys.path.insert(0, ".")
from marian_amd.ops import hip_ext


def bench(name, fn, iters=50, bytes_moved=None, flops=None):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    line = f"{name}: {dt*1e6:.1f} us"
    if bytes_moved:
        line += f", {bytes_moved/dt/1e12:.2f} TB/s"
    if flops:
        line += f", {flops/dt/1e12:.1f} TF/s"
    print(line)


def main():
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    iters = int(sys.argv[2]) if len(sys.argv) > 2 else 50
    ext = hip_ext(required=True)
    dev = "cuda"
    torch.manual_seed(0)

    if which in ("attention", "all"):
        B, H, T, D = 64, 16, 512, 64   # bigger T shows the flash tiling
        q = torch.randn(B, H, T, D, device=dev, dtype=torch.bfloat16)
        k, v = torch.randn_like(q), torch.randn_like(q)
        flops = 4 * B * H * T * T * D  # QK^T + PV
        bench("attn_fwd  B64 H16 T512 D64",
              lambda: ext.attention_fwd(q, k, v, None, False, D ** -0.5, False),
              iters, flops=flops)
        o, lse = ext.attention_fwd(q, k, v, None, False, D ** -0.5, True)
        do = torch.randn_like(o)
        bench("attn_bwd  B64 H16 T512 D64",
              lambda: ext.attention_bwd(q, k, v, o, do, lse, None, False,
                                        D ** -0.5),
              iters, flops=int(2.5 * flops))
        B2, T2 = 512, 64               # the NMT training shape
        q2 = torch.randn(B2, H, T2, D, device=dev, dtype=torch.bfloat16)
        k2, v2 = torch.randn_like(q2), torch.randn_like(q2)
        f2 = 4 * B2 * H * T2 * T2 * D
        bench("attn_fwd  B512 H16 T64 D64",
              lambda: ext.attention_fwd(q2, k2, v2, None, False, D ** -0.5,
                                        False), iters, flops=f2)
        o2, lse2 = ext.attention_fwd(q2, k2, v2, None, False, D ** -0.5, True)
        do2 = torch.randn_like(o2)
        bench("attn_bwd  B512 H16 T64 D64",
              lambda: ext.attention_bwd(q2, k2, v2, o2, do2, lse2, None,
                                        False, D ** -0.5),
              iters, flops=int(2.5 * f2))
        bench("attn_bwd  B512 H16 T64 D64 causal",
              lambda: ext.attention_bwd(q2, k2, v2, o2, do2, lse2, None,
                                        True, D ** -0.5),
              iters, flops=int(2.5 * f2))

    if which in ("layernorm", "ln", "all"):
        R, d = 32768, 1024
        x = torch.randn(R, d, device=dev, dtype=torch.bfloat16)
        g = torch.rand(d, device=dev, dtype=torch.bfloat16) + 0.5
        b = torch.randn(d, device=dev, dtype=torch.bfloat16)
        nbytes = R * d * 2 * 2  # in+out (stats negligible)
        bench("ln_fwd    32768x1024", lambda: ext.layernorm_fwd(x, g, b, 1e-9),
              iters, bytes_moved=nbytes)
        y, mean, rstd = ext.layernorm_fwd(x, g, b, 1e-9)
        dy = torch.randn_like(x)
        bench("ln_bwd    32768x1024",
              lambda: ext.layernorm_bwd(x, g, mean, rstd, dy),
              iters, bytes_moved=R * d * 2 * 3)
        r = torch.randn_like(x)
        bench("ln_res_fwd 32768x1024",
              lambda: ext.layernorm_res_fwd(x, r, g, b, 1e-9),
              iters, bytes_moved=R * d * 2 * 4)

    if which in ("ce", "all"):
        N, V = 32768, 32000
        logits = torch.randn(N, V, device=dev, dtype=torch.bfloat16)
        labels = torch.randint(0, V, (N,), device=dev)
        mask = torch.ones(N, device=dev)
        bench("ce_fwd    32768x32000",
              lambda: ext.ce_fwd(logits, labels, 0.1, mask),
              iters, bytes_moved=N * V * 2)
        loss, lse = ext.ce_fwd(logits, labels, 0.1, mask)
        dl = torch.ones(N, device=dev)
        bench("ce_bwd    32768x32000",
              lambda: ext.ce_bwd(logits, labels, lse, dl, 0.1, mask),
              iters, bytes_moved=N * V * 2 * 2)

    if which in ("adam", "all"):
        n = 210_000_000
        p = torch.randn(n, device=dev)
        gr = torch.randn(n, device=dev, dtype=torch.bfloat16)
        m = torch.zeros(n, device=dev)
        v2 = torch.zeros(n, device=dev)
        bench("adam      210M params",
              lambda: ext.adam_step(p, gr, m, v2, 1e-3, 0.9, 0.98, 1e-9,
                                    0.0, 5, 1.0),
              10, bytes_moved=n * 26)

    if which in ("fp8",):
        bench_fp8()

    if which in ("softmax", "all"):
        R, C = 32768, 32000
        x = torch.randn(R, C, device=dev, dtype=torch.bfloat16)
        bench("softmax   32768x32000", lambda: ext.softmax_fwd(x), 20,
              bytes_moved=R * C * 2 * 2)


def bench_fp8():
    """Is torch._scaled_mm actually hitting fp8 MFMA rate on gfx950?"""
    import torch
    dev = "cuda"
    for (M, K, N) in [(32768, 1024, 1024), (32768, 1024, 4096),
                      (32768, 1024, 32000), (8192, 8192, 8192)]:
        a = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        b = torch.randn(K, N, device=dev, dtype=torch.bfloat16)
        flops = 2.0 * M * K * N
        bench(f"bf16 matmul {M}x{K}x{N}", lambda: a @ b, 20, flops=flops)
        a8 = a.to(torch.float8_e4m3fn)
        b8 = b.t().contiguous().t().to(torch.float8_e4m3fn)
        sa = torch.tensor(1.0, device=dev)
        sb = torch.tensor(1.0, device=dev)
        bench(f"fp8 scaled_mm {M}x{K}x{N}",
              lambda: torch._scaled_mm(a8, b8, scale_a=sa, scale_b=sb,
                                       out_dtype=torch.bfloat16),
              20, flops=flops)
        from marian_amd.ops.fp8 import fp8_linear
        bench(f"fp8_linear(+quant) {M}x{K}x{N}",
              lambda: fp8_linear(a, b), 20, flops=flops)


if __name__ == "__main__":
    main()
