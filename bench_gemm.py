"""Per-shape GEMM accounting for the transformer-big training step.

Enumerates every GEMM the B1024xT64 bf16 step actually runs (exact
layouts/transposes as the model code invokes them), times each under the
shipped TunableOp table, and prints TF/s + the aggregate step GEMM time.
This is the measurement VERDICT r1 #8 asks for: where hipBLASLt sits vs
roofline (bf16 MFMA dense peak ~2.5 PF/s) per shape, so hand-MFMA effort
goes only where the library actually loses.

Usage (GPU box):
  python bench_gemm.py            # measure with shipped tuning
  python bench_gemm.py --retune   # TunableOp full-search re-tune pass
"""
import argparse
import os
import sys

_ROOT = os.path.dirname(os.path.abspath(__file__))
os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME",
                      os.path.join(_ROOT, "profiles", "tunableop", "tuned.csv"))

import torch  # noqa: E402


def timeit(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    e0 = torch.cuda.Event(enable_timing=True)
    e1 = torch.cuda.Event(enable_timing=True)
    e0.record()
    for _ in range(iters):
        fn()
    e1.record()
    torch.cuda.synchronize()
    return e0.elapsed_time(e1) / iters * 1e-3  # seconds


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--retune", action="store_true")
    ap.add_argument("--compare", action="store_true",
                    help="also time lt_mm (full-pool algo search) per shape")
    ap.add_argument("--batch", type=int, default=1024)
    ap.add_argument("--seq", type=int, default=64)
    ap.add_argument("--dim", type=int, default=1024)
    ap.add_argument("--ffn", type=int, default=4096)
    ap.add_argument("--vocab", type=int, default=32000)
    ap.add_argument("--enc", type=int, default=6)
    ap.add_argument("--dec", type=int, default=6)
    args = ap.parse_args()
    if args.retune:
        os.environ["PYTORCH_TUNABLEOP_TUNING"] = "1"
        torch.cuda.tunable.tuning_enable(True)

    dev = torch.device("cuda")
    dt = torch.bfloat16
    M = args.batch * args.seq
    d, F, V = args.dim, args.ffn, args.vocab
    NE, ND = args.enc, args.dec

    def t(*shape, transposed=False):
        x = torch.randn(*shape, device=dev, dtype=dt)
        return x.t().contiguous().t() if transposed else x

    # (name, count/step, fn) — layouts mirror the model exactly:
    # fwd uses addmm (bias) or the lt relu epilogue; dgrad is dy @ W.t()
    # (W row-major -> NT); wgrad is x.t() @ dy (TN). Logits weight is the
    # TRANSPOSED VIEW of the tied embedding [V,d].t().
    work = []

    def gemm(name, count, flops, make):
        work.append((name, count, flops, make))

    E = t(V, d)           # tied embedding, row-major [V, d]
    bias = {n: torch.randn(n, device=dev, dtype=dt) for n in (d, 3 * d, 2 * d, F, V)}

    def addmm(m, k, n, Wmk=None):
        x = t(m, k)
        W = Wmk if Wmk is not None else t(k, n)
        b = bias[n]
        fn = lambda: torch.addmm(b, x, W)
        if W.stride(1) == 1:
            fn._ltbias = (x, W, b)
        return fn

    def dgrad(m, n, k):
        dy = t(m, n)
        W = t(k, n)   # row-major weight, dgrad uses .t() view (NT)
        fn = lambda: torch.mm(dy, W.t())
        fn._operands = (dy, W.t())
        return fn

    def wgrad(m, k, n):
        x = t(m, k)
        dy = t(m, n)
        fn = lambda: torch.mm(x.t(), dy)
        fn._operands = (x.t(), dy)
        fn._wgrad = (x, dy)
        return fn

    n_self = NE + ND      # self-attn blocks
    gemm("qkv_fwd", n_self, 2 * M * d * 3 * d, lambda: addmm(M, d, 3 * d))
    gemm("qkv_dgrad", n_self, 2 * M * d * 3 * d, lambda: dgrad(M, 3 * d, d))
    gemm("qkv_wgrad", n_self, 2 * M * d * 3 * d, lambda: wgrad(M, d, 3 * d))
    n_out = NE + 2 * ND   # out-proj of self+cross attn
    gemm("attnout_fwd", n_out, 2 * M * d * d, lambda: addmm(M, d, d))
    gemm("attnout_dgrad", n_out, 2 * M * d * d, lambda: dgrad(M, d, d))
    gemm("attnout_wgrad", n_out, 2 * M * d * d, lambda: wgrad(M, d, d))
    gemm("crossq_fwd", ND, 2 * M * d * d, lambda: addmm(M, d, d))
    gemm("crossq_dgrad", ND, 2 * M * d * d, lambda: dgrad(M, d, d))
    gemm("crossq_wgrad", ND, 2 * M * d * d, lambda: wgrad(M, d, d))
    gemm("crosskv_fwd", ND, 2 * M * d * 2 * d, lambda: addmm(M, d, 2 * d))
    gemm("crosskv_dgrad", ND, 2 * M * d * 2 * d, lambda: dgrad(M, 2 * d, d))
    gemm("crosskv_wgrad", ND, 2 * M * d * 2 * d, lambda: wgrad(M, d, 2 * d))

    def ffn1_lt():
        from marian_amd import ops
        x = t(M, d)
        W = t(d, F)
        b = bias[F]
        ext = ops.hip_ext(required=True)
        return lambda: ext.lt_linear_act(x, W, b, True)

    n_ffn = NE + ND
    gemm("ffn1_fwd(lt-relu)", n_ffn, 2 * M * d * F, ffn1_lt)
    gemm("ffn1_dgrad", n_ffn, 2 * M * d * F, lambda: dgrad(M, F, d))
    gemm("ffn1_wgrad", n_ffn, 2 * M * d * F, lambda: wgrad(M, d, F))
    gemm("ffn2_fwd", n_ffn, 2 * M * F * d, lambda: addmm(M, F, d))
    gemm("ffn2_dgrad", n_ffn, 2 * M * F * d, lambda: dgrad(M, d, F))
    gemm("ffn2_wgrad", n_ffn, 2 * M * F * d, lambda: wgrad(M, F, d))

    def logits_fwd():
        x = t(M, d)
        W = E.t()          # transposed VIEW, exactly as output_logits
        b = bias[V]
        return lambda: torch.addmm(b, x, W)

    def logits_dgrad():
        dy = t(M, V)
        fn = lambda: torch.mm(dy, E)   # dx = dy @ (E.t()).t() == dy @ E
        fn._operands = (dy, E)
        return fn

    def logits_wgrad():
        x = t(M, d)
        dy = t(M, V)
        fn = lambda: torch.mm(x.t(), dy)  # [d,V] into E.t().grad
        fn._operands = (x.t(), dy)
        fn._wgrad = (x, dy)
        return fn

    gemm("logits_fwd", 1, 2 * M * d * V, logits_fwd)
    gemm("logits_dgrad", 1, 2 * M * d * V, logits_dgrad)
    gemm("logits_wgrad", 1, 2 * M * d * V, logits_wgrad)

    # --compare: time the torch route vs the full-pool-searched lt_mm
    # (split-K included) on the same operands
    from marian_amd import ops as mops
    ext = mops.hip_ext(required=True) if args.compare else None

    total_ms = 0.0
    total_tf = 0.0
    best_ms = 0.0
    print(f"{'name':24s} {'count':>5s} {'ms/call':>8s} {'TF/s':>7s} "
          f"{'lt ms':>8s} {'lt TF/s':>7s} {'ms/step':>8s} {'TFLOP/step':>10s}")
    rows = []
    for name, count, flops, make in work:
        fn = make()
        sec = timeit(fn)
        tf = flops / sec / 1e12
        lt_txt = ("", "")
        lt_sec = None
        if ext is not None and hasattr(fn, "_operands"):
            A, B = fn._operands
            ref = fn()
            got = ext.lt_mm(A, B)
            err = (got.float() - (ref if ref.dtype == torch.float32
                                  else ref.float())).abs().max().item()
            den = ref.float().abs().max().item()
            assert err <= 0.02 * max(den, 1.0), (name, err, den)
            lt_fn = lambda: ext.lt_mm(A, B)
            lt_sec = timeit(lt_fn)
            lt_txt = (f"{lt_sec*1e3:8.3f}", f"{flops/lt_sec/1e12:7.0f}")
        if ext is not None and hasattr(fn, "_ltbias"):
            xb, Wb, bb = fn._ltbias
            ref = fn().float()
            got = ext.lt_linear_act(xb, Wb, bb, False).float()
            err = (got - ref).abs().max().item()
            ok = err <= 0.02 * max(ref.abs().max().item(), 1.0)
            bsec = timeit(lambda: ext.lt_linear_act(xb, Wb, bb, False))
            print(f"{'  ^ lt-bias fwd':24s} {'':5s} {bsec*1e3:8.3f} "
                  f"{flops/bsec/1e12:7.0f}   {'OK' if ok else 'FAIL'}")
        if ext is not None and hasattr(fn, "_wgrad"):
            xw, dyw = fn._wgrad
            ref = fn().float()
            got = ext.wgrad_bf16(xw, dyw, 0).float()
            err = (got - ref).abs().max().item()
            den = ref.abs().max().item()
            ok = err <= 0.03 * max(den, 1.0)
            wsec = timeit(lambda: ext.wgrad_bf16(xw, dyw, 0))
            print(f"{'  ^ hand-mfma wgrad':24s} {'':5s} {wsec*1e3:8.3f} "
                  f"{flops/wsec/1e12:7.0f}   err={err:.4f}/{den:.1f} "
                  f"{'OK' if ok else 'FAIL'}")
            if wsec < (lt_sec or sec) and ok:
                lt_sec = wsec
        use = min(sec, lt_sec) if lt_sec else sec
        ms_step = sec * 1e3 * count
        best_ms += use * 1e3 * count
        total_ms += ms_step
        total_tf += flops * count / 1e12
        rows.append((name, count, sec * 1e3, tf, ms_step, flops * count / 1e12))
        print(f"{name:24s} {count:5d} {sec*1e3:8.3f} {tf:7.0f} "
              f"{lt_txt[0]:>8s} {lt_txt[1]:>7s} "
              f"{ms_step:8.2f} {flops*count/1e12:10.2f}")
    print(f"{'TOTAL':24s} {'':5s} {'':8s} "
          f"{total_tf/ (total_ms/1e3) :7.0f} {'':8s} {'':7s} {total_ms:8.2f} {total_tf:10.2f}")
    if args.compare:
        print(f"# best-of(torch, lt): {best_ms:.1f} ms/step at "
              f"{total_tf/(best_ms/1e3):.0f} TF/s effective")
    print(f"# aggregate: {total_ms:.1f} ms/step of GEMM at "
          f"{total_tf/(total_ms/1e3):.0f} TF/s effective "
          f"({100*total_tf/(total_ms/1e3)/2495:.0f}% of bf16 dense peak)")

    if args.retune:
        torch.cuda.tunable.write_file()
        print("# wrote re-tuned table to", os.environ["PYTORCH_TUNABLEOP_FILENAME"])


if __name__ == "__main__":
    main()
