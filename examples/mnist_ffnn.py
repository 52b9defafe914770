"""FFNN classifier on synthetic MNIST-shaped data — demonstrates that the
framework's op layer is a generic NN toolkit, mirroring the reference's
examples/mnist (ref: src/examples/mnist/ — FFNN + LeNet-style models on
MNIST via the expression graph).

Run:  python examples/mnist_ffnn.py [epochs]
Uses marian_amd.ops (linear / linear_act / cross_entropy / adam_step) and
the same fused HIP kernels as the NMT models when a GPU is present.
"""
import sys

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])
from marian_amd import ops  # noqa: E402


def synthetic_mnist(n, seed=0, device="cpu"):
    g = torch.Generator().manual_seed(42)   # shared prototypes
    protos = torch.rand(10, 784, generator=g)
    g2 = torch.Generator().manual_seed(seed)
    y = torch.randint(0, 10, (n,), generator=g2)
    x = protos[y] + 0.35 * torch.randn(n, 784, generator=g2)
    return x.clamp(0, 1).to(device), y.to(device)


def main(epochs=3):
    device = "cuda" if torch.cuda.is_available() else "cpu"
    dt = torch.bfloat16 if device == "cuda" else torch.float32
    x, y = synthetic_mnist(8192, device=device)
    xv, yv = synthetic_mnist(1024, seed=1, device=device)
    dims = [784, 512, 256, 10]
    Ws = [torch.nn.init.xavier_uniform_(
        torch.empty(a, b, device=device, dtype=dt)).requires_grad_(True)
        for a, b in zip(dims, dims[1:])]
    bs = [torch.zeros(b, device=device, dtype=dt, requires_grad=True)
          for b in dims[1:]]
    params = Ws + bs
    opt = torch.optim.Adam(params, lr=1e-3)
    for ep in range(epochs):
        perm = torch.randperm(len(x), device=device)
        total = 0.0
        for i in range(0, len(x), 256):
            idx = perm[i:i + 256]
            h = x[idx].to(dt)
            for W, b in list(zip(Ws, bs))[:-1]:
                h = ops.linear_act(h, W, b, "relu")
            logits = ops.linear(h, Ws[-1], bs[-1])
            losses = ops.cross_entropy(logits.float(), y[idx])
            loss = losses.mean()
            opt.zero_grad()
            loss.backward()
            opt.step()
            total += float(loss) * len(idx)
        with torch.no_grad():
            h = xv.to(dt)
            for W, b in list(zip(Ws, bs))[:-1]:
                h = ops.linear_act(h, W, b, "relu")
            acc = (ops.linear(h, Ws[-1], bs[-1]).argmax(-1) == yv)
            acc = acc.float().mean()
        print(f"epoch {ep + 1}: train ce {total / len(x):.4f}  "
              f"valid acc {float(acc):.3f}")
    return float(acc)


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 3)
