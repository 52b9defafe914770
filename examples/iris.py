"""Tiny softmax-regression on a synthetic iris-shaped dataset
(ref: src/examples/iris/ — shows the op layer handles small dense
problems end-to-end)."""
import sys

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])
from marian_amd import ops  # noqa: E402


def main(steps=200):
    g = torch.Generator().manual_seed(0)
    centers = torch.tensor([[5.0, 3.4, 1.5, 0.2],
                            [5.9, 2.8, 4.3, 1.3],
                            [6.6, 3.0, 5.6, 2.1]])
    x = (centers.repeat_interleave(50, 0)
         + 0.3 * torch.randn(150, 4, generator=g))
    y = torch.arange(3).repeat_interleave(50)
    W = torch.zeros(4, 3, requires_grad=True)
    b = torch.zeros(3, requires_grad=True)
    opt = torch.optim.Adam([W, b], lr=0.1)
    for _ in range(steps):
        loss = ops.cross_entropy(ops.linear(x, W, b), y).mean()
        opt.zero_grad()
        loss.backward()
        opt.step()
    acc = float((ops.linear(x, W, b).argmax(-1) == y).float().mean())
    print(f"train accuracy {acc:.3f}")
    return acc


if __name__ == "__main__":
    main()
