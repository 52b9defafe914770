"""CPU reference correctness of the ops layer (the same contracts the HIP
kernels are tested against on the GPU box in tests/test_kernels_gpu.py)."""
import math

import pytest
import torch

from marian_amd import ops


def test_cross_entropy_matches_manual():
    torch.manual_seed(0)
    logits = torch.randn(12, 17)
    labels = torch.randint(0, 17, (12,))
    got = ops.cross_entropy(logits, labels, 0.0)
    want = torch.nn.functional.cross_entropy(logits, labels, reduction="none")
    assert torch.allclose(got, want, atol=1e-6)


def test_cross_entropy_label_smoothing():
    torch.manual_seed(1)
    V = 11
    eps = 0.1
    logits = torch.randn(5, V)
    labels = torch.randint(0, V, (5,))
    got = ops.cross_entropy(logits, labels, eps)
    lsm = torch.log_softmax(logits, -1)
    want = (1 - eps) * (-lsm.gather(1, labels[:, None]).squeeze(1)) \
        + eps * (-lsm.mean(1))
    assert torch.allclose(got, want, atol=1e-6)


def test_cross_entropy_mask():
    logits = torch.randn(4, 9)
    labels = torch.randint(0, 9, (4,))
    mask = torch.tensor([1.0, 0.0, 1.0, 0.0])
    got = ops.cross_entropy(logits, labels, 0.0, mask)
    assert got[1] == 0 and got[3] == 0


def test_attention_reference_matches_manual():
    torch.manual_seed(2)
    B, H, Tq, Tk, D = 2, 3, 4, 5, 8
    q = torch.randn(B, H, Tq, D)
    k = torch.randn(B, H, Tk, D)
    v = torch.randn(B, H, Tk, D)
    out = ops.attention(q, k, v)
    scores = q @ k.transpose(-1, -2) / math.sqrt(D)
    want = torch.softmax(scores, -1) @ v
    assert torch.allclose(out, want, atol=1e-5)


def test_attention_causal():
    torch.manual_seed(3)
    q = torch.randn(1, 1, 4, 8)
    k = torch.randn(1, 1, 4, 8)
    v = torch.randn(1, 1, 4, 8)
    out = ops.attention(q, k, v, causal=True)
    # causal: row t must not attend to keys > t -> equals truncated attention
    for t in range(4):
        ref = ops.attention(q[:, :, t:t + 1], k[:, :, :t + 1], v[:, :, :t + 1])
        assert torch.allclose(out[:, :, t], ref[:, :, 0], atol=1e-5)


def test_attention_mask_blocks_positions():
    torch.manual_seed(4)
    q = torch.randn(2, 2, 3, 8)
    k = torch.randn(2, 2, 5, 8)
    v = torch.randn(2, 2, 5, 8)
    mask = torch.zeros(2, 1, 1, 5)
    mask[:, :, :, 3:] = -1e9  # block keys 3,4
    out = ops.attention(q, k, v, mask)
    ref = ops.attention(q, k[:, :, :3], v[:, :, :3])
    assert torch.allclose(out, ref, atol=1e-5)


def test_layer_norm_matches_torch():
    torch.manual_seed(5)
    x = torch.randn(6, 32)
    scale = torch.rand(32) + 0.5
    bias = torch.randn(32)
    got = ops.layer_norm(x, scale, bias)
    want = torch.nn.functional.layer_norm(x, (32,), scale, bias, 1e-9)
    assert torch.allclose(got, want, atol=1e-5)


def test_adam_step_matches_torch_adam():
    torch.manual_seed(6)
    n = 64
    p_ref = torch.randn(n)
    p_ours = p_ref.clone()
    g = torch.randn(n)
    m = torch.zeros(n)
    v = torch.zeros(n)
    lr, b1, b2, eps = 1e-3, 0.9, 0.999, 1e-8
    topt = torch.optim.Adam([torch.nn.Parameter(p_ref)], lr=lr,
                            betas=(b1, b2), eps=eps)
    param = list(topt.param_groups[0]["params"])[0]
    param.grad = g.clone()
    topt.step()
    ops.adam_step(p_ours, g, m, v, lr, b1, b2, eps, 0.0, 1)
    assert torch.allclose(param.detach(), p_ours, atol=1e-6)


def test_topk_matches_torch():
    torch.manual_seed(7)
    x = torch.randn(5, 100)
    v1, i1 = ops.topk(x, 4)
    v2, i2 = torch.topk(x, 4, dim=-1)
    assert torch.equal(v1, v2)
    assert torch.equal(i1, i2)


def test_softmax_log_softmax():
    x = torch.randn(4, 33)
    assert torch.allclose(ops.softmax(x), torch.softmax(x, -1), atol=1e-6)
    assert torch.allclose(ops.log_softmax(x), torch.log_softmax(x, -1),
                          atol=1e-6)
