"""CPU sanity tests of the eager fp32 reference ops (the ground truth the
HIP kernels are compared against in test_ops_gpu.py)."""
import math

import torch
import torch.nn.functional as F

from distributed_training_guide_amd.ops import reference as R


def test_rmsnorm_ref_matches_manual():
    torch.manual_seed(0)
    x = torch.randn(4, 7, 64)
    w = torch.randn(64)
    y = R.rmsnorm_ref(x, w, 1e-5)
    expected = x / (x.pow(2).mean(-1, keepdim=True) + 1e-5).sqrt() * w
    assert torch.allclose(y, expected, atol=1e-5)


def test_rope_ref_inverse():
    torch.manual_seed(0)
    x = torch.randn(2, 16, 4, 32)
    cos, sin = R.rope_tables(32, 16, 10000.0)
    y = R.rope_ref(x, cos, sin)
    back = R.rope_ref(y, cos, sin, backward=True)
    assert torch.allclose(back, x, atol=1e-5)
    # norm-preserving rotation
    assert torch.allclose(y.norm(), x.norm(), atol=1e-4)


def test_rope_ref_positions():
    torch.manual_seed(0)
    x = torch.randn(1, 8, 2, 16)
    cos, sin = R.rope_tables(16, 64, 10000.0)
    pos = torch.arange(8, 16, dtype=torch.int32)
    y_off = R.rope_ref(x, cos, sin, positions=pos)
    x_pad = torch.cat([torch.randn(1, 8, 2, 16), x], dim=1)
    y_full = R.rope_ref(x_pad, cos, sin)
    assert torch.allclose(y_off, y_full[:, 8:], atol=1e-5)


def test_silu_mul_ref():
    torch.manual_seed(0)
    gu = torch.randn(3, 5, 32)
    y = R.silu_mul_ref(gu)
    g, u = gu.chunk(2, dim=-1)
    assert torch.allclose(y, F.silu(g) * u, atol=1e-6)


def test_attention_ref_vs_manual():
    torch.manual_seed(0)
    B, S, H, D = 2, 32, 4, 16
    q = torch.randn(B, S, H, D)
    k = torch.randn(B, S, H, D)
    v = torch.randn(B, S, H, D)
    scale = 1 / math.sqrt(D)
    o = R.attention_ref(q, k, v, scale)
    # manual causal softmax
    qt = q.permute(0, 2, 1, 3)
    kt = k.permute(0, 2, 1, 3)
    vt = v.permute(0, 2, 1, 3)
    s = qt @ kt.transpose(-1, -2) * scale
    mask = torch.triu(torch.ones(S, S, dtype=torch.bool), 1)
    s = s.masked_fill(mask, float("-inf"))
    expected = (s.softmax(-1) @ vt).permute(0, 2, 1, 3)
    assert torch.allclose(o, expected, atol=1e-5)


def test_attention_ref_gqa():
    torch.manual_seed(0)
    B, S, Hq, Hkv, D = 1, 16, 8, 2, 16
    q = torch.randn(B, S, Hq, D)
    k = torch.randn(B, S, Hkv, D)
    v = torch.randn(B, S, Hkv, D)
    o = R.attention_ref(q, k, v, 1 / math.sqrt(D))
    # expanding kv to Hq must give the same result
    k2 = k.repeat_interleave(Hq // Hkv, dim=2)
    v2 = v.repeat_interleave(Hq // Hkv, dim=2)
    o2 = R.attention_ref(q, k2, v2, 1 / math.sqrt(D))
    assert torch.allclose(o, o2, atol=1e-6)


def test_cross_entropy_ref_shift():
    torch.manual_seed(0)
    B, S, V = 2, 8, 32
    logits = torch.randn(B, S, V)
    labels = torch.randint(0, V, (B, S))
    loss = R.cross_entropy_ref(logits, labels)
    expected = F.cross_entropy(logits[:, :-1].reshape(-1, V),
                               labels[:, 1:].reshape(-1))
    assert torch.allclose(loss, expected, atol=1e-6)


def test_cross_entropy_ref_ignore():
    torch.manual_seed(0)
    logits = torch.randn(1, 6, 16)
    labels = torch.randint(0, 16, (1, 6))
    labels[0, 3] = -100
    loss = R.cross_entropy_ref(logits, labels)
    assert torch.isfinite(loss)


def test_add_rmsnorm_cpu_fallback():
    """Fused residual-add + rmsnorm CPU path == manual add + rmsnorm."""
    import torch

    from distributed_training_guide_amd.ops.rmsnorm import add_rmsnorm
    from distributed_training_guide_amd.ops.reference import rmsnorm_ref

    torch.manual_seed(0)
    res = torch.randn(2, 8, 64)
    delta = torch.randn(2, 8, 64)
    w = torch.randn(64)
    y, res_out = add_rmsnorm(res, delta, w, 1e-5)
    assert torch.allclose(res_out, res + delta)
    assert torch.allclose(y, rmsnorm_ref(res + delta, w, 1e-5), atol=1e-5)


def test_shifted_flat_labels():
    """Causal shift folded into a flat label vector: out[b*S+s] =
    labels[b,s+1], last position of each row = ignore_index."""
    import torch

    from distributed_training_guide_amd.ops.cross_entropy import IGNORE_INDEX
    from distributed_training_guide_amd.ops.fused_linear_ce import \
        _shifted_flat_labels

    labels = torch.arange(12).view(3, 4)
    out = _shifted_flat_labels(labels).view(3, 4)
    for b in range(3):
        for s in range(3):
            assert out[b, s] == labels[b, s + 1]
        assert out[b, 3] == IGNORE_INDEX
