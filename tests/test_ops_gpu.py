"""GPU numerics tests: every HIP kernel vs its plain PyTorch fp32 reference
(SURVEY.md §4 test strategy — bf16 tolerances)."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu

from distributed_training_guide_amd.ops import reference as R
from distributed_training_guide_amd import ops


def _rel_err(a, b):
    a = a.float()
    b = b.float()
    return ((a - b).norm() / b.norm().clamp_min(1e-12)).item()


# ---------------- rmsnorm ----------------
@pytest.mark.parametrize("shape", [(4, 128, 4096), (2, 33, 2048), (1, 7, 768)])
def test_rmsnorm_fwd_bwd(shape):
    torch.manual_seed(0)
    x = torch.randn(*shape, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    w = torch.randn(shape[-1], device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    y = ops.rmsnorm(x, w, 1e-5)
    xr = x.detach().float().requires_grad_(True)
    wr = w.detach().float().requires_grad_(True)
    yr = R.rmsnorm_ref(xr, wr, 1e-5)
    assert _rel_err(y, yr) < 2e-2
    dy = torch.randn_like(y)
    y.backward(dy)
    yr.backward(dy.float())
    assert _rel_err(x.grad, xr.grad) < 3e-2
    assert _rel_err(w.grad, wr.grad) < 3e-2


# ---------------- rope ----------------
@pytest.mark.parametrize("D", [64, 128])
def test_rope_fwd_bwd(D):
    torch.manual_seed(0)
    B, S, H = 2, 96, 4
    x = torch.randn(B, S, H, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    y = ops.rope(x, theta=10000.0)
    cos, sin = R.rope_tables(D, S, 10000.0, device="cuda")
    yr = R.rope_ref(x.detach().float(), cos, sin)
    assert _rel_err(y, yr) < 1e-2
    dy = torch.randn_like(y)
    y.backward(dy)
    # rope backward = inverse rotation of dy
    gr = R.rope_ref(dy.float(), cos, sin, backward=True)
    assert _rel_err(x.grad, gr) < 1e-2


def test_rope_positions():
    torch.manual_seed(0)
    D = 64
    x = torch.randn(1, 16, 2, D, device="cuda", dtype=torch.bfloat16)
    pos = torch.arange(100, 116, dtype=torch.int32, device="cuda")
    y = ops.rope(x, positions=pos, max_pos=256)
    cos, sin = R.rope_tables(D, 256, 10000.0, device="cuda")
    yr = R.rope_ref(x.float(), cos, sin, positions=pos)
    assert _rel_err(y, yr) < 1e-2


# ---------------- silu_mul ----------------
def test_silu_mul_fwd_bwd():
    torch.manual_seed(0)
    gu = torch.randn(8, 64, 2 * 1408, device="cuda", dtype=torch.bfloat16,
                     requires_grad=True)
    y = ops.silu_mul(gu)
    gur = gu.detach().float().requires_grad_(True)
    yr = R.silu_mul_ref(gur)
    assert _rel_err(y, yr) < 2e-2
    dy = torch.randn_like(y)
    y.backward(dy)
    yr.backward(dy.float())
    assert _rel_err(gu.grad, gur.grad) < 3e-2


# ---------------- attention ----------------
@pytest.mark.parametrize("cfg", [
    dict(B=2, S=128, Hq=4, Hkv=4, D=64),
    dict(B=2, S=256, Hq=8, Hkv=2, D=128),
    dict(B=1, S=1024, Hq=4, Hkv=1, D=128),
    dict(B=1, S=200, Hq=2, Hkv=2, D=128),  # ragged S
    dict(B=1, S=48, Hq=2, Hkv=2, D=64),    # S < tile
    dict(B=1, S=4096, Hq=4, Hkv=1, D=128),  # chapter-5 seq length
])
def test_attention_fwd_bwd(cfg):
    torch.manual_seed(0)
    B, S, Hq, Hkv, D = cfg["B"], cfg["S"], cfg["Hq"], cfg["Hkv"], cfg["D"]
    q = torch.randn(B, S, Hq, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    v = torch.randn(B, S, Hkv, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    scale = 1 / math.sqrt(D)
    o = ops.flash_attention(q, k, v, scale)
    qr = q.detach().float().requires_grad_(True)
    kr = k.detach().float().requires_grad_(True)
    vr = v.detach().float().requires_grad_(True)
    orf = R.attention_ref(qr, kr, vr, scale)
    assert _rel_err(o, orf) < 3e-2, f"fwd rel err {_rel_err(o, orf)}"
    do = torch.randn_like(o)
    o.backward(do)
    orf.backward(do.float())
    assert _rel_err(q.grad, qr.grad) < 5e-2, f"dq {_rel_err(q.grad, qr.grad)}"
    assert _rel_err(k.grad, kr.grad) < 5e-2, f"dk {_rel_err(k.grad, kr.grad)}"
    assert _rel_err(v.grad, vr.grad) < 5e-2, f"dv {_rel_err(v.grad, vr.grad)}"


# ---------------- cross entropy ----------------
@pytest.mark.parametrize("V", [512, 128256])
def test_cross_entropy_fwd_bwd(V):
    torch.manual_seed(0)
    B, S = 2, 64
    logits = torch.randn(B, S, V, device="cuda", dtype=torch.bfloat16,
                         requires_grad=True)
    labels = torch.randint(0, V, (B, S), device="cuda")
    labels[0, 5] = -100
    loss = ops.causal_lm_loss(logits, labels)
    lr = logits.detach().float().requires_grad_(True)
    loss_ref = R.cross_entropy_ref(lr, labels)
    assert abs(loss.item() - loss_ref.item()) / loss_ref.item() < 1e-2
    loss.backward()
    loss_ref.backward()
    assert _rel_err(logits.grad, lr.grad) < 3e-2


# ---------------- adamw ----------------
def test_fused_adamw_matches_torch():
    torch.manual_seed(0)
    shapes = [(1024,), (333, 55), (4096, 64)]
    params_hip = [torch.randn(*s, device="cuda", dtype=torch.bfloat16,
                              requires_grad=True) for s in shapes]
    params_ref = [p.detach().float().clone().requires_grad_(True)
                  for p in params_hip]
    opt_hip = ops.FusedAdamW(params_hip, lr=1e-2, weight_decay=0.1)
    opt_ref = torch.optim.AdamW(params_ref, lr=1e-2, weight_decay=0.1)
    for step in range(5):
        g = [torch.randn_like(p, dtype=torch.float32) for p in params_ref]
        for p, pr, gr in zip(params_hip, params_ref, g):
            p.grad = gr.to(torch.bfloat16)
            pr.grad = gr.clone()
        opt_hip.step()
        opt_ref.step()
    for p, pr in zip(params_hip, params_ref):
        assert _rel_err(p, pr) < 2e-2


def test_fused_adamw_fp32_exact():
    torch.manual_seed(1)
    p_hip = torch.randn(2048, 128, device="cuda", requires_grad=True)
    p_ref = p_hip.detach().clone().requires_grad_(True)
    opt_hip = ops.FusedAdamW([p_hip], lr=3e-3, weight_decay=0.05)
    opt_ref = torch.optim.AdamW([p_ref], lr=3e-3, weight_decay=0.05)
    for _ in range(3):
        g = torch.randn_like(p_ref)
        p_hip.grad = g.clone()
        p_ref.grad = g.clone()
        opt_hip.step()
        opt_ref.step()
    assert _rel_err(p_hip, p_ref) < 1e-5


# ---------------- model-level smoke ----------------
def test_model_train_step_gpu():
    from distributed_training_guide_amd.models import build_model

    torch.manual_seed(0)
    model = build_model("llama-debug", device="cuda", dtype=torch.bfloat16)
    opt = ops.FusedAdamW(model.parameters(), lr=1e-3)
    ids = torch.randint(0, 1024, (2, 128), device="cuda")
    losses = []
    for _ in range(5):
        out = model(input_ids=ids, labels=ids)
        out.loss.backward()
        opt.step()
        opt.zero_grad(set_to_none=True)
        losses.append(out.loss.item())
    assert all(math.isfinite(l) for l in losses)
    assert losses[-1] < losses[0]  # memorizing one batch must reduce loss


def test_qkv_rope_fused_matches_unfused():
    from distributed_training_guide_amd.ops import qkv_rope, rope

    torch.manual_seed(3)
    B, S, Hq, Hkv, D = 2, 256, 8, 2, 64
    W = (Hq + 2 * Hkv) * D
    qkv = torch.randn(B, S, W, device="cuda", dtype=torch.bfloat16,
                      requires_grad=True)
    q, k, v = qkv_rope(qkv, Hq, Hkv, D, theta=1e4)
    # unfused reference path
    qkv2 = qkv.detach().clone().requires_grad_(True)
    q2, k2, v2 = qkv2.split([Hq * D, Hkv * D, Hkv * D], dim=-1)
    q2 = rope(q2.view(B, S, Hq, D), 1e4)
    k2 = rope(k2.view(B, S, Hkv, D), 1e4)
    v2 = v2.view(B, S, Hkv, D).contiguous()
    assert torch.equal(v, v2)
    assert _rel_err(q, q2.float()) < 1e-2
    assert _rel_err(k, k2.float()) < 1e-2
    dq = torch.randn_like(q)
    dk = torch.randn_like(k)
    dv = torch.randn_like(v)
    torch.autograd.backward([q, k, v], [dq, dk, dv])
    torch.autograd.backward([q2, k2, v2], [dq, dk, dv])
    assert _rel_err(qkv.grad, qkv2.grad.float()) < 1e-2
