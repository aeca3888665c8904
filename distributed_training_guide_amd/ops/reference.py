"""Eager fp32 reference implementations of every HIP op.

These are the ground truth the HIP kernels are tested against
(tests/test_ops_gpu.py) and the CPU execution path for the chapters that run
without a GPU (01-single-gpu plumbing, CPU unit tests).  Pure PyTorch, fp32
math, no fused tricks — deliberately boring.
"""
import torch
import torch.nn.functional as F


def rmsnorm_ref(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    rstd = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * rstd * w.float()).to(x.dtype)


def rope_tables(dim: int, max_pos: int, theta: float,
                device=None) -> tuple[torch.Tensor, torch.Tensor]:
    """cos/sin tables [max_pos, dim/2] in fp32 (host-precomputed — on-device
    trig per element would be VALU-bound; guide App. B)."""
    inv_freq = 1.0 / (theta ** (torch.arange(0, dim, 2, device=device,
                                             dtype=torch.float32) / dim))
    pos = torch.arange(max_pos, device=device, dtype=torch.float32)
    freqs = torch.outer(pos, inv_freq)  # [max_pos, dim/2]
    return freqs.cos().contiguous(), freqs.sin().contiguous()


def rope_ref(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
             positions: torch.Tensor | None = None,
             backward: bool = False) -> torch.Tensor:
    """Rotate-half RoPE on [B, S, H, D] (pairs (i, i + D/2))."""
    B, S, H, D = x.shape
    if positions is None:
        c = cos[:S]
        s = sin[:S]
    else:
        c = cos[positions.long()]
        s = sin[positions.long()]
    c = c.view(1, S, 1, D // 2).float()
    s = s.view(1, S, 1, D // 2).float()
    x0 = x[..., : D // 2].float()
    x1 = x[..., D // 2:].float()
    if backward:
        y0 = x0 * c + x1 * s
        y1 = -x0 * s + x1 * c
    else:
        y0 = x0 * c - x1 * s
        y1 = x0 * s + x1 * c
    return torch.cat([y0, y1], dim=-1).to(x.dtype)


def silu_mul_ref(gu: torch.Tensor) -> torch.Tensor:
    I = gu.shape[-1] // 2
    g = gu[..., :I].float()
    u = gu[..., I:].float()
    return (F.silu(g) * u).to(gu.dtype)


def attention_ref(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                  scale: float) -> torch.Tensor:
    """Causal GQA SDPA on [B, S, H, D] inputs, fp32 math."""
    B, S, Hq, D = q.shape
    Hkv = k.shape[2]
    group = Hq // Hkv
    qt = q.permute(0, 2, 1, 3).float()          # [B, Hq, S, D]
    kt = k.permute(0, 2, 1, 3).float()
    vt = v.permute(0, 2, 1, 3).float()
    if group > 1:
        kt = kt.repeat_interleave(group, dim=1)
        vt = vt.repeat_interleave(group, dim=1)
    o = F.scaled_dot_product_attention(qt, kt, vt, is_causal=True, scale=scale)
    return o.permute(0, 2, 1, 3).to(q.dtype)


def cross_entropy_ref(logits: torch.Tensor, labels: torch.Tensor,
                      ignore_index: int = -100) -> torch.Tensor:
    """Shifted causal-LM CE, mean over valid tokens (fp32)."""
    shift_logits = logits[..., :-1, :].float().contiguous()
    shift_labels = labels[..., 1:].contiguous()
    return F.cross_entropy(
        shift_logits.view(-1, shift_logits.size(-1)),
        shift_labels.view(-1),
        ignore_index=ignore_index,
    )
