"""RoPE op: HIP kernel on GPU (rope.hip), eager reference on CPU.

cos/sin tables are precomputed once per (head_dim, theta, device) and grown
on demand; explicit int32 `positions` support mirrors the reference's
position_ids handling under sequence parallelism
(/root/reference/06-tensor-parallel/train_llm.py:210-212).
"""
import torch

from .._ext import ext, use_hip
from .reference import rope_ref, rope_tables

_TABLE_CACHE: dict = {}


def get_rope_tables(dim: int, max_pos: int, theta: float, device):
    key = (dim, float(theta), str(device))
    entry = _TABLE_CACHE.get(key)
    if entry is None or entry[0].shape[0] < max_pos:
        size = max(max_pos, 1024)
        cos, sin = rope_tables(dim, size, theta, device=device)
        _TABLE_CACHE[key] = (cos, sin)
        entry = _TABLE_CACHE[key]
    return entry


class _RoPEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos, sin, positions):
        y = ext().rope(x, cos, sin, positions, False)
        ctx.cos, ctx.sin, ctx.positions = cos, sin, positions
        return y

    @staticmethod
    def backward(ctx, dy):
        dx = ext().rope(dy.contiguous(), ctx.cos, ctx.sin, ctx.positions, True)
        return dx, None, None, None


class _QKVRopeFn(torch.autograd.Function):
    """Fused qkv split + RoPE (qkv_rope.hip): one pass produces rotated
    contiguous q/k plus v from the packed projection output; backward
    re-packs dq/dk/dv into dqkv with the inverse rotation (replacing the
    split -> rope -> contiguous chain and the grad cat)."""

    @staticmethod
    def forward(ctx, qkv, cos, sin, positions, Hq, Hkv, D):
        q, k, v = ext().qkv_rope_fwd(qkv, cos, sin, positions, Hq, Hkv, D)
        ctx.cos, ctx.sin, ctx.positions = cos, sin, positions
        return q, k, v

    @staticmethod
    def backward(ctx, dq, dk, dv):
        dqkv = ext().qkv_rope_bwd(dq.contiguous(), dk.contiguous(),
                                  dv.contiguous(), ctx.cos, ctx.sin,
                                  ctx.positions)
        return dqkv, None, None, None, None, None, None


def qkv_rope(qkv: torch.Tensor, Hq: int, Hkv: int, D: int,
             theta: float = 10000.0,
             positions: torch.Tensor | None = None,
             max_pos: int | None = None):
    """Split packed qkv [B,S,(Hq+2Hkv)*D] and apply RoPE to q/k in one
    fused pass. Returns (q [B,S,Hq,D], k, v [B,S,Hkv,D])."""
    B, S, W = qkv.shape
    need = max_pos or S
    if positions is not None:
        positions = positions.to(torch.int32).contiguous()
        need = max(need, int(positions.max().item()) + 1)
    cos, sin = get_rope_tables(D, need, theta, qkv.device)
    if use_hip(qkv):
        return _QKVRopeFn.apply(qkv.contiguous(), cos, sin, positions,
                                Hq, Hkv, D)
    # CPU fallback: split views + eager rope
    q, k, v = qkv.split([Hq * D, Hkv * D, Hkv * D], dim=-1)
    q = rope_ref(q.view(B, S, Hq, D), cos, sin, positions)
    k = rope_ref(k.view(B, S, Hkv, D), cos, sin, positions)
    return q, k, v.view(B, S, Hkv, D).contiguous()


def rope(x: torch.Tensor, theta: float = 10000.0,
         positions: torch.Tensor | None = None,
         max_pos: int | None = None) -> torch.Tensor:
    """Apply rotate-half RoPE to x [B, S, H, D]."""
    B, S, H, D = x.shape
    need = max_pos or S
    if positions is not None:
        positions = positions.to(torch.int32).contiguous()
        need = max(need, int(positions.max().item()) + 1)
    cos, sin = get_rope_tables(D, need, theta, x.device)
    if use_hip(x):
        return _RoPEFn.apply(x.contiguous(), cos, sin, positions)
    return rope_ref(x, cos, sin, positions)
