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


def _table_size(S: int, positions, max_pos) -> int:
    """Table rows needed.  When the caller supplies max_pos (the
    config-known max position, as models/llama.py does) we trust it and
    avoid a GPU->CPU `positions.max().item()` sync per attention layer;
    only positions on CPU or with no declared bound pay the reduction."""
    need = max_pos or S
    if positions is None:
        return need
    if max_pos is not None and not positions.is_cuda:
        return max(need, int(positions.max()) + 1)
    if max_pos is None:
        return max(need, int(positions.max().item()) + 1)
    return need


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
    need = _table_size(S, positions, max_pos)
    cos, sin = get_rope_tables(D, need, theta, qkv.device)
    if positions is not None:
        positions = positions.to(torch.int32).contiguous()
    if use_hip(qkv) and D % 16 == 0:
        return _QKVRopeFn.apply(qkv.contiguous(), cos, sin, positions,
                                Hq, Hkv, D)
    if use_hip(qkv):
        # head_dim not 16-aligned: the fused kernel's 8-pair vectorization
        # doesn't apply — split + per-tensor HIP rope instead.
        q, k, v = qkv.split([Hq * D, Hkv * D, Hkv * D], dim=-1)
        q = _RoPEFn.apply(q.reshape(B, S, Hq, D).contiguous(), cos, sin,
                          positions)
        k = _RoPEFn.apply(k.reshape(B, S, Hkv, D).contiguous(), cos, sin,
                          positions)
        return q, k, v.reshape(B, S, Hkv, D).contiguous()
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
    need = _table_size(S, positions, max_pos)
    cos, sin = get_rope_tables(D, need, theta, x.device)
    if positions is not None:
        positions = positions.to(torch.int32).contiguous()
    if use_hip(x):
        return _RoPEFn.apply(x.contiguous(), cos, sin, positions)
    return rope_ref(x, cos, sin, positions)
