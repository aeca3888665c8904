"""Causal flash attention op (BSHD layout, GQA).

HIP MFMA kernels on GPU (attention_fwd.hip / attention_bwd.hip) — the
replacement for the reference's flash-attn-2 dependency
(attn_implementation="flash_attention_2",
/root/reference/05-training-llama-405b/train_llm.py:87-94).  CPU path uses
torch SDPA in fp32 (differentiable, reference numerics).
"""
import math

import torch

from .._ext import ext, use_hip
from .reference import attention_ref


class _FlashAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale):
        o, lse = ext().attn_fwd(q, k, v, scale)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, dout):
        q, k, v, o, lse = ctx.saved_tensors
        dq, dk, dv = ext().attn_bwd(dout.contiguous(), q, k, v, o, lse,
                                    ctx.scale)
        return dq, dk, dv, None


def flash_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    scale: float | None = None) -> torch.Tensor:
    """q [B,S,Hq,D], k/v [B,S,Hkv,D] -> o [B,S,Hq,D]; always causal."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if use_hip(q, k, v):
        return _FlashAttnFn.apply(q.contiguous(), k.contiguous(),
                                  v.contiguous(), scale)
    return attention_ref(q, k, v, scale)
