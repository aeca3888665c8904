"""Fused SwiGLU gate op: y = silu(gu[..., :I]) * gu[..., I:].

HIP kernel on GPU (silu_mul.hip); the packed gate_up layout lets the Llama
MLP run gate+up as ONE GEMM (SURVEY.md §2b "SiLU-gated MLP").
"""
import torch

from .._ext import ext, use_hip
from .reference import silu_mul_ref


class _SiluMulFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gu):
        y = ext().silu_mul_fwd(gu)
        ctx.save_for_backward(gu)
        return y

    @staticmethod
    def backward(ctx, dy):
        (gu,) = ctx.saved_tensors
        return ext().silu_mul_bwd(dy.contiguous(), gu)


def silu_mul(gu: torch.Tensor) -> torch.Tensor:
    if use_hip(gu):
        return _SiluMulFn.apply(gu.contiguous())
    return silu_mul_ref(gu)
