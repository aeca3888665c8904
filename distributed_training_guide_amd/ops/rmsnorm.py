"""RMSNorm op: HIP kernel on GPU (rmsnorm.hip), eager fp32 reference on CPU.

Replaces transformers' LlamaRMSNorm in the reference model stack
(/root/reference/04-fully-sharded-data-parallel/train_llm.py:32-44 patches
its reset; every Llama forward invokes it — SURVEY.md §2b).
"""
import torch

from .._ext import ext, use_hip
from .reference import rmsnorm_ref


class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, eps):
        y, rstd = ext().rmsnorm_fwd(x, w, eps)
        ctx.save_for_backward(x, w, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, rstd = ctx.saved_tensors
        dx, dw = ext().rmsnorm_bwd(dy.contiguous(), x, w, rstd)
        return dx, dw, None


def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-5):
    if use_hip(x):
        return _RMSNormFn.apply(x.contiguous(), w, eps)
    return rmsnorm_ref(x, w, eps)


class RMSNorm(torch.nn.Module):
    def __init__(self, hidden_size: int, eps: float = 1e-5,
                 device=None, dtype=None):
        super().__init__()
        self.weight = torch.nn.Parameter(
            torch.empty(hidden_size, device=device, dtype=dtype))
        self.eps = eps
        self.reset_parameters()

    def reset_parameters(self):
        with torch.no_grad():
            if not self.weight.is_meta:
                self.weight.fill_(1.0)

    def forward(self, x):
        return rmsnorm(x, self.weight, self.eps)

    def extra_repr(self):
        return f"{self.weight.shape[0]}, eps={self.eps}"
