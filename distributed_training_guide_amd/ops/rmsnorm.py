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


class _AddRMSNormFn(torch.autograd.Function):
    """Fused residual-add + RMSNorm (add_rmsnorm.hip): res_out = res+delta,
    y = rmsnorm(res_out)*w in one pass; backward folds the downstream
    residual gradient into dx (no separate elementwise adds)."""

    @staticmethod
    def forward(ctx, res, delta, w, eps):
        y, res_out, rstd = ext().add_rmsnorm_fwd(res, delta, w, eps)
        ctx.save_for_backward(res_out, w, rstd)
        return y, res_out

    @staticmethod
    def backward(ctx, dy, dres_out):
        res_out, w, rstd = ctx.saved_tensors
        if dres_out is None:
            dres_out = torch.zeros_like(res_out)
        dx, dw = ext().add_rmsnorm_bwd(dy.contiguous(),
                                       dres_out.contiguous(), res_out, w,
                                       rstd)
        # d_res == d_delta == dx (the add distributes the gradient)
        return dx, dx, dw, None


def add_rmsnorm(res: torch.Tensor, delta: torch.Tensor, w: torch.Tensor,
                eps: float = 1e-5):
    """Returns (rmsnorm(res+delta)*w, res+delta)."""
    H = res.shape[-1]
    if use_hip(res, delta) and H % 8 == 0 and H <= 16384:
        return _AddRMSNormFn.apply(res.contiguous(), delta.contiguous(), w,
                                   eps)
    s = res + delta
    return rmsnorm_ref(s, w, eps) if not use_hip(res, delta) \
        else rmsnorm(s, w, eps), s
