"""LayerNorm + tanh-GELU ops: HIP kernels on GPU (layernorm.hip), eager
fp32 reference on CPU — the GPT-2 block's norm/activation, so chapter-1's
smoke model (reference 01-single-gpu/README.md:9-12 trains HF gpt2) runs
fully on the in-repo HIP path like the Llama chapters."""
import torch
import torch.nn as nn
import torch.nn.functional as F

from .._ext import ext, use_hip


class _LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, eps):
        y, mu, rstd = ext().layernorm_fwd(x, w, b, eps)
        ctx.save_for_backward(x, w, mu, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, mu, rstd = ctx.saved_tensors
        dx, dw, db = ext().layernorm_bwd(dy.contiguous(), x, w, mu, rstd)
        return dx, dw, db, None


class LayerNorm(nn.Module):
    """Drop-in for nn.LayerNorm (weight+bias over the last dim) on the
    gfx950 kernel; fp32 eager math on CPU."""

    def __init__(self, hidden: int, eps: float = 1e-5, device=None,
                 dtype=None):
        super().__init__()
        self.eps = eps
        self.normalized_shape = (hidden,)
        self.weight = nn.Parameter(torch.ones(hidden, device=device,
                                              dtype=dtype))
        self.bias = nn.Parameter(torch.zeros(hidden, device=device,
                                             dtype=dtype))

    def reset_parameters(self):
        with torch.no_grad():
            if not self.weight.is_meta:
                self.weight.fill_(1.0)
                self.bias.zero_()

    def forward(self, x):
        if use_hip(x):
            return _LayerNormFn.apply(x.contiguous(), self.weight,
                                      self.bias, self.eps)
        xf = x.float()
        y = F.layer_norm(xf, self.normalized_shape, self.weight.float(),
                         self.bias.float(), self.eps)
        return y.to(x.dtype)

    def extra_repr(self):
        return f"{self.normalized_shape}, eps={self.eps}"


class _GeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.save_for_backward(x)
        return ext().gelu_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        return ext().gelu_bwd(dy.contiguous(), x)


def gelu(x: torch.Tensor) -> torch.Tensor:
    """tanh-approximation GELU (what HF gpt2 uses)."""
    if use_hip(x) and x.numel() % 8 == 0:
        return _GeluFn.apply(x.contiguous())
    return F.gelu(x, approximate="tanh")
