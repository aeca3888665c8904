"""Chunked fused lm_head projection + causal-LM cross-entropy.

The reference's biggest single activation is the [B,S,V] logits tensor
(/root/reference/06-tensor-parallel/README.md:243-271 motivates
loss-parallel for exactly this); at Llama-3 vocab (128256) and bs24/s1024
that is ~6.3 GB bf16 forward + ~6.3 GB gradient.  Here the final
projection and the CE loss are computed in row chunks: each chunk's logits
live only long enough to feed the online-LSE CE kernel (forward) or the
softmax-grad + two GEMMs (backward, logits recomputed), so the full logits
tensor is NEVER materialized.  Costs one extra lm_head GEMM in backward
(~2%% of step at bs24); saves ~12 GB peak at bs24.

The causal shift is folded into a precomputed flat shifted-label vector
(position s predicts labels[b,s+1]; the last position of every row becomes
ignore_index), which lets chunks be plain contiguous row ranges of the
flattened [B*S, H] input — chunk GEMMs need no copies and batch-row
boundaries need no special casing.

dW is accumulated across chunks in fp32 and cast to the weight dtype once.
"""
import torch

from .._ext import ext
from .cross_entropy import IGNORE_INDEX
from .reference import cross_entropy_ref


def _shifted_flat_labels(labels: torch.Tensor) -> torch.Tensor:
    """[B,S] labels -> flat [B*S] where out[b*S+s] = labels[b,s+1], and the
    last position of each row is ignore_index (never a loss row)."""
    B, S = labels.shape
    flat = labels.reshape(-1)
    out = torch.empty_like(flat)
    out[: B * S - 1] = flat[1:]
    out[B * S - 1] = IGNORE_INDEX
    out.view(B, S)[:, S - 1] = IGNORE_INDEX
    return out.contiguous()


class _FusedLinearCEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, labels, chunk_rows):
        B, S, H = x.shape
        V = weight.shape[0]
        x2 = x.reshape(B * S, H)
        N = B * S
        shifted = _shifted_flat_labels(labels)
        n_valid = max(int((shifted != IGNORE_INDEX).sum().item()), 1)
        lse = torch.empty(N, dtype=torch.float32, device=x.device)
        loss_sum = torch.zeros((), dtype=torch.float32, device=x.device)
        wt = weight.t()
        for r0 in range(0, N, chunk_rows):
            r1 = min(r0 + chunk_rows, N)
            logits_c = torch.matmul(x2[r0:r1], wt)  # [L,V] bf16, transient
            lc, lse_c = ext().ce_fwd(logits_c.unsqueeze(0), shifted[r0:r1],
                                     r1 - r0, IGNORE_INDEX)
            loss_sum += lc.sum()
            lse[r0:r1] = lse_c
        ctx.save_for_backward(x2, weight, shifted, lse)
        ctx.n_valid = n_valid
        ctx.chunk_rows = chunk_rows
        ctx.in_shape = (B, S, H)
        return loss_sum / n_valid

    @staticmethod
    def backward(ctx, dloss):
        x2, weight, shifted, lse = ctx.saved_tensors
        B, S, H = ctx.in_shape
        V = weight.shape[0]
        N = B * S
        scale = float(dloss.item()) / ctx.n_valid
        dx2 = torch.empty_like(x2)
        dw32 = torch.zeros(V, H, dtype=torch.float32, device=x2.device)
        wt = weight.t()
        for r0 in range(0, N, ctx.chunk_rows):
            r1 = min(r0 + ctx.chunk_rows, N)
            L = r1 - r0
            logits_c = torch.matmul(x2[r0:r1], wt)  # recompute, transient
            dlogits_c = ext().ce_bwd(logits_c.unsqueeze(0), shifted[r0:r1],
                                     lse[r0:r1], scale, L, 0, IGNORE_INDEX,
                                     False).squeeze(0)
            torch.matmul(dlogits_c, weight, out=dx2[r0:r1])
            dw32.add_(torch.matmul(dlogits_c.t(), x2[r0:r1]))
        return (dx2.view(B, S, H), dw32.to(weight.dtype), None, None)


def fused_causal_lm_loss(x: torch.Tensor, weight: torch.Tensor,
                         labels: torch.Tensor,
                         chunk_rows: int = 8192) -> torch.Tensor:
    """Mean causal-LM CE of linear(x, weight) against shifted labels,
    without materializing [B,S,V] logits.  x [B,S,H] bf16 contiguous,
    weight [V,H] (V padded to a multiple of 8), labels [B,S] int64."""
    if x.is_cuda:
        return _FusedLinearCEFn.apply(x.contiguous(), weight,
                                      labels.contiguous(), chunk_rows)
    # CPU fallback: full logits + eager reference (tests; memory is moot)
    logits = torch.matmul(x, weight.t())
    return cross_entropy_ref(logits, labels, IGNORE_INDEX)
