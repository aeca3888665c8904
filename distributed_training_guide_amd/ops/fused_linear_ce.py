"""Chunked fused lm_head projection + causal-LM cross-entropy.

The reference's biggest single activation is the [B,S,V] logits tensor
(/root/reference/06-tensor-parallel/README.md:243-271 motivates
loss-parallel for exactly this); at Llama-3 vocab (128256) and bs24/s1024
that is ~6.3 GB bf16 forward + ~6.3 GB gradient.  Two fused modes:

* keep_logits=True ("semi"): one full forward GEMM whose logits are saved,
  backward chunks the softmax-grad + dgrad/wgrad GEMMs so the full
  [B,S,V] GRADIENT never exists.  ~free; saves ~6.3 GB at bs24.
* keep_logits=False ("fused"): forward also runs in chunks and saves
  nothing; backward recomputes each logits chunk.  Costs one extra
  lm_head GEMM (~2.5%% of step at bs24); saves ~12.6 GB.

DTGA_CE_MODE selects: auto (default — fused only when free HBM is tight),
semi, fused, unfused (full logits + plain CE, the fastest when memory is
plentiful).

The causal shift is folded into a precomputed flat shifted-label vector
(position s predicts labels[b,s+1]; the last position of every row becomes
ignore_index), which lets chunks be plain contiguous row ranges of the
flattened [B*S, H] input — chunk GEMMs need no copies and batch-row
boundaries need no special casing.

Multi-chunk dW is accumulated in fp32 and cast to the weight dtype once.
"""
import os

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
    # fill_ (a device kernel) rather than scalar assignment (a pageable
    # H2D copy): keeps this hipGraph-capturable
    out[B * S - 1:].fill_(IGNORE_INDEX)
    out.view(B, S)[:, S - 1].fill_(IGNORE_INDEX)
    return out.contiguous()


class _FusedLinearCEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, labels, chunk_rows, keep_logits):
        B, S, H = x.shape
        x2 = x.reshape(B * S, H)
        N = B * S
        shifted = _shifted_flat_labels(labels)
        # device scalar: no GPU->CPU sync per step (see cross_entropy.py)
        n_valid = (shifted != IGNORE_INDEX).sum().clamp_min(1)
        logits = None
        if keep_logits:
            logits = torch.matmul(x2, weight.t())  # [N,V], saved
            loss_rows, lse = ext().ce_fwd(logits.unsqueeze(0), shifted, N,
                                          IGNORE_INDEX)
            loss_sum = loss_rows.sum()
        else:
            lse = torch.empty(N, dtype=torch.float32, device=x.device)
            loss_sum = torch.zeros((), dtype=torch.float32, device=x.device)
            wt = weight.t()
            for r0 in range(0, N, chunk_rows):
                r1 = min(r0 + chunk_rows, N)
                logits_c = torch.matmul(x2[r0:r1], wt)  # transient
                lc, lse_c = ext().ce_fwd(logits_c.unsqueeze(0),
                                         shifted[r0:r1], r1 - r0,
                                         IGNORE_INDEX)
                loss_sum += lc.sum()
                lse[r0:r1] = lse_c
        if keep_logits:
            ctx.save_for_backward(x2, weight, shifted, lse, n_valid, logits)
        else:
            ctx.save_for_backward(x2, weight, shifted, lse, n_valid)
        ctx.chunk_rows = chunk_rows
        ctx.in_shape = (B, S, H)
        ctx.keep_logits = keep_logits
        return loss_sum / n_valid

    @staticmethod
    def backward(ctx, dloss):
        if ctx.keep_logits:
            x2, weight, shifted, lse, n_valid, logits = ctx.saved_tensors
        else:
            x2, weight, shifted, lse, n_valid = ctx.saved_tensors
            logits = None
        B, S, H = ctx.in_shape
        V = weight.shape[0]
        N = B * S
        scale_t = (dloss.detach() / n_valid).to(torch.float32) \
            .reshape(1).contiguous()
        dx2 = torch.empty_like(x2)
        one_chunk = ctx.chunk_rows >= N
        # multi-chunk: accumulate dW in fp32 across chunks; single chunk:
        # one bf16 GEMM, exactly the unfused wgrad
        dw32 = None if one_chunk else torch.zeros(
            V, H, dtype=torch.float32, device=x2.device)
        dw = None
        wt = weight.t()
        for r0 in range(0, N, ctx.chunk_rows):
            r1 = min(r0 + ctx.chunk_rows, N)
            L = r1 - r0
            logits_c = (logits[r0:r1] if logits is not None
                        else torch.matmul(x2[r0:r1], wt))  # recompute
            dlogits_c = ext().ce_bwd(logits_c.unsqueeze(0), shifted[r0:r1],
                                     lse[r0:r1], 0.0, L, 0, IGNORE_INDEX,
                                     False, scale_t).squeeze(0)
            del logits_c
            torch.matmul(dlogits_c, weight, out=dx2[r0:r1])
            if one_chunk:
                dw = torch.matmul(dlogits_c.t(), x2[r0:r1])
            else:
                dw32.add_(torch.matmul(dlogits_c.t(), x2[r0:r1]))
        if dw is None:
            dw = dw32.to(weight.dtype)
        return (dx2.view(B, S, H), dw, None, None, None)


def fused_causal_lm_loss(x: torch.Tensor, weight: torch.Tensor,
                         labels: torch.Tensor,
                         chunk_rows: int | None = None) -> torch.Tensor:
    """Mean causal-LM CE of linear(x, weight) against shifted labels
    without materializing the full [B,S,V] logits gradient (and, under
    memory pressure, without the logits themselves).  x [B,S,H] bf16
    contiguous, weight [V,H] (V padded to a multiple of 8), labels [B,S]
    int64.  DTGA_CE_MODE / DTGA_CE_CHUNK override the policy."""
    if not x.is_cuda:
        # CPU fallback: full logits + eager reference (tests; memory moot)
        logits = torch.matmul(x, weight.t())
        return cross_entropy_ref(logits, labels, IGNORE_INDEX)
    mode = os.environ.get("DTGA_CE_MODE", "auto")
    if mode == "unfused":
        from .cross_entropy import causal_lm_loss

        return causal_lm_loss(torch.matmul(x, weight.t()), labels)
    if chunk_rows is None:
        chunk_rows = int(os.environ.get("DTGA_CE_CHUNK", "8192"))
    if mode == "semi":
        keep = True
    elif mode == "fused":
        keep = False
    else:  # auto: recompute-in-backward only when HBM is actually tight
        logits_bytes = x.numel() // x.shape[-1] * weight.shape[0] * 2
        free, _ = torch.cuda.mem_get_info(x.device)
        keep = free > 4 * logits_bytes
    return _FusedLinearCEFn.apply(x.contiguous(), weight,
                                  labels.contiguous(), chunk_rows, keep)
