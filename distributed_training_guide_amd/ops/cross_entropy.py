"""Fused causal-LM cross-entropy (shifted labels, mean over valid tokens).

HIP kernel on GPU (cross_entropy.hip): single online-logsumexp pass over the
bf16 logits — never materializes an fp32 logits copy, unlike the reference's
transformers loss path (SURVEY.md §2b "Causal-LM cross-entropy").

Also provides the vocab-sharded (loss-parallel) variant used by the TP
chapter (/root/reference/06-tensor-parallel/README.md:243-271): each rank
computes local (max, sumexp, gathered-logit) per token, the three get
all-reduced over the TP group, and forward/backward finish locally.
"""
import torch
import torch.distributed as dist

from .._ext import ext, use_hip
from .reference import cross_entropy_ref

IGNORE_INDEX = -100


class _CausalCEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels):
        B, S, V = logits.shape
        S_out = S - 1
        loss_rows, lse = ext().ce_fwd(logits, labels, S_out, IGNORE_INDEX)
        # valid-token count stays a DEVICE scalar: no GPU->CPU sync in the
        # hot loop (and the backward scale is computed on device too)
        n_valid = (labels[:, 1:] != IGNORE_INDEX).sum().clamp_min(1)
        ctx.save_for_backward(logits, labels, lse, n_valid)
        ctx.S_out = S_out
        return loss_rows.sum() / n_valid

    @staticmethod
    def backward(ctx, dloss):
        logits, labels, lse, n_valid = ctx.saved_tensors
        scale_t = (dloss.detach() / n_valid).to(torch.float32).reshape(1)
        dlogits = ext().ce_bwd(logits, labels, lse, 0.0, ctx.S_out, 0,
                               IGNORE_INDEX, False, scale_t.contiguous())
        return dlogits, None


def causal_lm_loss(logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """logits [B,S,V] (V padded to a multiple of 8), labels [B,S] int64."""
    if use_hip(logits):
        return _CausalCEFn.apply(logits.contiguous(), labels.contiguous())
    return cross_entropy_ref(logits, labels, IGNORE_INDEX)


class _ShardedCausalCEFn(torch.autograd.Function):
    """Loss-parallel CE over vocab-sharded logits [B,S,V/tp]."""

    @staticmethod
    def forward(ctx, logits, labels, vocab_start, group):
        B, S, Vl = logits.shape
        S_out = S - 1
        mx, sm, gathered = ext().ce_fwd_sharded(logits, labels, S_out,
                                                vocab_start, IGNORE_INDEX)
        # combine across shards: global max, rescaled sum, gathered logit
        gmax = mx.clone()
        dist.all_reduce(gmax, op=dist.ReduceOp.MAX, group=group)
        sm = sm * torch.exp(mx - gmax)
        dist.all_reduce(sm, op=dist.ReduceOp.SUM, group=group)
        # `gathered` is -inf except on the shard owning the label; MAX picks it
        dist.all_reduce(gathered, op=dist.ReduceOp.MAX, group=group)
        lse = gmax + torch.log(sm)
        shifted = labels[:, 1:]
        valid = shifted != IGNORE_INDEX
        n_valid = valid.sum().clamp_min(1)
        loss_rows = torch.where(valid.reshape(-1), lse - gathered,
                                torch.zeros_like(lse))
        ctx.save_for_backward(logits, labels, lse, n_valid)
        ctx.S_out = S_out
        ctx.vocab_start = vocab_start
        return loss_rows.sum() / n_valid

    @staticmethod
    def backward(ctx, dloss):
        logits, labels, lse, n_valid = ctx.saved_tensors
        scale_t = (dloss.detach() / n_valid).to(torch.float32).reshape(1)
        dlogits = ext().ce_bwd(logits, labels, lse, 0.0, ctx.S_out,
                               ctx.vocab_start, IGNORE_INDEX, True,
                               scale_t.contiguous())
        return dlogits, None, None, None


def sharded_causal_lm_loss(logits: torch.Tensor, labels: torch.Tensor,
                           vocab_start: int, group=None) -> torch.Tensor:
    """Loss-parallel CE: logits [B,S,V_local] are this rank's vocab shard
    starting at `vocab_start`; labels are full [B,S]."""
    if logits.is_cuda:
        return _ShardedCausalCEFn.apply(logits.contiguous(),
                                        labels.contiguous(), vocab_start,
                                        group)
    # CPU (gloo) reference path for tests: all-gather the shards and use the
    # eager loss. Differentiable through all_gather is not needed on CPU
    # tests of the loss value; gradient path tested on GPU.
    world = dist.get_world_size(group)
    shards = [torch.empty_like(logits) for _ in range(world)]
    dist.all_gather(shards, logits.contiguous(), group=group)
    rank = dist.get_rank(group)
    shards[rank] = logits  # keep autograd edge to the local shard
    full = torch.cat(shards, dim=-1)
    return cross_entropy_ref(full, labels, IGNORE_INDEX)
