"""Tensor + sequence parallelism — hand-placed re-implementation of the
reference's DTensor TP plan (/root/reference/06-tensor-parallel/
train_llm.py:79-121), DTensor-free per SURVEY.md §7 step 6:

  activations travel sequence-sharded ([B, S/tp, h], "Shard(1)") between
  blocks; norms run SequenceParallel on the local shard; entering attention
  and MLP the activation is ALL-GATHERED over seq ("Shard(1)->Replicate",
  06:90-95,102-105); q/k/v and gate/up are column-sharded GEMMs; o_proj and
  down_proj are row-sharded and REDUCE-SCATTER their partial outputs back
  to Shard(1) (06:99,108); the embedding is vocab-parallel with a
  reduce-scatter epilogue (06:79-83 semantics); lm_head is column-sharded
  over vocab with a seq all-gather prologue and either a vocab all-gather
  (Replicate output, 06:114-119) or loss-parallel fused CE
  (06-.../README.md:243-271).

Every boundary collective is an autograd.Function whose backward is the
transpose collective (all-gather <-> reduce-scatter).  Norm weights are
replicated across tp; their per-rank partial grads are summed over the tp
group by a post-accumulate-grad hook (DTensor gave the reference this for
free; here it is explicit).
"""
import logging

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from ..models.llama import CausalLMOutput, LlamaConfig
from ..ops import RMSNorm, flash_attention, rope, silu_mul
from ..ops.cross_entropy import causal_lm_loss, sharded_causal_lm_loss

LOGGER = logging.getLogger(__name__)


# ---------------------------------------------------------------- collectives
#
# Layout strategy (MI355X: every copy is a full-activation HBM round trip,
# 4 boundaries x n_layers x fwd+bwd of them per step): the seq-sharded
# [B, s, H] activation is ALREADY a contiguous rank chunk of the
# [tp, B, s, H] gathered stack, so the all-gather needs NO input copy —
# only one permute copy to interleave ranks back into [B, tp*s, H].  The
# reduce-scatter is its mirror: one permute copy into [tp, B, s, H] rank
# order, and the collective writes the output directly in final layout.
# One copy per boundary, versus two with the [B,S,H]<->[S,B,H] transpose
# pair this replaces.

def _gather_seq_1copy(x, group):
    """[B, s, H] -> [B, tp*s, H] with one permute copy after the gather."""
    world = dist.get_world_size(group)
    xc = x.contiguous()
    B, s, H = xc.shape
    parts = torch.empty((world, B, s, H), dtype=x.dtype, device=x.device)
    if x.is_cuda:
        dist.all_gather_into_tensor(parts, xc, group=group)
    else:
        dist.all_gather(list(parts.unbind(0)), xc, group=group)
    return parts.permute(1, 0, 2, 3).reshape(B, world * s, H)


def _reduce_scatter_seq_1copy(x, group):
    """[B, S, H] partial sums -> [B, S/tp, H] with one permute copy before
    the reduce-scatter; the output lands directly in its final layout."""
    world = dist.get_world_size(group)
    B, S, H = x.shape
    s = S // world
    inp = x.reshape(B, world, s, H).permute(1, 0, 2, 3).contiguous()
    out = torch.empty((B, s, H), dtype=x.dtype, device=x.device)
    if x.is_cuda:
        dist.reduce_scatter_tensor(out, inp, group=group)
    else:  # gloo: all-reduce then take this rank's chunk
        dist.all_reduce(inp, group=group)
        out.copy_(inp[dist.get_rank(group)])
    return out


class _GatherSeq(torch.autograd.Function):
    """[B, S/tp, H] -> [B, S, H]; backward reduce-scatters the grad."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _gather_seq_1copy(x, group)

    @staticmethod
    def backward(ctx, g):
        return _reduce_scatter_seq_1copy(g, ctx.group), None


class _ReduceScatterSeq(torch.autograd.Function):
    """[B, S, H] (partial sums) -> [B, S/tp, H]; backward all-gathers."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _reduce_scatter_seq_1copy(x, group)

    @staticmethod
    def backward(ctx, g):
        return _gather_seq_1copy(g, ctx.group), None


class _GatherVocab(torch.autograd.Function):
    """[B, S, V/tp] -> [B, S, V] (Replicate output of the column-sharded
    lm_head); backward takes the local vocab slice of the grad."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        world = dist.get_world_size(group)
        ctx.vlocal = x.shape[-1]
        xc = x.contiguous()
        parts = torch.empty((world, *xc.shape), dtype=x.dtype,
                            device=x.device)
        if x.is_cuda:
            dist.all_gather_into_tensor(parts, xc, group=group)
        else:
            dist.all_gather(list(parts.unbind(0)), xc, group=group)
        # concat rank chunks on the vocab (last) dim: one permute copy
        perm = (*range(1, x.dim()), 0, x.dim())
        return parts.permute(perm).reshape(*x.shape[:-1],
                                           world * ctx.vlocal)

    @staticmethod
    def backward(ctx, g):
        r = dist.get_rank(ctx.group)
        return g[..., r * ctx.vlocal: (r + 1) * ctx.vlocal].contiguous(), None


# DTGA_FORCE_COLLECTIVES=1 runs the real collective branch even at
# world=1 (RCCL supports single-rank communicators: collectives become
# device copies) — lets a 1-GPU box execute and profile the exact RCCL
# call pattern of the multi-GPU path.
def _force_collectives():
    import os

    return os.environ.get("DTGA_FORCE_COLLECTIVES") == "1"


def gather_seq(x, group):
    if dist.get_world_size(group) == 1 and not _force_collectives():
        return x
    return _GatherSeq.apply(x, group)


def reduce_scatter_seq(x, group):
    if dist.get_world_size(group) == 1 and not _force_collectives():
        return x
    return _ReduceScatterSeq.apply(x, group)


def gather_vocab(x, group):
    if dist.get_world_size(group) == 1 and not _force_collectives():
        return x
    return _GatherVocab.apply(x, group)


# ---------------------------------------------------------------- modules
def _shard_rows(full: torch.Tensor, segments, tp, tp_rank):
    """Slice the rows of `full` ([out, ...]) column-parallel-wise, where the
    out dim is a concatenation of `segments` (packed qkv / gate_up): each
    segment is sharded independently and the local slices concatenated."""
    outs = []
    off = 0
    for seg in segments:
        local = seg // tp
        outs.append(full.narrow(0, off + tp_rank * local, local))
        off += seg
    return torch.cat(outs, dim=0)


class ColwiseLinear(nn.Module):
    """Column-parallel (output-sharded) linear; `segments` handles packed
    projections (q|k|v, gate|up) so each logical segment shards evenly."""

    def __init__(self, in_features, out_features, mesh, segments=None,
                 device=None, dtype=None):
        super().__init__()
        self.tp = mesh.tp_size
        self.segments = segments or [out_features]
        assert all(s % self.tp == 0 for s in self.segments)
        self.out_local = out_features // self.tp
        self.weight = nn.Parameter(torch.empty(
            self.out_local, in_features, device=device, dtype=dtype))

    def load_full_weight(self, full, tp_rank):
        with torch.no_grad():
            self.weight.copy_(_shard_rows(full, self.segments, self.tp,
                                          tp_rank).to(self.weight.dtype))

    def forward(self, x):
        return F.linear(x, self.weight)


class RowwiseLinear(nn.Module):
    """Row-parallel (input-sharded) linear producing partial sums."""

    def __init__(self, in_features, out_features, mesh, device=None,
                 dtype=None):
        super().__init__()
        self.tp = mesh.tp_size
        assert in_features % self.tp == 0
        self.in_local = in_features // self.tp
        self.weight = nn.Parameter(torch.empty(
            out_features, self.in_local, device=device, dtype=dtype))

    def load_full_weight(self, full, tp_rank):
        with torch.no_grad():
            self.weight.copy_(full.narrow(
                1, tp_rank * self.in_local, self.in_local)
                .to(self.weight.dtype))

    def forward(self, x):
        return F.linear(x, self.weight)


class VocabParallelEmbedding(nn.Module):
    def __init__(self, vocab, hidden, mesh, device=None, dtype=None):
        super().__init__()
        self.tp = mesh.tp_size
        assert vocab % self.tp == 0
        self.vlocal = vocab // self.tp
        self.start = mesh.tp_rank * self.vlocal
        self.weight = nn.Parameter(torch.empty(
            self.vlocal, hidden, device=device, dtype=dtype))

    def load_full_weight(self, full, tp_rank):
        with torch.no_grad():
            self.weight.copy_(full.narrow(0, tp_rank * self.vlocal,
                                          self.vlocal).to(self.weight.dtype))

    def forward(self, ids):
        local = ids - self.start
        mask = (local >= 0) & (local < self.vlocal)
        emb = F.embedding(local.clamp(0, self.vlocal - 1), self.weight)
        return emb * mask.unsqueeze(-1).to(emb.dtype)


class TPLlamaAttention(nn.Module):
    def __init__(self, config: LlamaConfig, mesh, device=None, dtype=None):
        super().__init__()
        self.config = config
        tp = mesh.tp_size
        assert config.num_attention_heads % tp == 0
        assert config.num_key_value_heads % tp == 0
        d = config.head_dim
        self.num_heads = config.num_attention_heads // tp
        self.num_kv_heads = config.num_key_value_heads // tp
        self.head_dim = d
        h = config.hidden_size
        self.qkv_proj = ColwiseLinear(
            h, (config.num_attention_heads + 2 * config.num_key_value_heads) * d,
            mesh,
            segments=[config.num_attention_heads * d,
                      config.num_key_value_heads * d,
                      config.num_key_value_heads * d],
            device=device, dtype=dtype)
        self.o_proj = RowwiseLinear(config.num_attention_heads * d, h, mesh,
                                    device=device, dtype=dtype)

    def forward(self, x, position_ids=None):
        B, S, _ = x.shape  # x is REPLICATED (post seq-gather), full S
        d = self.head_dim
        qkv = self.qkv_proj(x)
        q, k, v = qkv.split([self.num_heads * d, self.num_kv_heads * d,
                             self.num_kv_heads * d], dim=-1)
        q = q.view(B, S, self.num_heads, d)
        k = k.view(B, S, self.num_kv_heads, d)
        v = v.view(B, S, self.num_kv_heads, d).contiguous()
        theta = self.config.rope_theta
        maxp = self.config.max_position_embeddings
        q = rope(q, theta, positions=position_ids, max_pos=maxp)
        k = rope(k, theta, positions=position_ids, max_pos=maxp)
        o = flash_attention(q, k, v)
        return self.o_proj(o.reshape(B, S, self.num_heads * d))


class TPLlamaMLP(nn.Module):
    def __init__(self, config: LlamaConfig, mesh, device=None, dtype=None):
        super().__init__()
        h, i = config.hidden_size, config.intermediate_size
        self.gate_up_proj = ColwiseLinear(h, 2 * i, mesh, segments=[i, i],
                                          device=device, dtype=dtype)
        self.down_proj = RowwiseLinear(i, h, mesh, device=device, dtype=dtype)

    def forward(self, x):
        return self.down_proj(silu_mul(self.gate_up_proj(x)))


class TPLlamaDecoderLayer(nn.Module):
    def __init__(self, config: LlamaConfig, mesh, device=None, dtype=None):
        super().__init__()
        self.mesh = mesh
        self.input_layernorm = RMSNorm(config.hidden_size,
                                       config.rms_norm_eps, device, dtype)
        self.self_attn = TPLlamaAttention(config, mesh, device, dtype)
        self.post_attention_layernorm = RMSNorm(
            config.hidden_size, config.rms_norm_eps, device, dtype)
        self.mlp = TPLlamaMLP(config, mesh, device, dtype)

    def forward(self, x, position_ids=None):
        # x: [B, S/tp, h] sequence-sharded
        g = self.mesh.tp_group
        h = gather_seq(self.input_layernorm(x), g)          # AG (06:90-95)
        x = x + reduce_scatter_seq(self.self_attn(h, position_ids), g)  # RS
        h = gather_seq(self.post_attention_layernorm(x), g)  # AG (06:102-105)
        x = x + reduce_scatter_seq(self.mlp(h), g)           # RS (06:108)
        return x


class TPLlamaForCausalLM(nn.Module):
    """Sequence+tensor-parallel Llama over a DeviceMesh2D's tp dimension."""

    def __init__(self, config: LlamaConfig, mesh, device=None, dtype=None,
                 loss_parallel=False):
        super().__init__()
        self.config = config
        self.mesh = mesh
        self.loss_parallel = loss_parallel
        v = config.padded_vocab_size
        h = config.hidden_size
        self.embed_tokens = VocabParallelEmbedding(v, h, mesh, device, dtype)
        self.layers = nn.ModuleList(
            TPLlamaDecoderLayer(config, mesh, device, dtype)
            for _ in range(config.num_hidden_layers))
        self.norm = RMSNorm(h, config.rms_norm_eps, device, dtype)
        self.lm_head = ColwiseLinear(h, v, mesh, device=device, dtype=dtype)
        self.vocab_start = mesh.tp_rank * (v // mesh.tp_size)
        self.init_weights()
        self._register_replicated_grad_hooks()

    # ---- init: generate the FULL weight per tensor (same seed on every
    # rank) and slice the local shard — numerics identical to the
    # single-process model under the same torch.manual_seed ----
    def init_weights(self):
        if next(self.parameters()).is_meta:
            return
        cfg = self.config
        std = cfg.initializer_range
        v, h, i = cfg.padded_vocab_size, cfg.hidden_size, cfg.intermediate_size
        d = cfg.head_dim
        hq, hkv = cfg.num_attention_heads, cfg.num_key_value_heads
        tr = self.mesh.tp_rank
        dev = self.norm.weight.device

        def full(shape):
            return torch.empty(shape, device=dev,
                               dtype=self.norm.weight.dtype).normal_(0, std)

        with torch.no_grad():
            self.embed_tokens.load_full_weight(full((v, h)), tr)
            for layer in self.layers:
                layer.input_layernorm.weight.fill_(1.0)
                layer.post_attention_layernorm.weight.fill_(1.0)
                layer.self_attn.qkv_proj.load_full_weight(
                    full(((hq + 2 * hkv) * d, h)), tr)
                layer.self_attn.o_proj.load_full_weight(full((h, hq * d)), tr)
                layer.mlp.gate_up_proj.load_full_weight(full((2 * i, h)), tr)
                layer.mlp.down_proj.load_full_weight(full((h, i)), tr)
            self.norm.weight.fill_(1.0)
            self.lm_head.load_full_weight(full((v, h)), tr)

    # NOTE: init order above MUST match LlamaForCausalLM.init_weights'
    # modules() order for seed-parity tests: embed, per-layer (qkv, o,
    # gate_up, down), norm, lm_head — norms use fill so consume no RNG.

    def _register_replicated_grad_hooks(self):
        """Norm weights are tp-replicated; each rank's grad covers only its
        sequence shard -> sum over the tp group when the grad is ready."""
        g = self.mesh.tp_group
        if self.mesh.tp_size == 1:
            return

        def hook(param):
            dist.all_reduce(param.grad, group=g)

        for m in self.modules():
            if isinstance(m, RMSNorm):
                m.weight.register_post_accumulate_grad_hook(hook)

    def forward(self, input_ids, labels=None, attention_mask=None,
                position_ids=None, **_):
        g = self.mesh.tp_group
        if position_ids is not None and position_ids.dim() == 2:
            position_ids = position_ids[0]
        x = reduce_scatter_seq(self.embed_tokens(input_ids), g)
        for layer in self.layers:
            x = layer(x, position_ids)
        x = gather_seq(self.norm(x), g)
        logits_local = self.lm_head(x)
        loss = None
        logits = None
        if labels is not None and self.loss_parallel:
            loss = sharded_causal_lm_loss(logits_local, labels,
                                          self.vocab_start, g)
        else:
            logits = gather_vocab(logits_local, g)
            if labels is not None:
                loss = causal_lm_loss(logits, labels)
        return CausalLMOutput(loss=loss, logits=logits)

    # ---- checkpoint helpers: local tp shards (DCP-style file per rank) ----
    def tp_state_dict(self):
        return {k: v.detach().to("cpu") for k, v in self.state_dict().items()}

    def load_tp_state_dict(self, sd):
        self.load_state_dict({k: v for k, v in sd.items()})

    def reconstruct_full_tensor(self, name: str, parts: list):
        """Reassemble the FULL tensor for parameter `name` from all
        tp_old ranks' shards (parts[r] = rank r's tensor), per that
        module's sharding rule — the reshard-on-load primitive (the
        reference leans on torch DCP's planner for tp-size changes;
        SURVEY.md §7 'DCP-compatible sharded checkpoint with resharding')."""
        mod = dict(self.named_modules())[name.rsplit(".", 1)[0]]
        tp_old = len(parts)
        if isinstance(mod, ColwiseLinear):
            # old shard = concat over segments of that segment's slice
            full_segs = []
            off = 0
            for seg in mod.segments:
                loc = seg // tp_old
                full_segs.append(torch.cat([p[off: off + loc]
                                            for p in parts]))
                off += loc
            return torch.cat(full_segs)
        if isinstance(mod, RowwiseLinear):
            return torch.cat(parts, dim=1)
        if isinstance(mod, VocabParallelEmbedding):
            return torch.cat(parts, dim=0)
        return parts[0]  # replicated (norm weights)

    def shard_tensor(self, name: str, full: torch.Tensor) -> torch.Tensor:
        """Slice a FULL tensor down to this rank's shard per parameter
        `name`'s sharding rule (the inverse of reconstruct_full_tensor);
        also used to reshard optimizer moments, which follow their
        parameter's layout."""
        mod = dict(self.named_modules())[name.rsplit(".", 1)[0]]
        tr = self.mesh.tp_rank
        if isinstance(mod, ColwiseLinear):
            return _shard_rows(full, mod.segments, mod.tp, tr).contiguous()
        if isinstance(mod, RowwiseLinear):
            return full.narrow(1, tr * mod.in_local,
                               mod.in_local).contiguous()
        if isinstance(mod, VocabParallelEmbedding):
            return full.narrow(0, tr * mod.vlocal, mod.vlocal).contiguous()
        return full

    def load_tp_state_dict_resharded(self, shards: list):
        """Load from a DIFFERENT tp size: `shards` is every old rank's
        tp_state_dict() in rank order; each full tensor is reconstructed
        and re-sliced for this model's (tp, tp_rank)."""
        tr = self.mesh.tp_rank
        mods = dict(self.named_modules())
        with torch.no_grad():
            for name, p in self.named_parameters():
                full = self.reconstruct_full_tensor(
                    name, [sd[name] for sd in shards])
                mod = mods[name.rsplit(".", 1)[0]]
                if isinstance(mod, (ColwiseLinear, RowwiseLinear,
                                    VocabParallelEmbedding)):
                    mod.load_full_weight(full.to(p.device), tr)
                else:
                    p.copy_(full.to(p.device, p.dtype))
