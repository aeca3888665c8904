"""Llama-family causal LM built directly on the gfx950 HIP ops.

Config-compatible shapes with transformers' Llama (the reference builds its
models via AutoModelForCausalLM.from_config,
/root/reference/01-single-gpu/train_llm.py:47-49) but implemented natively:
RMSNorm / RoPE / flash attention / SwiGLU / fused CE are this repo's HIP
kernels; the projections are plain GEMMs (hipBLASLt via torch.linear).

MI355X-first choices:
  * BSHD activations everywhere — q/k/v go from the packed qkv GEMM into the
    attention kernel without a single transpose.
  * qkv packed as ONE GEMM (q | k | v segments), gate+up packed as ONE GEMM
    (halved launch count; the TP layer shards these segment-aware).
  * vocab padded to a multiple of 64 (GEMM-friendly; CE kernel needs /8).
  * meta-device init + init_weights()/reset_parameters() flow supported for
    the FSDP chapters (reference 04:74-95, 06:123-125).
"""
from dataclasses import dataclass, field

import torch
import torch.nn as nn

from ..ops import (RMSNorm, causal_lm_loss, flash_attention, qkv_rope,
                   silu_mul)
from ..ops.embedding import Embedding
from ..ops.fused_linear_ce import fused_causal_lm_loss
from ..ops.rmsnorm import add_rmsnorm


def pad_vocab(v: int, multiple: int = 64) -> int:
    return (v + multiple - 1) // multiple * multiple


@dataclass
class LlamaConfig:
    vocab_size: int = 32000
    hidden_size: int = 4096
    intermediate_size: int = 11008
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 32
    max_position_embeddings: int = 4096
    rope_theta: float = 10000.0
    rms_norm_eps: float = 1e-5
    initializer_range: float = 0.02
    tie_word_embeddings: bool = False
    model_type: str = field(default="llama")

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_attention_heads

    @property
    def padded_vocab_size(self) -> int:
        return pad_vocab(self.vocab_size)

    def num_parameters(self) -> int:
        h, i, v = self.hidden_size, self.intermediate_size, self.padded_vocab_size
        d = self.head_dim
        attn = h * (self.num_attention_heads + 2 * self.num_key_value_heads) * d + h * h
        mlp = 3 * h * i
        per_layer = attn + mlp + 2 * h
        embed = v * h * (1 if self.tie_word_embeddings else 2)
        return per_layer * self.num_hidden_layers + embed + h


@dataclass
class CausalLMOutput:
    loss: torch.Tensor | None = None
    logits: torch.Tensor | None = None


class LlamaAttention(nn.Module):
    def __init__(self, config: LlamaConfig, device=None, dtype=None):
        super().__init__()
        self.config = config
        h = config.hidden_size
        d = config.head_dim
        self.num_heads = config.num_attention_heads
        self.num_kv_heads = config.num_key_value_heads
        self.head_dim = d
        # packed q|k|v projection — one GEMM
        self.qkv_proj = nn.Linear(
            h, (self.num_heads + 2 * self.num_kv_heads) * d, bias=False,
            device=device, dtype=dtype)
        self.o_proj = nn.Linear(self.num_heads * d, h, bias=False,
                                device=device, dtype=dtype)

    def forward(self, x, position_ids=None):
        B, S, _ = x.shape
        d = self.head_dim
        qkv = self.qkv_proj(x)
        # fused split + RoPE: one pass, no intermediate copies, backward
        # writes dqkv directly (no grad cat)
        q, k, v = qkv_rope(qkv, self.num_heads, self.num_kv_heads, d,
                           self.config.rope_theta, positions=position_ids,
                           max_pos=self.config.max_position_embeddings)
        o = flash_attention(q, k, v)
        return self.o_proj(o.reshape(B, S, self.num_heads * d))


class LlamaMLP(nn.Module):
    def __init__(self, config: LlamaConfig, device=None, dtype=None):
        super().__init__()
        h, i = config.hidden_size, config.intermediate_size
        # packed gate|up projection — one GEMM
        self.gate_up_proj = nn.Linear(h, 2 * i, bias=False, device=device,
                                      dtype=dtype)
        self.down_proj = nn.Linear(i, h, bias=False, device=device,
                                   dtype=dtype)

    def forward(self, x):
        return self.down_proj(silu_mul(self.gate_up_proj(x)))


class LlamaDecoderLayer(nn.Module):
    def __init__(self, config: LlamaConfig, device=None, dtype=None):
        super().__init__()
        self.input_layernorm = RMSNorm(config.hidden_size,
                                       config.rms_norm_eps, device, dtype)
        self.self_attn = LlamaAttention(config, device, dtype)
        self.post_attention_layernorm = RMSNorm(
            config.hidden_size, config.rms_norm_eps, device, dtype)
        self.mlp = LlamaMLP(config, device, dtype)

    def forward(self, x, position_ids=None, residual=None):
        """Two call forms (both via __call__ so FSDP/checkpoint hooks fire):
        - forward(x): the reference chain x + attn(norm(x)) + mlp(...).
        - forward(delta, residual=res): fused (delta, residual) threading —
          each residual add is fused into the next RMSNorm kernel; returns
          (new_delta, new_residual), equivalent to forward(res + delta).
        """
        if residual is None:
            x = x + self.self_attn(self.input_layernorm(x), position_ids)
            x = x + self.mlp(self.post_attention_layernorm(x))
            return x
        normed, residual = add_rmsnorm(residual, x,
                                       self.input_layernorm.weight,
                                       self.input_layernorm.eps)
        a = self.self_attn(normed, position_ids)
        normed, residual = add_rmsnorm(residual, a,
                                       self.post_attention_layernorm.weight,
                                       self.post_attention_layernorm.eps)
        return self.mlp(normed), residual


class LlamaForCausalLM(nn.Module):
    def __init__(self, config: LlamaConfig, device=None, dtype=None):
        super().__init__()
        self.config = config
        v = config.padded_vocab_size
        h = config.hidden_size
        self.embed_tokens = Embedding(v, h, device=device, dtype=dtype)
        self.layers = nn.ModuleList(
            LlamaDecoderLayer(config, device, dtype)
            for _ in range(config.num_hidden_layers))
        self.norm = RMSNorm(h, config.rms_norm_eps, device, dtype)
        self.lm_head = nn.Linear(h, v, bias=False, device=device, dtype=dtype)
        if config.tie_word_embeddings:
            self.lm_head.weight = self.embed_tokens.weight
        self.init_weights()

    # ---- init flow (meta-device friendly; reference 04:74-95, 06:123-125) --
    def init_weights(self):
        std = self.config.initializer_range
        with torch.no_grad():
            for m in self.modules():
                self._reset_module(m, std)

    def reset_param_by_name(self, name: str, tensor: torch.Tensor):
        """Initialize one (materialized) parameter by its qualified name —
        used by the FSDP engine's unit-by-unit meta materialization
        (reference flow 04:74-95)."""
        std = self.config.initializer_range
        with torch.no_grad():
            if "layernorm" in name or name.startswith("norm."):
                tensor.fill_(1.0)
            elif name.endswith(".bias"):
                tensor.zero_()
            else:
                tensor.normal_(0.0, std)

    @staticmethod
    def _reset_module(m, std):
        if isinstance(m, (nn.Linear, nn.Embedding, Embedding)):
            if not m.weight.is_meta:
                m.weight.normal_(0.0, std)
            if isinstance(m, nn.Linear) and m.bias is not None \
                    and not m.bias.is_meta:
                m.bias.zero_()
        elif isinstance(m, RMSNorm):
            m.reset_parameters()

    def forward(self, input_ids, labels=None, attention_mask=None,
                position_ids=None, **_):
        # attention_mask accepted for API parity (reference data pipeline
        # emits it, 01:69); attention is always causal over packed rows.
        if position_ids is not None and position_ids.dim() == 2:
            position_ids = position_ids[0]
        x = self.embed_tokens(input_ids)
        if x.is_cuda:
            # fused (delta, residual) path: every residual add is fused
            # into the next RMSNorm kernel (add_rmsnorm.hip). Every layer
            # is entered through __call__ so FSDP / activation-checkpoint
            # hooks fire; the zero first delta costs one extra H-read.
            residual = x
            delta = torch.zeros_like(x)
            for layer in self.layers:
                delta, residual = layer(delta, position_ids,
                                        residual=residual)
            x, _ = add_rmsnorm(residual, delta, self.norm.weight,
                               self.norm.eps)
        else:
            for layer in self.layers:
                x = layer(x, position_ids)
            x = self.norm(x)
        if labels is not None and x.is_cuda:
            # chunked fused projection+CE: the [B,S,V] logits (+grad) are
            # never materialized (ops/fused_linear_ce.py) — logits=None,
            # like the reference's loss-only training consumers (02:149-163
            # only reads outputs.loss)
            loss = fused_causal_lm_loss(x, self.lm_head.weight, labels)
            return CausalLMOutput(loss=loss, logits=None)
        logits = self.lm_head(x)
        loss = None
        if labels is not None:
            loss = causal_lm_loss(logits, labels)
        return CausalLMOutput(loss=loss, logits=logits)
