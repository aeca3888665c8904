"""GPT-2 family causal LM (the reference's chapter-1 smoke model,
/root/reference/01-single-gpu/README.md:9-12 trains `gpt2` = 124M).

Learned positional embeddings, pre-LayerNorm blocks, GELU MLP, tied
embeddings — config-compatible shapes with HF gpt2.  Fully on the in-repo
gfx950 kernels: the flash attention kernel (head_dim 64), fused LayerNorm
and tanh-GELU (layernorm.hip) and the fused lm_head+CE loss.
"""
from dataclasses import dataclass, field

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import causal_lm_loss, flash_attention
from ..ops.embedding import Embedding
from ..ops.fused_linear_ce import fused_causal_lm_loss
from ..ops.layernorm import LayerNorm, gelu
from .llama import CausalLMOutput, pad_vocab


@dataclass
class GPT2Config:
    vocab_size: int = 50257
    hidden_size: int = 768
    num_hidden_layers: int = 12
    num_attention_heads: int = 12
    max_position_embeddings: int = 1024
    layer_norm_epsilon: float = 1e-5
    initializer_range: float = 0.02
    tie_word_embeddings: bool = True
    model_type: str = field(default="gpt2")

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_attention_heads

    @property
    def intermediate_size(self) -> int:
        return 4 * self.hidden_size

    @property
    def padded_vocab_size(self) -> int:
        return pad_vocab(self.vocab_size)

    def num_parameters(self) -> int:
        h = self.hidden_size
        per_block = 12 * h * h + 13 * h  # qkv/o/fc GEMMs+bias, 2 LN
        embed = self.padded_vocab_size * h * \
            (1 if self.tie_word_embeddings else 2) + \
            self.max_position_embeddings * h
        return per_block * self.num_hidden_layers + embed + 2 * h


class GPT2Block(nn.Module):
    def __init__(self, config: GPT2Config, device=None, dtype=None):
        super().__init__()
        h = config.hidden_size
        self.ln_1 = LayerNorm(h, eps=config.layer_norm_epsilon,
                              device=device, dtype=dtype)
        self.qkv_proj = nn.Linear(h, 3 * h, device=device, dtype=dtype)
        self.o_proj = nn.Linear(h, h, device=device, dtype=dtype)
        self.ln_2 = LayerNorm(h, eps=config.layer_norm_epsilon,
                              device=device, dtype=dtype)
        self.fc_in = nn.Linear(h, config.intermediate_size, device=device,
                               dtype=dtype)
        self.fc_out = nn.Linear(config.intermediate_size, h, device=device,
                                dtype=dtype)
        self.n_heads = config.num_attention_heads
        self.head_dim = config.head_dim

    def forward(self, x):
        B, S, h = x.shape
        y = self.ln_1(x)
        q, k, v = self.qkv_proj(y).split(h, dim=-1)
        q = q.view(B, S, self.n_heads, self.head_dim)
        k = k.view(B, S, self.n_heads, self.head_dim).contiguous()
        v = v.view(B, S, self.n_heads, self.head_dim).contiguous()
        o = flash_attention(q, k, v)
        x = x + self.o_proj(o.reshape(B, S, h))
        x = x + self.fc_out(gelu(self.fc_in(self.ln_2(x))))
        return x


class GPT2ForCausalLM(nn.Module):
    def __init__(self, config: GPT2Config, device=None, dtype=None):
        super().__init__()
        self.config = config
        v = config.padded_vocab_size
        h = config.hidden_size
        self.wte = Embedding(v, h, device=device, dtype=dtype)
        self.wpe = Embedding(config.max_position_embeddings, h,
                             device=device, dtype=dtype)
        self.blocks = nn.ModuleList(
            GPT2Block(config, device, dtype)
            for _ in range(config.num_hidden_layers))
        self.ln_f = LayerNorm(h, eps=config.layer_norm_epsilon,
                              device=device, dtype=dtype)
        self.lm_head = nn.Linear(h, v, bias=False, device=device, dtype=dtype)
        if config.tie_word_embeddings:
            self.lm_head.weight = self.wte.weight
        self.init_weights()

    def init_weights(self):
        std = self.config.initializer_range
        with torch.no_grad():
            for m in self.modules():
                if isinstance(m, (nn.Linear, nn.Embedding, Embedding)):
                    if not m.weight.is_meta:
                        m.weight.normal_(0.0, std)
                    if isinstance(m, nn.Linear) and m.bias is not None \
                            and not m.bias.is_meta:
                        m.bias.zero_()
                elif isinstance(m, LayerNorm):
                    m.reset_parameters()

    def reset_param_by_name(self, name: str, tensor: torch.Tensor):
        std = self.config.initializer_range
        with torch.no_grad():
            if ".ln_1." in name or ".ln_2." in name or name.startswith("ln_f."):
                if name.endswith(".bias"):
                    tensor.zero_()
                else:
                    tensor.fill_(1.0)
            elif name.endswith(".bias"):
                tensor.zero_()
            else:
                tensor.normal_(0.0, std)

    def forward(self, input_ids, labels=None, attention_mask=None,
                position_ids=None, **_):
        B, S = input_ids.shape
        if position_ids is None:
            position_ids = torch.arange(S, device=input_ids.device)
        elif position_ids.dim() == 2:
            position_ids = position_ids[0]
        x = self.wte(input_ids) + self.wpe(position_ids)[None, :, :]
        for block in self.blocks:
            x = block(x)
        x = self.ln_f(x)
        if labels is not None and x.is_cuda:
            # fused projection+CE (no retained [B,S,V] grad; see
            # ops/fused_linear_ce.py) — logits are 65x the hidden size
            # for gpt2, the dominant activation
            loss = fused_causal_lm_loss(x, self.lm_head.weight, labels)
            return CausalLMOutput(loss=loss, logits=None)
        logits = self.lm_head(x)
        loss = None
        if labels is not None:
            loss = causal_lm_loss(logits, labels)
        return CausalLMOutput(loss=loss, logits=logits)
