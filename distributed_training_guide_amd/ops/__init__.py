"""HIP-kernel-backed ops (GPU) with eager fp32 references (CPU/tests)."""
from .adamw import FusedAdamW
from .attention import flash_attention
from .cross_entropy import causal_lm_loss, sharded_causal_lm_loss
from .fused_linear_ce import fused_causal_lm_loss
from .layernorm import LayerNorm, gelu
from .rmsnorm import RMSNorm, rmsnorm
from .rope import qkv_rope, rope
from .swiglu import silu_mul

__all__ = [
    "FusedAdamW",
    "flash_attention",
    "causal_lm_loss",
    "sharded_causal_lm_loss",
    "fused_causal_lm_loss",
    "LayerNorm",
    "gelu",
    "RMSNorm",
    "rmsnorm",
    "qkv_rope",
    "rope",
    "silu_mul",
]
