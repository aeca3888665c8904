from .gpt2 import GPT2Config, GPT2ForCausalLM
from .llama import (CausalLMOutput, LlamaConfig, LlamaDecoderLayer,
                    LlamaForCausalLM)
from .registry import build_model, get_config, resolve_name

__all__ = [
    "GPT2Config", "GPT2ForCausalLM", "CausalLMOutput", "LlamaConfig",
    "LlamaDecoderLayer", "LlamaForCausalLM", "build_model", "get_config",
    "resolve_name",
]
