"""Model registry: name -> config, mirroring the reference's
AutoConfig.from_pretrained(model_name) surface
(/root/reference/01-single-gpu/train_llm.py:48) for the models the chapters
use — resolved offline from built-in configs (no network, random init).
"""
from .gpt2 import GPT2Config, GPT2ForCausalLM
from .llama import LlamaConfig, LlamaForCausalLM

_LLAMA_CONFIGS = {
    # name -> (hidden, intermediate, layers, heads, kv_heads, vocab, theta, max_pos)
    "llama-2-7b": (4096, 11008, 32, 32, 32, 32000, 1e4, 4096),
    "llama-2-13b": (5120, 13824, 40, 40, 40, 32000, 1e4, 4096),
    "llama-3-8b": (4096, 14336, 32, 32, 8, 128256, 5e5, 8192),
    "llama-3-70b": (8192, 28672, 80, 64, 8, 128256, 5e5, 8192),
    "llama-3-405b": (16384, 53248, 126, 128, 8, 128256, 5e5, 8192),
    "llama-3.2-1b": (2048, 8192, 16, 32, 8, 128256, 5e5, 8192),
    "llama-3.2-3b": (3072, 8192, 28, 24, 8, 128256, 5e5, 8192),
    # small configs for tests / smoke runs
    "llama-debug": (256, 688, 4, 4, 2, 1024, 1e4, 2048),
    "llama-60m": (512, 1376, 8, 8, 8, 32000, 1e4, 2048),
}

_GPT2_CONFIGS = {
    # name -> (hidden, layers, heads); head_dim is 64 throughout
    "gpt2": (768, 12, 12),
    "gpt2-medium": (1024, 24, 16),
    "gpt2-large": (1280, 36, 20),
    "gpt2-xl": (1600, 48, 25),
}

_ALIASES = {
    "gpt2": "gpt2",
    "openai-community/gpt2": "gpt2",
    "openai-community/gpt2-medium": "gpt2-medium",
    "openai-community/gpt2-large": "gpt2-large",
    "openai-community/gpt2-xl": "gpt2-xl",
    "meta-llama/llama-2-7b-hf": "llama-2-7b",
    "meta-llama/llama-2-13b-hf": "llama-2-13b",
    "meta-llama/meta-llama-3-8b": "llama-3-8b",
    "meta-llama/meta-llama-3.1-8b": "llama-3-8b",
    "meta-llama/llama-3.1-8b": "llama-3-8b",
    "meta-llama/meta-llama-3-70b": "llama-3-70b",
    "meta-llama/llama-3.1-70b": "llama-3-70b",
    "meta-llama/meta-llama-3.1-405b": "llama-3-405b",
    "meta-llama/llama-3.1-405b": "llama-3-405b",
    "meta-llama/llama-3.2-1b": "llama-3.2-1b",
    "meta-llama/llama-3.2-3b": "llama-3.2-3b",
}


def resolve_name(name: str) -> str:
    key = name.lower()
    return _ALIASES.get(key, key)


def get_config(name: str):
    key = resolve_name(name)
    if key in _GPT2_CONFIGS:
        h, L, heads = _GPT2_CONFIGS[key]
        return GPT2Config(hidden_size=h, num_hidden_layers=L,
                          num_attention_heads=heads)
    if key in _LLAMA_CONFIGS:
        h, i, L, hq, hkv, v, theta, mp = _LLAMA_CONFIGS[key]
        return LlamaConfig(
            vocab_size=v, hidden_size=h, intermediate_size=i,
            num_hidden_layers=L, num_attention_heads=hq,
            num_key_value_heads=hkv, max_position_embeddings=mp,
            rope_theta=theta,
            tie_word_embeddings=key.startswith("llama-3.2"),
        )
    raise ValueError(
        f"unknown model {name!r}; known: "
        f"{', '.join(_GPT2_CONFIGS)}, {', '.join(_LLAMA_CONFIGS)}")


def build_model(name_or_config, device=None, dtype=None):
    config = (name_or_config if not isinstance(name_or_config, str)
              else get_config(name_or_config))
    if isinstance(config, GPT2Config):
        return GPT2ForCausalLM(config, device=device, dtype=dtype)
    return LlamaForCausalLM(config, device=device, dtype=dtype)
