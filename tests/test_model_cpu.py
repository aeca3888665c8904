"""CPU model tests: forward/backward, init flow, registry."""
import math

import pytest
import torch

from distributed_training_guide_amd.models import (build_model, get_config,
                                                   LlamaForCausalLM)


def test_registry_known_models():
    cfg = get_config("meta-llama/Meta-Llama-3-8B")
    assert cfg.hidden_size == 4096 and cfg.num_key_value_heads == 8
    assert get_config("gpt2").hidden_size == 768
    cfg70 = get_config("llama-3-70b")
    assert cfg70.num_hidden_layers == 80
    with pytest.raises(ValueError):
        get_config("nope-model")


def test_param_count_formula():
    cfg = get_config("llama-debug")
    model = build_model(cfg)
    assert sum(p.numel() for p in model.parameters()) == cfg.num_parameters()


@pytest.mark.parametrize("name", ["llama-debug", "gpt2"])
def test_forward_backward(name):
    torch.manual_seed(0)
    cfg = get_config(name)
    if name == "gpt2":
        cfg.num_hidden_layers = 2
        cfg.hidden_size = 128
        cfg.num_attention_heads = 2
        cfg.vocab_size = 512
    model = build_model(cfg)
    V = cfg.vocab_size
    ids = torch.randint(0, V, (2, 32))
    out = model(input_ids=ids, labels=ids)
    assert out.loss is not None and math.isfinite(out.loss.item())
    # untrained loss should be near ln(padded vocab)
    assert abs(out.loss.item() - math.log(cfg.padded_vocab_size)) < 1.0
    out.loss.backward()
    for n, p in model.named_parameters():
        assert p.grad is not None, n
        assert torch.isfinite(p.grad).all(), n


def test_meta_init_then_materialize():
    cfg = get_config("llama-debug")
    with torch.device("meta"):
        model = LlamaForCausalLM(cfg)
    model = model.to_empty(device="cpu")
    model.init_weights()
    ids = torch.randint(0, cfg.vocab_size, (1, 16))
    out = model(input_ids=ids, labels=ids)
    assert math.isfinite(out.loss.item())


def test_position_ids_affect_output():
    torch.manual_seed(0)
    cfg = get_config("llama-debug")
    model = build_model(cfg)
    ids = torch.randint(0, cfg.vocab_size, (1, 16))
    out0 = model(input_ids=ids).logits
    pos = torch.arange(32, 48).unsqueeze(0)
    out1 = model(input_ids=ids, position_ids=pos).logits
    assert not torch.allclose(out0, out1)
