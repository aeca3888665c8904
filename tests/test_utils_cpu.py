"""Small-unit coverage: LocalTimer semantics, mem-stats keys, model
registry aliases/param formulas."""
import time

import pytest
import torch


def test_local_timer_avg_and_reset():
    from distributed_training_guide_amd.utils.timers import LocalTimer

    t = LocalTimer(torch.device("cpu"))
    assert t.avg_elapsed_ms() == 0.0
    with t:
        time.sleep(0.01)
    with t:
        time.sleep(0.03)
    avg = t.avg_elapsed_ms()
    assert 5 < avg < 200
    t.reset()
    assert t.avg_elapsed_ms() == 0.0


def test_timer_skips_on_exception():
    from distributed_training_guide_amd.utils.timers import LocalTimer

    t = LocalTimer(torch.device("cpu"))
    with pytest.raises(ValueError):
        with t:
            raise ValueError("boom")
    # failed phase not recorded (reference semantics: tb aborts the sample)
    assert t.measurements == []


def test_mem_stats_keys_cpu():
    from distributed_training_guide_amd.utils.timers import get_mem_stats

    stats = get_mem_stats(torch.device("cpu"))
    assert set(stats) == {"total_gb", "curr_alloc_gb", "peak_alloc_gb",
                          "curr_resv_gb", "peak_resv_gb"}


def test_registry_aliases_and_params():
    from distributed_training_guide_amd.models import (build_model,
                                                       get_config,
                                                       resolve_name)

    assert resolve_name("meta-llama/Meta-Llama-3-8B") == "llama-3-8b"
    assert resolve_name("openai-community/gpt2") == "gpt2"
    c = get_config("llama-3-8b")
    assert abs(c.num_parameters() / 1e9 - 8.0) < 0.2
    with pytest.raises(ValueError):
        get_config("nonexistent-model")
    # formula == construction for one model of each family
    for name in ("gpt2-medium", "llama-3.2-1b"):
        cfg = get_config(name)
        m = build_model(name)
        assert sum(p.numel() for p in m.parameters()) == \
            cfg.num_parameters(), name
