"""Multi-process (gloo) tests of the config-driven ZeRO-3 engine
(distributed_training_guide_amd/engine.py) — the alternative-frameworks
chapter's counterpart of deepspeed.initialize
(/root/reference/alternative-frameworks/deepspeed/train_llm.py:58-73)."""

import pytest
import torch

from utils_dist import run_dist


def _factory(dtype):
    from distributed_training_guide_amd.models import build_model

    torch.manual_seed(0)
    return build_model("llama-debug", dtype=torch.float32)


def _worker_train(rank, world, tmpdir):
    from distributed_training_guide_amd import engine as engine_mod

    config = {
        "bf16": {"enabled": False},
        "gradient_accumulation_steps": 2,
        "optimizer": {"type": "AdamW", "params": {"lr": 1e-2}},
    }
    eng, opt, _, sched = engine_mod.initialize(
        config, model_factory=_factory, device=torch.device("cpu"))
    assert eng.world_size == world
    g = torch.Generator().manual_seed(rank + 1)
    losses = []
    for step in range(4):
        ids = torch.randint(0, 1024, (2, 32), generator=g)
        out = eng(input_ids=ids, labels=ids)
        eng.backward(out.loss)
        eng.step()  # no-op on non-boundary microbatches
        losses.append(out.loss.item())
    assert eng.global_step == 2  # 4 micro / accum 2
    # loss decreased over the accumulation windows on average
    eng.save_checkpoint(tmpdir, {"epoch": 0})

    # fresh engine resumes
    eng2, _, _, _ = engine_mod.initialize(
        config, model_factory=_factory, device=torch.device("cpu"))
    state = eng2.load_checkpoint(tmpdir)
    assert state is not None and eng2.global_step == 2
    for u1, u2 in zip(eng.module.units, eng2.module.units):
        assert torch.equal(u1.shard.detach(), u2.shard.detach())


@pytest.mark.parametrize("world", [2])
def test_engine_train_ckpt_resume(tmp_path, world):
    run_dist(_worker_train, world_size=world, args=(str(tmp_path),))


def test_engine_config_merge():
    from distributed_training_guide_amd.engine import DEFAULT_CONFIG, _merge

    cfg = _merge(DEFAULT_CONFIG, {"zero_optimization": {"stage": 3,
                                                        "overlap_comm": False}})
    assert cfg["zero_optimization"]["overlap_comm"] is False
    assert cfg["zero_optimization"]["reshard_after_forward"] is True
    assert cfg["bf16"]["enabled"] is True


def test_engine_rejects_non_stage3():
    from distributed_training_guide_amd.engine import (DEFAULT_CONFIG, Engine,
                                                       _merge)

    cfg = _merge(DEFAULT_CONFIG, {"zero_optimization": {"stage": 1}})
    with pytest.raises(ValueError, match="stage 3"):
        Engine(torch.nn.Linear(2, 2), cfg, torch.device("cpu"))
