"""bench.py driver-contract tests on CPU: every --parallelism path must
produce the one-line JSON contract, single-process and world_size=2 (gloo),
so the driver's round-end 8-GPU scaling run cannot hit an untested launch
path (BASELINE.json configs: dp / fsdp / tp / 2d)."""
import json
import os
import subprocess
import sys
from pathlib import Path

import pytest

from utils_dist import free_port

REPO = Path(__file__).resolve().parent.parent

COMMON = ["--model", "llama-debug", "--batch-size", "2",
          "--seq-length", "64", "--steps", "2", "--warmup", "1",
          "--device", "cpu"]


def _check_contract(line, n_gpus):
    rec = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in rec, key
    assert rec["n_gpus"] == n_gpus
    assert rec["value"] > 0
    assert rec["dtype"] == "bf16"
    assert rec["data"] == "synthetic"
    for key in ("model", "global_batch", "seq_len", "parallelism"):
        assert key in rec["config"], key
    return rec


@pytest.mark.parametrize("par", ["auto", "single", "fsdp", "tp", "2d"])
def test_bench_single_process(par):
    out = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--parallelism", par]
        + COMMON, capture_output=True, text=True, timeout=300, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [ln for ln in out.stdout.splitlines() if ln.startswith("{")][-1]
    rec = _check_contract(line, 1)
    assert rec["config"]["global_batch"] == 2


@pytest.mark.parametrize("par,label,gbatch", [
    ("auto", "dp2", 4),
    ("ddp", "dp2", 4),
    ("fsdp", "fsdp2", 4),
    ("tp", "tp2", 2),
    ("2d", "2d_fsdp1_tp2", 2),
])
def test_bench_world2(par, label, gbatch):
    port = free_port()
    env = dict(os.environ)
    env.pop("RANK", None), env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), str(REPO / "bench.py"),
         "--gpus", "2", "--parallelism", par] + COMMON,
        capture_output=True, text=True, timeout=600, cwd=REPO, env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [ln for ln in out.stdout.splitlines() if ln.startswith("{")][-1]
    rec = _check_contract(line, 2)
    assert rec["config"]["parallelism"] == label
    assert rec["config"]["global_batch"] == gbatch


def test_bench_2d_world4():
    """dp2 x tp2: both mesh communicator sets live at once (FSDP over the
    dp dim of TP-sharded layers) — the composition the 8-GPU round-end
    run exercises at dp4 x tp2."""
    port = free_port()
    env = dict(os.environ)
    env.pop("RANK", None), env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
         "--master-port", str(port), str(REPO / "bench.py"),
         "--gpus", "4", "--parallelism", "2d", "--tensor-parallel", "2"]
        + COMMON,
        capture_output=True, text=True, timeout=600, cwd=REPO, env=env)
    assert out.returncode == 0, out.stderr[-2500:]
    line = [ln for ln in out.stdout.splitlines() if ln.startswith("{")][-1]
    rec = _check_contract(line, 4)
    assert rec["config"]["parallelism"] == "2d_fsdp2_tp2"
    assert rec["config"]["global_batch"] == 4  # dp=2 replicas x bs2


def test_bench_collectives_world2():
    port = free_port()
    env = dict(os.environ)
    env.pop("RANK", None), env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), str(REPO / "tools/bench_collectives.py"),
         "--sizes-mb", "0.5", "--iters", "2", "--warmup", "1",
         "--device", "cpu"],
        capture_output=True, text=True, timeout=300, cwd=REPO, env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    recs = [json.loads(ln) for ln in out.stdout.splitlines()
            if ln.startswith("{")]
    ops = {r["op"] for r in recs}
    assert {"all_reduce", "all_gather", "reduce_scatter", "broadcast",
            "barrier"} <= ops
    for r in recs:
        if r["op"] != "barrier":
            assert r["algbw_gbs"] > 0
