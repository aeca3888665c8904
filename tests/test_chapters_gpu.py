"""Chapter entrypoints end-to-end on the GPU (single process): run a few
steps through the full HIP path, write the reference checkpoint layout,
resume — the runnable-command verification on real
hardware."""
import json
import os
import subprocess
import sys
from pathlib import Path

import pytest

pytestmark = pytest.mark.gpu

REPO = Path(__file__).resolve().parent.parent

COMMON = ["-m", "llama-60m", "-d", "synthetic", "-s", "256", "-b", "4",
          "--num-samples", "64", "--num-workers", "0", "--log-freq", "2",
          "--max-steps", "3", "--ckpt-freq", "2", "--num-epochs", "2"]


def _env():
    env = dict(os.environ)
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR",
              "MASTER_PORT"):
        env.pop(k, None)
    return env


def test_chapter1_gpu(tmp_path):
    out = subprocess.run(
        [sys.executable, str(REPO / "01-single-gpu" / "train_llm.py"),
         *COMMON, "--save-dir", str(tmp_path), "-e", "exp"],
        capture_output=True, text=True, timeout=300, cwd=REPO, env=_env())
    assert out.returncode == 0, out.stderr[-2000:]
    exp = tmp_path / "exp"
    for f in ("model.pt", "optimizer.pt", "lr_scheduler.pt", "state.json"):
        assert (exp / f).exists(), f
    # resume past the saved step
    out = subprocess.run(
        [sys.executable, str(REPO / "01-single-gpu" / "train_llm.py"),
         *COMMON, "--save-dir", str(tmp_path), "-e", "exp",
         "--max-steps", "5"],
        capture_output=True, text=True, timeout=300, cwd=REPO, env=_env())
    assert out.returncode == 0, out.stderr[-2000:]
    assert json.loads((exp / "state.json").read_text())["global_step"] >= 4


def test_chapter4_fsdp_gpu_world1_forced(tmp_path):
    """ch4 under torchrun nproc=1 with the real RCCL collective branches
    forced — the FSDP chapter's full loop incl. sharded checkpoint."""
    from utils_dist import free_port

    env = _env()
    env["DTGA_FORCE_COLLECTIVES"] = "1"
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "1", "--master-addr", "127.0.0.1",
         "--master-port", str(free_port()),
         str(REPO / "04-fully-sharded-data-parallel" / "train_llm.py"),
         *COMMON, "--save-dir", str(tmp_path), "-e", "exp"],
        capture_output=True, text=True, timeout=400, cwd=REPO, env=env)
    assert out.returncode == 0, out.stderr[-2500:]
    exp = tmp_path / "exp"
    assert (exp / "checkpoint" / "shard_rank0.pt").exists()
    assert (exp / "checkpoint" / "metadata.json").exists()
    assert json.loads((exp / "state.json").read_text())["global_step"] == 2


def test_bench_hip_graphs_mode(tmp_path):
    """--hip-graphs captures the whole step and replays it (opt-in
    experiment); must produce the normal JSON contract."""
    out = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--model", "llama-60m",
         "--batch-size", "4", "--seq-length", "256", "--steps", "3",
         "--warmup", "1", "--hip-graphs"],
        capture_output=True, text=True, timeout=300, cwd=REPO, env=_env())
    assert out.returncode == 0, out.stderr[-2000:]
    line = [ln for ln in out.stdout.splitlines() if ln.startswith("{")][-1]
    rec = json.loads(line)
    assert rec["value"] > 0
