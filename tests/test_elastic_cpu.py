"""Elastic restart path: run the toy under torchrun --max-restarts and check
progress survives rank crashes via the shared state file (SURVEY.md §4 item
2; reference related-topics/elastic-training/toy.py:32-44, README.md:26-35).
No GPU required (gloo)."""
import os
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent


@pytest.mark.timeout(420)
def test_toy_elastic_restart(tmp_path):
    env = dict(os.environ)
    env["TOY_STATE_FILE"] = str(tmp_path / "toy-state.json")
    env["TOY_FAIL_AT"] = "3"  # deterministic: rank 0 dies once at iter 3
    # single attempt: the toy's interruptible_barrier keeps the surviving
    # rank signal-responsive, so OUR teardown is fast.  A separate,
    # torch-internal race remains: on restart, gloo connectFullMesh can
    # read the dead incarnation's address from the store and burn ~100 s
    # before failing, which torchelastic treats as one more worker
    # failure — the generous --max-restarts budget lets the run ride
    # through it (each clean run needs exactly 1 restart).
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run",
         "--standalone", "--local-addr", "127.0.0.1",
         "--nproc-per-node", "2", "--max-restarts", "6",
         str(REPO / "related-topics" / "elastic-training" / "toy.py")],
        env=env, cwd=str(tmp_path), capture_output=True, text=True,
        timeout=400)
    out = proc.stdout + proc.stderr
    assert proc.returncode == 0, out[-2000:]
    assert "done: {'iteration': 20}" in out, out[-2000:]
    # the state file is cleaned up on success
    assert not (tmp_path / "toy-state.json").exists()


@pytest.mark.timeout(120)
def test_toy_no_failures(tmp_path):
    """With failure injection off the toy runs straight through."""
    env = dict(os.environ)
    env["TOY_STATE_FILE"] = str(tmp_path / "toy-state.json")
    env["TOY_FAIL_PROB"] = "0"
    env["TOY_FAIL_AT"] = "-1"
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--standalone",
         "--local-addr", "127.0.0.1", "--nproc-per-node", "2",
         "--max-restarts", "0",
         str(REPO / "related-topics" / "elastic-training" / "toy.py")],
        env=env, cwd=str(tmp_path), capture_output=True, text=True,
        timeout=110)
    assert proc.returncode == 0, (proc.stdout + proc.stderr)[-2000:]
