#!/usr/bin/env python3
"""Elastic-restart demonstration — counterpart of
/root/reference/related-topics/elastic-training/toy.py: each rank randomly
crashes; torchrun --max-restarts restarts ALL workers; progress survives in
a shared state file written by rank 0 under barrier fencing.  No GPU
required (gloo).

    torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 4 \
        --max-restarts 3 related-topics/elastic-training/toy.py
"""
import datetime
import json
import os
import random
import time

import torch.distributed as dist
from torch.distributed.elastic.multiprocessing.errors import record

STATE_FILE = os.environ.get("TOY_STATE_FILE", "toy-state.json")


def interruptible_barrier(timeout_s: float = 20.0):
    """Async barrier polled from Python.  A plain dist.barrier() blocks in
    native gloo code holding the GIL, so a surviving rank cannot process
    torchelastic's SIGTERM until the barrier times out — the restart
    stalls for the full collective timeout.  Polling with time.sleep
    keeps signal delivery immediate (teardown in ~ms, not ~20 s)."""
    work = dist.barrier(async_op=True)
    deadline = time.time() + timeout_s
    while not work.is_completed():
        if time.time() > deadline:
            raise RuntimeError("barrier timed out (peer died?)")
        time.sleep(0.01)


@record
def main():
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    # store-barrier before gloo's address exchange: across torchelastic
    # restarts, out-of-phase re-inits can read a dead incarnation's
    # listener address (lowers the race rate but does not eliminate it)
    os.environ.setdefault("TORCH_DIST_INIT_BARRIER", "1")
    # bounded collectives AND a bounded gloo connect: the connectFullMesh
    # deadline is ~4x this timeout, so 10 s keeps a raced re-init cheap.
    # When the race fires anyway (stale peer address -> connection
    # refused), retry the init IN-PROCESS: each attempt uses a fresh
    # store prefix, so both workers re-pair without burning a
    # torchelastic restart (restarting the processes re-enters the same
    # race; retrying the group init resolves it).
    for attempt in range(4):
        try:
            dist.init_process_group(
                "gloo", timeout=datetime.timedelta(seconds=10))
            break
        except RuntimeError as e:
            if "connectFullMesh" not in str(e) or attempt == 3:
                raise
            try:
                dist.destroy_process_group()
            except (RuntimeError, ValueError):
                pass
            print(f"[rank={rank}] init race (attempt {attempt}); retrying")
            time.sleep(1.0 + 0.5 * rank)
    # NOTE: rank/world_size are NOT stable across restarts — reload shared
    # progress from the state file, never from process memory.
    state = {"iteration": 0}
    if os.path.exists(STATE_FILE):
        with open(STATE_FILE) as fp:
            state = json.load(fp)
    print(f"[rank={rank}/{world} restart={os.environ.get('TORCHELASTIC_RESTART_COUNT', 0)}] "
          f"resuming at iteration {state['iteration']}")

    fail_prob = float(os.environ.get("TOY_FAIL_PROB", "0.05"))
    # deterministic injection for tests: rank 0 fails once at this iter on
    # the first incarnation (TOY_FAIL_AT=-1 disables)
    fail_at = int(os.environ.get("TOY_FAIL_AT", "-1"))
    restarts = int(os.environ.get("TORCHELASTIC_RESTART_COUNT", 0))
    for it in range(state["iteration"], 20):
        time.sleep(0.1)  # "training"
        if fail_at >= 0:
            if rank == 0 and restarts == 0 and it == fail_at:
                raise RuntimeError(f"rank {rank} injected failure at {it}")
        elif random.random() < fail_prob:
            raise RuntimeError(f"rank {rank} simulated failure at iter {it}")
        state["iteration"] = it + 1
        interruptible_barrier()
        if rank == 0:
            with open(STATE_FILE, "w") as fp:
                json.dump(state, fp)
        interruptible_barrier()
    if rank == 0:
        print("done:", state)
        os.unlink(STATE_FILE)


if __name__ == "__main__":
    main()
