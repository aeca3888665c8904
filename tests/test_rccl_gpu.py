"""RCCL-branch de-risking on a single GPU.

The multi-GPU engines' CUDA collective branches (all_gather_into_tensor,
reduce_scatter_tensor, broadcast(group_src=...), async work.wait() stream
semantics) can run on ONE GPU via a single-rank RCCL communicator —
collectives degenerate to device copies but exercise the exact API calls,
argument forms and stream ordering the 8-GPU run will use.
DTGA_FORCE_COLLECTIVES=1 makes the engines take those branches at world=1
(parallel/fsdp.py, parallel/ddp.py, parallel/tp.py)."""
import json
import os
import subprocess
import sys
from pathlib import Path

import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = Path(__file__).resolve().parent.parent


@pytest.fixture()
def nccl_world1():
    import torch.distributed as dist

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29531")
    if not dist.is_initialized():
        dist.init_process_group("nccl", rank=0, world_size=1,
                                device_id=torch.device("cuda:0"))
    yield dist
    if dist.is_initialized():
        dist.destroy_process_group()


def test_dist_api_surface_rccl(nccl_world1):
    """Every c10d call form the engines issue, on a real RCCL
    communicator: signatures, kwargs, async semantics."""
    dist = nccl_world1
    dev = torch.device("cuda:0")
    shard = torch.randn(1024, device=dev, dtype=torch.bfloat16)
    full = torch.empty(1024, device=dev, dtype=torch.bfloat16)
    # fsdp.py unshard / zero1.py step
    w = dist.all_gather_into_tensor(full, shard, async_op=True)
    w.wait()
    assert torch.equal(full, shard)
    # fsdp.py reduce_scatter_grads (fp32 reduce dtype)
    big = torch.randn(2048, device=dev, dtype=torch.float32)
    out = torch.empty(2048, device=dev, dtype=torch.float32)
    w = dist.reduce_scatter_tensor(out, big, async_op=True)
    w.wait()
    assert torch.allclose(out, big)
    # ddp.py _broadcast_module_states / fsdp.py load_full_state_dict
    t = torch.randn(512, device=dev)
    dist.broadcast(t, group_src=0)
    # ddp.py bucket all-reduce (async, waited by the engine callback)
    g = torch.randn(4096, device=dev, dtype=torch.bfloat16)
    ref = g.clone()
    w = dist.all_reduce(g, async_op=True)
    w.wait()
    assert torch.equal(g, ref)
    # in-place all-gather (zero1.py: input is a view of the output)
    buf = torch.randn(2048, device=dev, dtype=torch.bfloat16)
    my = buf[0:2048]
    dist.all_gather_into_tensor(buf, my)
    dist.barrier()
    torch.cuda.synchronize()


def test_ddp_forced_collectives_world1(nccl_world1, monkeypatch):
    """DDP bucket engine's real RCCL branch (broadcast at construction +
    async bucket all-reduce + engine-callback wait) at world=1."""
    monkeypatch.setenv("DTGA_FORCE_COLLECTIVES", "1")
    from distributed_training_guide_amd.models import build_model
    from distributed_training_guide_amd.parallel.ddp import \
        DistributedDataParallel
    from distributed_training_guide_amd.parallel.zero1 import \
        ZeroRedundancyOptimizer

    torch.manual_seed(0)
    model = build_model("llama-debug", device=torch.device("cuda"),
                        dtype=torch.bfloat16)
    model = DistributedDataParallel(model, bucket_cap_mb=8)
    # ZeRO-1's in-place shard all-gather also runs under the forced flag
    opt = ZeroRedundancyOptimizer(model.parameters(), lr=1e-4)
    ids = torch.randint(0, 1024, (2, 64), device="cuda")
    for _ in range(2):
        out = model(input_ids=ids, labels=ids)
        out.loss.backward()
        opt.step()
        opt.zero_grad(set_to_none=True)
    torch.cuda.synchronize()
    assert torch.isfinite(out.loss)


def _run_bench(par, extra=()):
    import sys as _sys

    _sys.path.insert(0, str(REPO / "tests"))
    from utils_dist import free_port

    env = dict(os.environ)
    env["DTGA_FORCE_COLLECTIVES"] = "1"
    env.pop("RANK", None), env.pop("WORLD_SIZE", None)
    # never inherit the fixture's rendezvous port (EADDRINUSE)
    env["MASTER_ADDR"] = "127.0.0.1"
    env["MASTER_PORT"] = str(free_port())
    out = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--parallelism", par,
         "--model", "llama-60m", "--batch-size", "2", "--seq-length", "128",
         "--steps", "2", "--warmup", "1", *extra],
        capture_output=True, text=True, timeout=600, cwd=REPO, env=env)
    assert out.returncode == 0, (out.stdout[-1000:], out.stderr[-3000:])
    line = [ln for ln in out.stdout.splitlines() if ln.startswith("{")][-1]
    return json.loads(line)


def test_bench_fsdp_forced_rccl_gpu():
    """bench --parallelism fsdp at N=1 with the engines' real RCCL
    branches forced: AG/RS/prefetch/work.wait() stream path end to end."""
    rec = _run_bench("fsdp")
    assert rec["config"]["parallelism"] == "fsdp1"
    assert rec["value"] > 0


def test_bench_tp_forced_rccl_gpu():
    """TP boundary collectives (seq AG/RS, sharded loss all_reduce) on a
    real RCCL communicator at tp=1."""
    rec = _run_bench("tp")
    assert rec["config"]["parallelism"] == "tp1"
    assert rec["value"] > 0


def test_bench_2d_forced_rccl_gpu():
    rec = _run_bench("2d")
    assert rec["config"]["parallelism"] == "2d_fsdp1_tp1"
    assert rec["value"] > 0
