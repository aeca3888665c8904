"""Multi-process (gloo, world 2) tests of the DDP bucket engine and ZeRO-1
— CPU stand-ins for the RCCL path per SURVEY.md §4."""

import torch

from utils_dist import run_dist


def _make_model(seed=0, dtype=torch.float32):
    from distributed_training_guide_amd.models import build_model

    torch.manual_seed(seed)
    return build_model("llama-debug", dtype=dtype)


def _batch(rank, seed=7):
    g = torch.Generator().manual_seed(seed + rank)
    return torch.randint(0, 1024, (2, 32), generator=g)


# ---------------- DDP grad averaging ----------------

def _ddp_grads_match(rank, world):
    from distributed_training_guide_amd.parallel.ddp import \
        DistributedDataParallel

    model = _make_model(seed=rank)  # different init; broadcast must fix it
    ddp = DistributedDataParallel(model, bucket_cap_mb=1)
    ids = _batch(rank)
    out = ddp(input_ids=ids, labels=ids)
    out.loss.backward()

    # single-process reference: same (broadcast = rank 0) weights, grads
    # averaged over both ranks' batches
    ref = _make_model(seed=0)
    grads = {}
    for r in range(world):
        ref.zero_grad()
        o = ref(input_ids=_batch(r), labels=_batch(r))
        o.loss.backward()
        for n, p in ref.named_parameters():
            grads[n] = grads.get(n, 0) + p.grad / world

    for n, p in ddp.module.named_parameters():
        assert torch.allclose(p.grad, grads[n], atol=1e-5), \
            f"{n}: {(p.grad - grads[n]).abs().max()}"


def test_ddp_grads_match():
    run_dist(_ddp_grads_match, world_size=2)


# ---------------- no_sync ----------------

def _no_sync_then_sync(rank, world):
    from distributed_training_guide_amd.parallel.ddp import \
        DistributedDataParallel

    model = _make_model(seed=0)
    ddp = DistributedDataParallel(model, bucket_cap_mb=1)
    with ddp.no_sync():
        out = ddp(input_ids=_batch(rank), labels=_batch(rank))
        (out.loss / 2).backward()
    # grads now differ across ranks (no all-reduce happened)
    g = next(ddp.module.parameters()).grad.clone()
    gathered = [torch.empty_like(g) for _ in range(world)]
    torch.distributed.all_gather(gathered, g)
    assert not torch.allclose(gathered[0], gathered[1])
    # boundary microbatch syncs, accumulating on top
    out = ddp(input_ids=_batch(rank + 10), labels=_batch(rank + 10))
    (out.loss / 2).backward()
    g2 = next(ddp.module.parameters()).grad.clone()
    gathered2 = [torch.empty_like(g2) for _ in range(world)]
    torch.distributed.all_gather(gathered2, g2)
    assert torch.allclose(gathered2[0], gathered2[1], atol=1e-6)


def test_ddp_no_sync():
    run_dist(_no_sync_then_sync, world_size=2)


# ---------------- zero_grad(set_to_none) re-aliasing ----------------

def _set_to_none_realias(rank, world):
    from distributed_training_guide_amd.parallel.ddp import \
        DistributedDataParallel

    model = _make_model(seed=0)
    ddp = DistributedDataParallel(model, bucket_cap_mb=1)
    for step in range(2):
        out = ddp(input_ids=_batch(rank, seed=step), labels=_batch(rank, seed=step))
        out.loss.backward()
        g = [p.grad.clone() for p in ddp.module.parameters()]
        for p in ddp.module.parameters():
            p.grad = None  # what optimizer.zero_grad(set_to_none=True) does
    # second step produced synced grads again
    gathered = [torch.empty_like(g[0]) for _ in range(world)]
    torch.distributed.all_gather(gathered, g[0])
    assert torch.allclose(gathered[0], gathered[1], atol=1e-6)


def test_ddp_set_to_none():
    run_dist(_set_to_none_realias, world_size=2)


# ---------------- ZeRO-1 ----------------

def _zero1_matches_plain(rank, world):
    from distributed_training_guide_amd.ops import FusedAdamW
    from distributed_training_guide_amd.parallel.ddp import \
        DistributedDataParallel
    from distributed_training_guide_amd.parallel.zero1 import \
        ZeroRedundancyOptimizer

    model = _make_model(seed=0)
    ddp = DistributedDataParallel(model, bucket_cap_mb=1)
    opt = ZeroRedundancyOptimizer(ddp.parameters(),
                                  optimizer_class=FusedAdamW, lr=1e-2)
    ref = _make_model(seed=0)
    ref_opt = FusedAdamW(ref.parameters(), lr=1e-2)

    for step in range(3):
        out = ddp(input_ids=_batch(rank, seed=step),
                  labels=_batch(rank, seed=step))
        out.loss.backward()
        opt.step()
        opt.zero_grad(set_to_none=True)

        ref.zero_grad()
        agg = {}
        for r in range(world):
            tmp = _make_model(seed=0)
            tmp.load_state_dict(ref.state_dict())
            o = tmp(input_ids=_batch(r, seed=step), labels=_batch(r, seed=step))
            o.loss.backward()
            for n, p in tmp.named_parameters():
                agg[n] = agg.get(n, 0) + p.grad / world
        for n, p in ref.named_parameters():
            p.grad = agg[n]
        ref_opt.step()
        ref_opt.zero_grad()

    for (n, p), (_, pr) in zip(ddp.module.named_parameters(),
                               ref.named_parameters()):
        assert torch.allclose(p, pr, atol=1e-5), \
            f"{n} diff {(p - pr).abs().max()}"
    # all ranks agree bitwise
    for p in ddp.module.parameters():
        gathered = [torch.empty_like(p) for _ in range(world)]
        torch.distributed.all_gather(gathered, p.detach())
        assert torch.equal(gathered[0], gathered[1])


def test_zero1_matches_plain_adamw():
    run_dist(_zero1_matches_plain, world_size=2)


# ---------------- chapter 2 end-to-end ----------------

def _chapter2_e2e(rank, world, tmpdir):
    import sys
    from pathlib import Path

    repo = Path(__file__).resolve().parent.parent
    sys.path.insert(0, str(repo / "02-distributed-data-parallel"))
    import importlib

    mod = importlib.import_module("train_llm")
    state = mod.main([
        "-m", "llama-debug", "-d", "synthetic", "-s", "32", "-b", "1",
        "--num-samples", "16", "--num-workers", "0", "--max-steps", "2",
        "-e", "ddp-e2e", "--ckpt-freq", "2", "--save-dir", tmpdir,
        "--device", "cpu", "--num-epochs", "1",
    ])
    assert state["global_step"] == 2


def test_chapter2_end_to_end(tmp_path):
    run_dist(_chapter2_e2e, world_size=2, args=(str(tmp_path),))
    assert (tmp_path / "ddp-e2e" / "model.pt").exists()
    assert (tmp_path / "ddp-e2e" / "state.json").exists()
    # ZeRO-1 drops optimizer.pt (reference README 02:308)
    assert not (tmp_path / "ddp-e2e" / "optimizer.pt").exists()
