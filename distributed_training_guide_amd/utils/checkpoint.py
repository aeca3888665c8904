"""Checkpoint / resume — both reference formats:

* Unsharded (chapters 1/2): {model.pt, optimizer.pt, lr_scheduler.pt,
  state.json} saved rank-0-only under barrier fencing
  (/root/reference/01-single-gpu/train_llm.py:181-187, 02:192-199), loaded
  with weights_only=True (01:95-101).
* Sharded (chapters 4-7): exp_dir/checkpoint/ with one shard file per rank
  plus rank-0 metadata — the reference's DCP "file per rank" layout
  (04-fully-sharded-data-parallel/README.md:168, 04:241-255), with
  lr_scheduler.pt and state.json still rank-0-only.

Resume trigger = state.json exists for the experiment (01:94).
RNG state save/restore implements the determinism recipe
(related-topics/determinism/README.md:46-68) behind save_rng/load_rng.
"""
import json
import random
from pathlib import Path

import numpy as np
import torch
import torch.distributed as dist


def _barrier():
    if dist.is_available() and dist.is_initialized():
        dist.barrier()


def optim_sd_cpu(optimizer) -> dict:
    """Optimizer state dict with all tensors copied to CPU, WITHOUT
    mutating the live state.  torch's state_dict() returns references to
    the live per-param state dicts: assigning `st[k] = v.to('cpu')` in
    place would swap the GPU moment buffers for CPU copies under a running
    FusedAdamW (whose chunk-descriptor cache holds raw device pointers) —
    every later fused step would write through dangling pointers."""
    sd = optimizer.state_dict()
    sd["state"] = {
        k: {kk: (vv.to("cpu") if isinstance(vv, torch.Tensor) else vv)
            for kk, vv in st.items()}
        for k, st in sd["state"].items()
    }
    return sd


def mkdir_rank0(path: Path, rank: int):
    """Mount-aware mkdir under barrier fencing (reference 04:160-168)."""
    _barrier()
    if rank == 0:
        path.mkdir(parents=True, exist_ok=True)
    _barrier()


# ---------------- unsharded (chapters 1/2) ----------------

def save_unsharded(exp_dir: Path, model, optimizer, lr_scheduler, state,
                   rank: int = 0, save_optimizer: bool = True):
    if rank == 0:
        if save_optimizer and optimizer is not None:
            torch.save(optimizer.state_dict(), exp_dir / "optimizer.pt")
        torch.save(model.state_dict(), exp_dir / "model.pt")
        torch.save(lr_scheduler.state_dict(), exp_dir / "lr_scheduler.pt")
        with open(exp_dir / "state.json", "w") as fp:
            json.dump(state, fp)
    _barrier()


def load_unsharded(exp_dir: Path, model, optimizer, lr_scheduler, device,
                   load_optimizer: bool = True):
    def _load(p):
        return torch.load(p, map_location=device, weights_only=True)

    model.load_state_dict(_load(exp_dir / "model.pt"))
    if load_optimizer and optimizer is not None \
            and (exp_dir / "optimizer.pt").exists():
        optimizer.load_state_dict(_load(exp_dir / "optimizer.pt"))
    lr_scheduler.load_state_dict(_load(exp_dir / "lr_scheduler.pt"))
    with open(exp_dir / "state.json") as fp:
        return json.load(fp)


# ---------------- sharded (chapters 4-7) ----------------

def save_sharded(exp_dir: Path, model_sd: dict, optim_sd: dict,
                 lr_scheduler, state, rank: int, world_size: int):
    """Every rank writes its own shard file; rank 0 writes metadata +
    scheduler + state.json.  model_sd/optim_sd are this rank's LOCAL shard
    state dicts (cpu tensors)."""
    ckpt = exp_dir / "checkpoint"
    mkdir_rank0(ckpt, rank)
    torch.save({"model": model_sd, "optimizer": optim_sd},
               ckpt / f"shard_rank{rank}.pt")
    if rank == 0:
        meta = {
            "world_size": world_size,
            "format": "dtg_amd_sharded_v1",
            "model_keys": sorted(model_sd.keys()),
        }
        with open(ckpt / "metadata.json", "w") as fp:
            json.dump(meta, fp)
        torch.save(lr_scheduler.state_dict(), exp_dir / "lr_scheduler.pt")
        with open(exp_dir / "state.json", "w") as fp:
            json.dump(state, fp)
    _barrier()


def load_sharded(exp_dir: Path, rank: int, world_size: int):
    """Returns (model_sd, optim_sd, state). World size must match the saved
    one for the direct per-rank path; resharding is handled by the caller
    via gather_full/reshard helpers in the FSDP engine."""
    ckpt = exp_dir / "checkpoint"
    with open(ckpt / "metadata.json") as fp:
        meta = json.load(fp)
    if meta["world_size"] != world_size:
        raise RuntimeError(
            f"checkpoint saved at world_size={meta['world_size']}, "
            f"loading at {world_size}: reshard via the engine's full-state "
            "path")
    blob = torch.load(ckpt / f"shard_rank{rank}.pt", map_location="cpu",
                      weights_only=True)
    with open(exp_dir / "state.json") as fp:
        state = json.load(fp)
    return blob["model"], blob["optimizer"], state


# ---------------- RNG (determinism recipe) ----------------

def save_rng(exp_dir: Path, rank: int):
    rng = {
        "torch": torch.get_rng_state(),
        "numpy": np.random.get_state(),
        "random": random.getstate(),
    }
    if torch.cuda.is_available():
        rng["cuda"] = torch.cuda.get_rng_state()
    torch.save(rng, exp_dir / f"rng_rank{rank}.pt")


def load_rng(exp_dir: Path, rank: int):
    p = exp_dir / f"rng_rank{rank}.pt"
    if not p.exists():
        return False
    rng = torch.load(p, map_location="cpu", weights_only=False)
    torch.set_rng_state(rng["torch"])
    np.random.set_state(rng["numpy"])
    random.setstate(rng["random"])
    if "cuda" in rng and torch.cuda.is_available():
        torch.cuda.set_rng_state(rng["cuda"])
    return True
