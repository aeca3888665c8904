"""Chapters 4/5 strategy: FSDP full-sharding (+ CPU offload, activation
checkpointing, prefetch depth, broadcast init)
(/root/reference/04-fully-sharded-data-parallel/train_llm.py,
05-training-llama-405b/train_llm.py)."""
import logging
import os
from pathlib import Path

import torch

from ..models import build_model
from ..ops import FusedAdamW
from ..trainer import pick_device
from ..utils import checkpoint as ckpt
from .fsdp import FSDP, apply_activation_checkpointing
from .pg import env_local_rank, init_distributed

LOGGER = logging.getLogger(__name__)


class FSDPStrategy:
    def __init__(self, args):
        self.local_rank = env_local_rank()
        self.device = pick_device(args, self.local_rank)
        if self.device.type == "cuda":
            torch.cuda.set_device(self.device)
        self.rank, _, self.world_size = init_distributed(self.device)
        self.dp_rank = self.rank
        self.dp_size = self.world_size
        self.dtype = torch.bfloat16
        self.cpu_offload = getattr(args, "cpu_offload", False)
        if self.cpu_offload:
            # leave CPU threads for the offloaded optimizer (reference
            # 05:69-72 tunes threads per rank)
            ngpus = max(torch.cuda.device_count(), 1)
            torch.set_num_threads(max(1, (os.cpu_count() or 8) // ngpus))

    def build(self, config, args):
        # meta init -> FSDP materializes unit by unit (peak = one unit)
        with torch.device("meta"):
            model = build_model(config, dtype=self.dtype)
        if getattr(args, "checkpoint_activations", False):
            apply_activation_checkpointing(model)
        model = FSDP(model, device=self.device,
                     cpu_offload=self.cpu_offload,
                     reduce_dtype=torch.float32)
        depth = getattr(args, "prefetch_layers", 1) or 1
        model.set_prefetch_depth(depth)
        weights_dir = getattr(args, "weights_dir", None)
        if weights_dir:
            # ch-5 pretrained path: rank 0 loads node-local safetensors
            # shards (written by 05-.../download.py) on CPU and broadcasts
            # (reference 05:76-85 from_pretrained + 05:118-126 broadcast)
            full = {}
            if self.rank == 0:
                full = _load_safetensors_dir(Path(weights_dir),
                                             args.model_name)
            model.load_full_state_dict(full, broadcast_from_rank0=True)
        elif getattr(args, "broadcast_init", False):
            # exercise the ch-5 pretrained-init path: rank 0 materializes a
            # full state dict on CPU and broadcasts shards (05:118-126)
            full = model.full_state_dict(rank0_only=True, offload_to_cpu=True)
            model.load_full_state_dict(full, broadcast_from_rank0=True)
        optimizer = FusedAdamW(model.parameters(), lr=args.lr)
        lr_scheduler = torch.optim.lr_scheduler.CosineAnnealingLR(
            optimizer, T_max=1000, eta_min=args.lr * 1e-2)
        return model, optimizer, lr_scheduler

    def no_sync(self, model):
        return model.no_sync()

    # ---- sharded checkpointing (reference 04:241-255; DCP-style layout) --
    def save_checkpoint(self, exp_dir: Path, model, optimizer, lr_scheduler,
                        state):
        ckpt.save_sharded(exp_dir, model.sharded_state_dict(),
                          ckpt.optim_sd_cpu(optimizer), lr_scheduler, state,
                          self.rank, self.world_size)

    def load_checkpoint(self, exp_dir: Path, model, optimizer, lr_scheduler):
        try:
            model_sd, optim_sd, state = ckpt.load_sharded(
                exp_dir, self.rank, self.world_size)
            model.load_sharded_state_dict(model_sd)
            optimizer.load_state_dict(optim_sd)
        except RuntimeError as e:
            if "reshard" not in str(e):
                raise
            state = _load_resharding(exp_dir, model, optimizer, self.rank)
        sched_sd = torch.load(exp_dir / "lr_scheduler.pt",
                              map_location="cpu", weights_only=True)
        lr_scheduler.load_state_dict(sched_sd)
        return state


def _load_resharding(exp_dir: Path, model, optimizer, rank):
    """World-size-changed load: reassemble each unit's full flat param and
    moments from the old per-rank shard files, then re-slice for the new
    world size (the reference leans on torch DCP's planner for this;
    our layout makes it a concat + slice)."""
    import json

    ckpt_dir = exp_dir / "checkpoint"
    with open(ckpt_dir / "metadata.json") as fp:
        meta = json.load(fp)
    w_old = meta["world_size"]
    blobs = [torch.load(ckpt_dir / f"shard_rank{r}.pt", map_location="cpu",
                        weights_only=True) for r in range(w_old)]
    LOGGER.info(f"resharding checkpoint from world_size={w_old} to "
                f"{model.world}")
    # model params
    full_model = {}
    for u in model.units:
        full = torch.cat([blobs[r]["model"][u.name] for r in range(w_old)])
        full_model[u.name] = full
        u.load_full_flat_param_from_concat(full)
    # optimizer moments: state keyed by param index, units order
    opt_sds = [b["optimizer"] for b in blobs]
    new_sd = {"state": {}, "param_groups": opt_sds[0]["param_groups"]}
    for i, u in enumerate(model.units):
        merged = {}
        for key in ("exp_avg", "exp_avg_sq"):
            if i in opt_sds[0]["state"] or str(i) in opt_sds[0]["state"]:
                def get(sd):
                    st = sd["state"]
                    return st[i] if i in st else st[str(i)]
                full = torch.cat([get(sd)[key] for sd in opt_sds])
                total = u.total
                full = full[:total]
                pad = u.padded - total
                if pad:
                    full = torch.cat([full, torch.zeros(pad, dtype=full.dtype)])
                merged[key] = full[u.rank * u.shard_numel:
                                   (u.rank + 1) * u.shard_numel].clone()
        if merged:
            new_sd["state"][i] = merged
    # fix param_groups "params" indices to the live optimizer's
    new_sd["param_groups"] = optimizer.state_dict()["param_groups"]
    optimizer.load_state_dict(new_sd)
    with open(exp_dir / "state.json") as fp:
        return json.load(fp)


def _load_safetensors_dir(weights_dir: Path, model_name: str):
    """Load a node-local safetensors shard directory written by
    05-training-llama-405b/download.py into a full CPU state dict
    (reference 05:76-85: from_pretrained on rank 0 CPU only)."""
    import json as _json

    from safetensors.torch import load_file

    d = weights_dir / model_name if (weights_dir / model_name).is_dir() \
        else weights_dir
    index = _json.loads((d / "model.safetensors.index.json").read_text())
    full = {}
    for fname in sorted(set(index["weight_map"].values())):
        full.update(load_file(str(d / fname)))
    return full
