"""Chapters 6/7 strategies: TP+SP (/root/reference/06-tensor-parallel/
train_llm.py) and 2D FSDP x TP (07-2d-parallel/train_llm.py)."""
import logging
from pathlib import Path

import torch

from ..ops import FusedAdamW
from ..trainer import pick_device
from ..utils import checkpoint as ckpt
from .ddp import DistributedDataParallel
from .fsdp import FSDP, apply_activation_checkpointing
from .mesh import DeviceMesh2D
from .pg import env_local_rank, init_distributed
from .tp import TPLlamaDecoderLayer, TPLlamaForCausalLM

LOGGER = logging.getLogger(__name__)


class TPStrategy:
    """Chapter 6: TP across the whole (or inner) mesh dim, DP across nodes
    via the DDP bucket engine when dp > 1."""

    def __init__(self, args, tp_size=None):
        self.local_rank = env_local_rank()
        self.device = pick_device(args, self.local_rank)
        if self.device.type == "cuda":
            torch.cuda.set_device(self.device)
        self.rank, _, self.world_size = init_distributed(self.device)
        tp = tp_size or getattr(args, "tensor_parallel", 0) or self.world_size
        self.mesh = DeviceMesh2D(tp_size=tp)
        LOGGER.info(f"{self.mesh}")
        self.dp_rank = self.mesh.dp_rank
        self.dp_size = self.mesh.dp_size
        self.dtype = torch.bfloat16
        self.loss_parallel = getattr(args, "loss_parallel", False)

    def build(self, config, args):
        model = TPLlamaForCausalLM(config, self.mesh, device=self.device,
                                   dtype=self.dtype,
                                   loss_parallel=self.loss_parallel)
        if self.mesh.dp_size > 1:
            model = DistributedDataParallel(
                model, bucket_cap_mb=getattr(args, "bucket_cap_mb", 128),
                process_group=self.mesh.dp_group)
        optimizer = FusedAdamW(model.parameters(), lr=args.lr)
        lr_scheduler = torch.optim.lr_scheduler.CosineAnnealingLR(
            optimizer, T_max=1000, eta_min=args.lr * 1e-2)
        return model, optimizer, lr_scheduler

    def no_sync(self, model):
        from contextlib import nullcontext

        return model.no_sync() if hasattr(model, "no_sync") else nullcontext()

    def save_checkpoint(self, exp_dir: Path, model, optimizer, lr_scheduler,
                        state):
        # DCP-style: every rank writes its tp shard (06:171-174)
        inner = model.module if hasattr(model, "module") else model
        ckpt.save_sharded(exp_dir, inner.tp_state_dict(),
                          ckpt.optim_sd_cpu(optimizer), lr_scheduler,
                          state, self.rank, self.world_size)

    def load_checkpoint(self, exp_dir: Path, model, optimizer, lr_scheduler):
        inner = model.module if hasattr(model, "module") else model
        try:
            msd, osd, state = ckpt.load_sharded(exp_dir, self.rank,
                                                self.world_size)
            inner.load_tp_state_dict(msd)
            optimizer.load_state_dict(osd)
        except RuntimeError as e:
            if "reshard" not in str(e):
                raise
            state = _load_tp_resharding(exp_dir, inner, optimizer)
        sched_sd = torch.load(exp_dir / "lr_scheduler.pt",
                              map_location="cpu", weights_only=True)
        lr_scheduler.load_state_dict(sched_sd)
        return state


def _load_tp_resharding(exp_dir: Path, inner, optimizer):
    """tp-size-changed load (ch6: tp == world): reconstruct every full
    tensor from the OLD ranks' shard files and re-slice for the new tp —
    model weights via the module sharding rules, optimizer moments via
    the same rules (they follow their parameter's layout).  The reference
    leans on torch DCP's planner for this."""
    import json

    ckpt_dir = exp_dir / "checkpoint"
    with open(ckpt_dir / "metadata.json") as fp:
        meta = json.load(fp)
    w_old = meta["world_size"]
    LOGGER.info(f"resharding TP checkpoint from tp={w_old} to "
                f"{inner.mesh.tp_size}")
    blobs = [torch.load(ckpt_dir / f"shard_rank{r}.pt", map_location="cpu",
                        weights_only=True) for r in range(w_old)]
    inner.load_tp_state_dict_resharded([b["model"] for b in blobs])
    # optimizer moments: state keyed by param index in parameters() order
    # (== named_parameters order)
    names = [n for n, _ in inner.named_parameters()]
    opt_sds = [b["optimizer"] for b in blobs]

    def get(sd, i):
        st = sd["state"]
        return st[i] if i in st else st.get(str(i))

    new_state = {}
    for i, name in enumerate(names):
        if get(opt_sds[0], i) is None:
            continue
        merged = {}
        for key in ("exp_avg", "exp_avg_sq"):
            full = inner.reconstruct_full_tensor(
                name, [get(sd, i)[key] for sd in opt_sds])
            merged[key] = inner.shard_tensor(name, full)
        new_state[i] = merged
    new_sd = {"state": new_state,
              "param_groups": optimizer.state_dict()["param_groups"]}
    optimizer.load_state_dict(new_sd)
    with open(exp_dir / "state.json") as fp:
        return json.load(fp)


class TwoDStrategy(TPStrategy):
    """Chapter 7: FSDP over the dp mesh dim layered over TP (07:121-123).

    Checkpoints are shard-of-shard (each (dp, tp) rank's FSDP shard of its
    local tp shard); loading requires the SAME (dp, tp) — 1-D resharding
    exists for pure FSDP (fsdp_strategy._load_resharding) and pure TP
    (_load_tp_resharding above), 2-D re-meshing is not implemented (the
    reference gets it from torch DCP's planner)."""

    def __init__(self, args):
        super().__init__(args, tp_size=getattr(args, "tensor_parallel", 8))

    def build(self, config, args):
        model = TPLlamaForCausalLM(config, self.mesh, device=self.device,
                                   dtype=self.dtype,
                                   loss_parallel=self.loss_parallel)
        if getattr(args, "checkpoint_activations", False):
            apply_activation_checkpointing(model,
                                           layer_cls=TPLlamaDecoderLayer)
        model = FSDP(model, layer_cls=TPLlamaDecoderLayer,
                     process_group=self.mesh.dp_group, device=self.device,
                     reduce_dtype=torch.float32,
                     cpu_offload=getattr(args, "cpu_offload", False))
        optimizer = FusedAdamW(model.parameters(), lr=args.lr)
        lr_scheduler = torch.optim.lr_scheduler.CosineAnnealingLR(
            optimizer, T_max=1000, eta_min=args.lr * 1e-2)
        return model, optimizer, lr_scheduler

    def no_sync(self, model):
        return model.no_sync()

    def save_checkpoint(self, exp_dir: Path, model, optimizer, lr_scheduler,
                        state):
        # shard-of-shard: each (dp, tp) rank saves its FSDP shard of the
        # local tp shard (the reference's 2D DCP save, 07:172-175)
        ckpt.save_sharded(exp_dir, model.sharded_state_dict(),
                          ckpt.optim_sd_cpu(optimizer), lr_scheduler,
                          state, self.rank, self.world_size)

    def load_checkpoint(self, exp_dir: Path, model, optimizer, lr_scheduler):
        msd, osd, state = ckpt.load_sharded(exp_dir, self.rank,
                                            self.world_size)
        model.load_sharded_state_dict(msd)
        optimizer.load_state_dict(osd)
        sched_sd = torch.load(exp_dir / "lr_scheduler.pt",
                              map_location="cpu", weights_only=True)
        lr_scheduler.load_state_dict(sched_sd)
        return state
