"""Chapter-2 strategy: DDP + ZeRO-1 over RCCL/xGMI
(/root/reference/02-distributed-data-parallel/train_llm.py)."""
from pathlib import Path

import torch

from ..models import build_model
from ..ops import FusedAdamW
from ..trainer import pick_device
from ..utils import checkpoint as ckpt
from .ddp import DistributedDataParallel
from .pg import env_local_rank, init_distributed
from .zero1 import ZeroRedundancyOptimizer


class DDPStrategy:
    def __init__(self, args):
        self.local_rank = env_local_rank()
        self.device = pick_device(args, self.local_rank)
        if self.device.type == "cuda":
            torch.cuda.set_device(self.device)
        self.rank, _, self.world_size = init_distributed(self.device)
        self.dp_rank = self.rank
        self.dp_size = self.world_size
        self.dtype = torch.bfloat16
        self.use_zero1 = getattr(args, "zero1", True)

    def build(self, config, args):
        model = build_model(config, device=self.device, dtype=self.dtype)
        model = DistributedDataParallel(
            model, bucket_cap_mb=getattr(args, "bucket_cap_mb", 128))
        if self.use_zero1 and self.world_size > 1:
            optimizer = ZeroRedundancyOptimizer(
                model.parameters(), optimizer_class=FusedAdamW, lr=args.lr)
        else:
            optimizer = FusedAdamW(model.parameters(), lr=args.lr)
        lr_scheduler = torch.optim.lr_scheduler.CosineAnnealingLR(
            optimizer, T_max=1000, eta_min=args.lr * 1e-2)
        return model, optimizer, lr_scheduler

    def no_sync(self, model):
        return model.no_sync()

    def save_checkpoint(self, exp_dir: Path, model, optimizer, lr_scheduler,
                        state):
        # rank-0 unsharded files; optimizer state dropped under ZeRO-1
        # (reference 02-.../README.md:308 drops optimizer.pt there too)
        ckpt.save_unsharded(exp_dir, model, optimizer, lr_scheduler, state,
                            rank=self.rank,
                            save_optimizer=not self.use_zero1)

    def load_checkpoint(self, exp_dir: Path, model, optimizer, lr_scheduler):
        return ckpt.load_unsharded(exp_dir, model, optimizer, lr_scheduler,
                                   self.device,
                                   load_optimizer=not self.use_zero1)
