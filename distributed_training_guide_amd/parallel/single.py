"""Chapter-1 strategy: one device, no process group
(/root/reference/01-single-gpu/train_llm.py)."""
from contextlib import nullcontext
from pathlib import Path

import torch

from ..models import build_model
from ..ops import FusedAdamW
from ..trainer import pick_device
from ..utils import checkpoint as ckpt


class SingleDeviceStrategy:
    def __init__(self, args):
        self.rank = 0
        self.local_rank = 0
        self.world_size = 1
        self.dp_rank = 0
        self.dp_size = 1
        self.device = pick_device(args, 0)
        self.dtype = torch.bfloat16
        if self.device.type == "cuda":
            torch.cuda.set_device(self.device)

    def build(self, config, args):
        model = build_model(config, device=self.device, dtype=self.dtype)
        optimizer = FusedAdamW(model.parameters(), lr=args.lr)
        lr_scheduler = torch.optim.lr_scheduler.CosineAnnealingLR(
            optimizer, T_max=1000, eta_min=args.lr * 1e-2)
        return model, optimizer, lr_scheduler

    def no_sync(self, model):
        return nullcontext()

    def save_checkpoint(self, exp_dir: Path, model, optimizer, lr_scheduler,
                        state):
        ckpt.save_unsharded(exp_dir, model, optimizer, lr_scheduler, state,
                            rank=0)

    def load_checkpoint(self, exp_dir: Path, model, optimizer, lr_scheduler):
        return ckpt.load_unsharded(exp_dir, model, optimizer, lr_scheduler,
                                   self.device)
