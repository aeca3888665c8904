"""Config-driven ZeRO-3 training engine — the MI355X-native counterpart of
the reference's DeepSpeed alternative
(/root/reference/alternative-frameworks/deepspeed/train_llm.py:30-196 and
ds_config.json:1-25).

The reference wraps the same trainer skeleton around `deepspeed.initialize`:
a JSON config chooses micro-batch size / bf16 / ZeRO stage, and the engine
object owns forward, `engine.backward(loss)`, `engine.step()` (optimizer +
LR schedule + zero_grad inside) and `engine.save_checkpoint/load_checkpoint`.
Here the same API shape is implemented on THIS repo's FSDP flat-param
engine (parallel/fsdp.py — ZeRO stage 3 semantics: param + grad + optimizer
state sharding with reduce-scatter/all-gather over RCCL/xGMI), the fused
HIP AdamW, and the sharded checkpoint layout of utils/checkpoint.py.

    engine, _, _, lr_scheduler = initialize(
        config="engine_config.json",
        model_factory=lambda dtype: build_model(cfg, dtype=dtype))
    loss = engine(**batch).loss
    engine.backward(loss)
    engine.step()
"""
import json
import logging
from pathlib import Path

import torch

from .ops import FusedAdamW
from .parallel.fsdp import FSDP, apply_activation_checkpointing
from .parallel.pg import env_local_rank, init_distributed
from .utils import checkpoint as ckpt

LOGGER = logging.getLogger(__name__)

DEFAULT_CONFIG = {
    "train_micro_batch_size_per_gpu": 1,
    "gradient_accumulation_steps": 1,
    "bf16": {"enabled": True},
    "zero_optimization": {
        "stage": 3,
        "overlap_comm": True,
        "reshard_after_forward": True,
        "offload_optimizer": {"device": "none"},
    },
    "optimizer": {
        "type": "AdamW",
        "params": {"lr": 3e-5, "betas": [0.9, 0.999], "eps": 1e-8,
                   "weight_decay": 0.0},
    },
    "scheduler": {"type": "CosineAnnealing", "params": {"t_max": 1000}},
    "activation_checkpointing": {"enabled": False},
}


def _merge(base: dict, override: dict) -> dict:
    out = dict(base)
    for k, v in override.items():
        if isinstance(v, dict) and isinstance(out.get(k), dict):
            out[k] = _merge(out[k], v)
        else:
            out[k] = v
    return out


class Engine(torch.nn.Module):
    """ZeRO-3 engine: owns the sharded model, optimizer, LR schedule and
    loss scaling for gradient accumulation — the call surface of the
    reference's deepspeed engine (fwd `:147`, backward `:153`, step `:155`,
    save/load checkpoint `:94-96,193-196`)."""

    def __init__(self, model, config: dict, device: torch.device):
        super().__init__()
        self.config = config
        self.device = device
        zero = config["zero_optimization"]
        if zero.get("stage", 3) != 3:
            raise ValueError("this engine implements ZeRO stage 3 only "
                             "(stages 0-1 = chapters 2; stage 3 = FSDP)")
        if config["activation_checkpointing"]["enabled"]:
            apply_activation_checkpointing(model)
        offload = zero.get("offload_optimizer", {}).get("device") == "cpu"
        self.module = FSDP(
            model, device=device, cpu_offload=offload,
            reshard_after_forward=zero.get("reshard_after_forward", True),
            reduce_dtype=torch.float32)
        self.cpu_offload = offload
        opt = config["optimizer"]["params"]
        self.optimizer = FusedAdamW(
            self.module.parameters(), lr=opt["lr"],
            betas=tuple(opt.get("betas", (0.9, 0.999))),
            eps=opt.get("eps", 1e-8),
            weight_decay=opt.get("weight_decay", 0.0))
        self.lr_scheduler = torch.optim.lr_scheduler.CosineAnnealingLR(
            self.optimizer,
            T_max=config["scheduler"]["params"].get("t_max", 1000),
            eta_min=opt["lr"] * 1e-2)
        self.accum = max(1, config["gradient_accumulation_steps"])
        self._micro = 0
        self.global_step = 0
        self.rank = self.module.rank
        self.world_size = self.module.world

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    def backward(self, loss):
        """Scales by 1/accum and skips the grad reduce-scatter on
        non-boundary microbatches (deepspeed semantics `:153`)."""
        loss = loss / self.accum
        if self._micro < self.accum - 1:
            with self.module.no_sync():
                loss.backward()
        else:
            loss.backward()
        self._micro += 1

    def is_gradient_accumulation_boundary(self) -> bool:
        return self._micro >= self.accum

    def step(self):
        """Optimizer + LR schedule + zero_grad, only at accumulation
        boundaries (the engine owns the schedule, reference `:155`)."""
        if not self.is_gradient_accumulation_boundary():
            return
        self.optimizer.step()
        self.lr_scheduler.step()
        self.optimizer.zero_grad(set_to_none=not self.cpu_offload)
        self._micro = 0
        self.global_step += 1

    # -- checkpointing (reference `:94-96,193-196`): sharded, all ranks --
    def save_checkpoint(self, save_dir, client_state: dict = None):
        exp_dir = Path(save_dir)
        state = dict(client_state or {})
        state["global_step"] = self.global_step
        ckpt.save_sharded(exp_dir, self.module.sharded_state_dict(),
                          ckpt.optim_sd_cpu(self.optimizer), self.lr_scheduler,
                          state, self.rank, self.world_size)

    def load_checkpoint(self, save_dir):
        exp_dir = Path(save_dir)
        if not (exp_dir / "state.json").exists():
            return None
        model_sd, optim_sd, state = ckpt.load_sharded(
            exp_dir, self.rank, self.world_size)
        self.module.load_sharded_state_dict(model_sd)
        self.optimizer.load_state_dict(optim_sd)
        sched = torch.load(exp_dir / "lr_scheduler.pt", map_location="cpu",
                           weights_only=True)
        self.lr_scheduler.load_state_dict(sched)
        self.global_step = state.get("global_step", 0)
        return state


def initialize(config, model_factory, device=None):
    """deepspeed.initialize-shaped entry (reference `:68-73`): returns
    (engine, optimizer, dataloader_placeholder, lr_scheduler)."""
    if isinstance(config, (str, Path)):
        with open(config) as fp:
            config = json.load(fp)
    config = _merge(DEFAULT_CONFIG, config or {})
    local_rank = env_local_rank()
    if device is None:
        device = (torch.device(f"cuda:{local_rank}")
                  if torch.cuda.is_available() else torch.device("cpu"))
    if device.type == "cuda":
        torch.cuda.set_device(device)
    init_distributed(device)
    dtype = torch.bfloat16 if config["bf16"]["enabled"] else torch.float32
    with torch.device("meta"):
        model = model_factory(dtype)
    engine = Engine(model, config, device)
    return engine, engine.optimizer, None, engine.lr_scheduler
