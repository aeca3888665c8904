#!/usr/bin/env python3
"""Alternative-frameworks chapter: the same trainer skeleton on a
config-driven ZeRO-3 engine, with wandb experiment tracking.

MI355X-native counterpart of
/root/reference/alternative-frameworks/deepspeed/train_llm.py: where the
reference hands the model to `deepspeed.initialize` with `ds_config.json`
(ZeRO stage-3 bf16, `:58-73`), this uses
distributed_training_guide_amd.engine.initialize — the same engine API
(`engine(**batch)` / `engine.backward(loss)` / `engine.step()` /
`engine.save_checkpoint`, reference `:147-155,94-96,193-196`) implemented on
this repo's FSDP flat-param sharding + fused HIP AdamW + RCCL collectives.
It is also the chapter that carries the reference's only in-tree wandb
integration (`:110-124,185-186`): rank-0 `wandb.init(project=...,
id=experiment_name, resume="must" if resumed)` + per-log-freq `wandb.log`.

    torchrun --standalone --nproc-per-node 8 \
        alternative-frameworks/zero3-engine/train_llm.py \
        -m llama-3-8b -d synthetic --engine-config engine_config.json
"""
import json
import logging
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent.parent))

import torch
from torch.distributed.elastic.multiprocessing.errors import record
from torch.utils.data import DataLoader

from distributed_training_guide_amd import engine as engine_mod
from distributed_training_guide_amd.data import (DistributedSampler,
                                                 default_collate,
                                                 load_and_preprocess_data)
from distributed_training_guide_amd.models import build_model, get_config
from distributed_training_guide_amd.parallel.pg import destroy
from distributed_training_guide_amd.trainer import get_parser
from distributed_training_guide_amd.utils.checkpoint import mkdir_rank0
from distributed_training_guide_amd.utils.logging import setup_logging
from distributed_training_guide_amd.utils.timers import (LocalTimer,
                                                         get_mem_stats,
                                                         reset_peak_memory_stats)

LOGGER = logging.getLogger(__name__)


def build_parser():
    p = get_parser()
    p.add_argument("--engine-config",
                   default=str(Path(__file__).parent / "engine_config.json"),
                   help="JSON engine config (ds_config.json equivalent)")
    return p


@record
def main(argv=None):
    args = build_parser().parse_args(argv)

    with open(args.engine_config) as fp:
        config_json = json.load(fp)
    # CLI batch size wins over the config file (reference passes batch size
    # through ds_config's train_micro_batch_size_per_gpu)
    config_json["train_micro_batch_size_per_gpu"] = args.batch_size
    config_json["gradient_accumulation_steps"] = max(
        1, getattr(args, "grad_accum_steps", 1))

    model_config = get_config(args.model_name)
    engine, _, _, lr_scheduler = engine_mod.initialize(
        config_json,
        model_factory=lambda dtype: build_model(model_config, dtype=dtype))
    rank, world_size = engine.rank, engine.world_size
    device = engine.device
    setup_logging(rank)
    LOGGER.info(args)
    torch.manual_seed(args.seed)

    train_data = load_and_preprocess_data(args, model_config)
    sampler = DistributedSampler(train_data, num_replicas=world_size,
                                 rank=rank, shuffle=True, seed=args.seed,
                                 drop_last=True)
    dataloader = DataLoader(train_data, batch_size=args.batch_size,
                            sampler=sampler, drop_last=True,
                            collate_fn=default_collate)

    exp_dir = Path(args.save_dir)
    is_experiment = args.experiment_name is not None
    if is_experiment:
        exp_dir = exp_dir / args.experiment_name

    state = {"epoch": 0, "epoch_step": 0, "running_loss": 0.0}
    resumed = False
    if is_experiment and (exp_dir / "state.json").exists():
        state = engine.load_checkpoint(exp_dir)
        resumed = True
    if is_experiment:
        LOGGER.info(f"Resumed={resumed} | {state}")
        mkdir_rank0(exp_dir, rank)

    # wandb on rank 0 with resume="must" (reference `:110-124`)
    wandb_run = None
    if getattr(args, "wandb", False) and rank == 0:
        try:
            import wandb

            wandb_run = wandb.init(
                project="distributed-training-guide-amd",
                id=args.experiment_name,
                resume="must" if resumed else None,
                config=dict(vars(args), engine_config=config_json))
        except ImportError:
            LOGGER.warning("wandb not installed; --wandb ignored")

    timers = {k: LocalTimer(device)
              for k in ["data", "forward", "backward", "update"]}
    accum = engine.accum
    tok_per_step = args.batch_size * args.seq_length * world_size * accum
    max_steps = getattr(args, "max_steps", 0)
    done = False

    for state["epoch"] in range(state["epoch"], args.num_epochs):
        if done:
            break
        sampler.set_epoch(state["epoch"])
        batches = iter(dataloader)
        n_steps = len(dataloader) // accum
        for i_step in range(n_steps):
            micro = []
            with timers["data"], torch.no_grad():
                for _ in range(accum):
                    batch = next(batches)
                    micro.append({k: v.to(device) for k, v in batch.items()})
            if i_step < state["epoch_step"]:
                continue

            total_loss = 0.0
            for batch in micro:
                with timers["forward"]:
                    outputs = engine(**batch)
                with timers["backward"]:
                    engine.backward(outputs.loss)
                total_loss += outputs.loss.item() / accum
            with timers["update"]:
                engine.step()

            state["epoch_step"] += 1
            state["running_loss"] += total_loss

            if engine.global_step % args.log_freq == 0:
                ms = sum(t.avg_elapsed_ms() for t in timers.values())
                info = {
                    "global_step": engine.global_step,
                    "lr": lr_scheduler.get_last_lr()[0],
                    "running_loss": state["running_loss"] / args.log_freq,
                    "epoch": state["epoch"],
                    **get_mem_stats(device),
                    "tokens_per_s": 1000 * tok_per_step / max(ms, 1e-9),
                    **{f"time/{k}": t.avg_elapsed_ms()
                       for k, t in timers.items()},
                }
                LOGGER.info(info)
                if wandb_run is not None:
                    wandb_run.log(info, step=engine.global_step)
                reset_peak_memory_stats(device)
                state["running_loss"] = 0.0
                for t in timers.values():
                    t.reset()

            if is_experiment and engine.global_step % args.ckpt_freq == 0:
                engine.save_checkpoint(exp_dir, dict(state))

            if max_steps and engine.global_step >= max_steps:
                done = True
                break
        if not done:
            state["epoch_step"] = 0

    if wandb_run is not None:
        wandb_run.finish()
    destroy()
    return state


if __name__ == "__main__":
    main()
