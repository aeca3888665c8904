"""Shared trainer loop (L4 of SURVEY.md §1) behind every chapter entrypoint.

The reference repeats this loop byte-identically in every chapter
(e.g. /root/reference/01-single-gpu/train_llm.py:115-189); here it is one
implementation parameterized by a parallelism Strategy (L3).  Observable
behavior matches the reference: CLI surface (01:289-303), phase timers
data/forward/backward/update with device-sync fences (01:113), the
per-log-freq info dict (01:155-172), checkpoint layout + resume batch
skipping (01:86-106,133-135,181-187), epoch/sampler semantics (02:137).

Extras the reference ships as diff-recipes are first-class flags:
--grad-accum-steps (related-topics/gradient-accumulation, incl. no_sync),
--deterministic + RNG checkpointing (related-topics/determinism),
--wandb (wandb-configurations topologies).
"""
import argparse
import logging
import os
from contextlib import nullcontext
from pathlib import Path

import torch
from torch.utils.data import DataLoader

from .data import DistributedSampler, default_collate, load_and_preprocess_data
from .models import get_config
from .utils.checkpoint import load_rng, save_rng
from .utils.logging import setup_logging
from .utils.timers import LocalTimer, get_mem_stats, reset_peak_memory_stats

LOGGER = logging.getLogger(__name__)


def get_parser(extra: bool = True) -> argparse.ArgumentParser:
    """Reference CLI surface (01:289-303) + chapter extras."""
    p = argparse.ArgumentParser()
    p.add_argument("-e", "--experiment-name", default=None)
    p.add_argument("-d", "--dataset-name", default="synthetic")
    p.add_argument("--dataset-subset", default=None)
    p.add_argument("-m", "--model-name", default=None, required=True)
    p.add_argument("--save-dir", default="../outputs")
    p.add_argument("--seed", default=0, type=int)
    p.add_argument("--num-epochs", default=100, type=int)
    p.add_argument("--lr", default=3e-5, type=float)
    p.add_argument("-b", "--batch-size", default=1, type=int)
    p.add_argument("--log-freq", default=10, type=int)
    p.add_argument("--ckpt-freq", default=500, type=int)
    p.add_argument("-s", "--seq-length", default=1024, type=int)
    if extra:
        p.add_argument("--num-samples", default=4096, type=int,
                       help="synthetic dataset size")
        p.add_argument("--max-steps", default=0, type=int,
                       help="stop after N optimizer steps (0 = unlimited)")
        p.add_argument("--grad-accum-steps", default=1, type=int)
        p.add_argument("--deterministic", action="store_true")
        p.add_argument("--wandb", action="store_true")
        p.add_argument("--wandb-mode", default="rank0",
                       choices=["rank0", "node0", "all"])
        p.add_argument("--device", default=None,
                       help="override device (cpu/cuda); default: auto")
        p.add_argument("--num-workers", default=1, type=int)
        p.add_argument("--measure-waiting", action="store_true",
                       help="time an explicit barrier before each data "
                            "fetch to separate straggler waiting from "
                            "loading (optimizing-data-loading recipe)")
    return p


def pick_device(args, local_rank: int) -> torch.device:
    if getattr(args, "device", None):
        return torch.device(args.device)
    if torch.cuda.is_available():
        return torch.device(f"cuda:{local_rank}")
    return torch.device("cpu")


class TrainerState(dict):
    @classmethod
    def fresh(cls):
        return cls(epoch=0, global_step=0, epoch_step=0, running_loss=0.0)


def run_training(args, strategy):
    """The canonical loop. `strategy` provides: rank/world_size/device/dtype,
    dp_rank/dp_size, build(config) -> (model, optimizer, lr_scheduler),
    save_checkpoint/load_checkpoint, no_sync()."""
    setup_logging(strategy.rank)
    LOGGER.debug(os.environ)
    LOGGER.info(args)

    torch.manual_seed(args.seed)
    if getattr(args, "deterministic", False):
        torch.use_deterministic_algorithms(True)

    config = get_config(args.model_name)
    model, optimizer, lr_scheduler = strategy.build(config, args)
    n_local = sum(p.numel() for p in model.parameters())
    # under sharding the local tensor count is NOT the model size; the
    # config knows the true total (reference logs the full count, 01:52)
    n_total = config.num_parameters() if hasattr(config, "num_parameters") \
        else n_local
    LOGGER.info(f"Training {n_total} model parameters "
                f"(local tensors: {n_local})")
    LOGGER.info(f"Initialized model uses "
                f"{get_mem_stats(strategy.device)['curr_alloc_gb']:.3f}gb")

    # local-rank-0 of each node loads first so it populates any node-local
    # dataset/tokenizer cache before its peers read it (reference 02:72-73
    # rank0_first; 06:130 rank_ordered for multi-node HF_HOME)
    from .parallel.pg import rank_ordered

    with rank_ordered(should_go_first=getattr(strategy, "local_rank", 0) == 0):
        train_data = load_and_preprocess_data(args, config)
    LOGGER.info(f"{len(train_data)} training samples")

    sampler = None
    shuffle = True
    if strategy.dp_size > 1 or strategy.world_size > 1:
        sampler = DistributedSampler(
            train_data, num_replicas=strategy.dp_size, rank=strategy.dp_rank,
            shuffle=True, seed=args.seed, drop_last=True)
        shuffle = False
    g = torch.Generator().manual_seed(args.seed)
    dataloader = DataLoader(
        train_data, batch_size=args.batch_size,
        shuffle=shuffle if sampler is None else False, sampler=sampler,
        drop_last=True, num_workers=args.num_workers,
        prefetch_factor=2 if args.num_workers > 0 else None,
        collate_fn=default_collate, generator=g)
    LOGGER.info(f"{len(dataloader)} batches per epoch")

    # experiment dir + resume (reference 01:80-110)
    is_experiment = args.experiment_name is not None
    exp_dir = Path(args.save_dir)
    if is_experiment:
        exp_dir = exp_dir / args.experiment_name

    state = TrainerState.fresh()
    resumed = False
    if is_experiment and (exp_dir / "state.json").exists():
        state = TrainerState(strategy.load_checkpoint(exp_dir, model,
                                                      optimizer, lr_scheduler))
        if getattr(args, "deterministic", False):
            load_rng(exp_dir, strategy.rank)
        resumed = True
    if is_experiment:
        LOGGER.info(f"Resumed={resumed} | {state}")
        from .utils.checkpoint import mkdir_rank0

        mkdir_rank0(exp_dir, strategy.rank)

    wandb_run = _maybe_init_wandb(args, strategy, resumed)

    phases = ["data", "forward", "backward", "update"]
    measure_waiting = (getattr(args, "measure_waiting", False)
                       and strategy.world_size > 1)
    if measure_waiting:
        phases = ["waiting"] + phases
    timers = {k: LocalTimer(strategy.device) for k in phases}
    accum = max(1, getattr(args, "grad_accum_steps", 1))
    max_steps = getattr(args, "max_steps", 0)
    tok_per_step = args.batch_size * args.seq_length * strategy.dp_size * accum
    done = False

    for state["epoch"] in range(state["epoch"], args.num_epochs):
        if done:
            break
        LOGGER.info(f"Begin epoch {state['epoch']} at step "
                    f"{state['epoch_step']}")
        if sampler is not None:
            sampler.set_epoch(state["epoch"])
        batches = iter(dataloader)
        n_batches = len(dataloader)
        # rank-0 progress bar (reference 02:132)
        progress_bar = _tqdm(range(n_batches // accum),
                             disable=strategy.rank > 0,
                             initial=min(state["epoch_step"],
                                         n_batches // accum))

        for i_step in range(n_batches // accum):
            if measure_waiting:
                # all ranks rendezvous BEFORE fetching: time spent here is
                # waiting on stragglers, not on this rank's loader
                import torch.distributed as dist

                with timers["waiting"]:
                    dist.barrier()
            micro = []
            with timers["data"], torch.no_grad():
                for _ in range(accum):
                    batch = next(batches)
                    micro.append({k: v.to(device=strategy.device)
                                  for k, v in batch.items()})
            if i_step < state["epoch_step"]:
                continue  # resume skip (01:133-135)

            total_loss = 0.0
            for mi, batch in enumerate(micro):
                # skip the inter-rank grad sync on non-boundary microbatches
                # (gradient-accumulation recipe, no_sync gating)
                sync_ctx = (strategy.no_sync(model)
                            if mi < accum - 1 else nullcontext())
                with sync_ctx:
                    with timers["forward"]:
                        outputs = model(**batch)
                    with timers["backward"]:
                        loss = outputs.loss / accum
                        loss.backward()
                total_loss += loss.item()
                del outputs, loss
            del micro

            with timers["update"]:
                optimizer.step()
                lr_scheduler.step()
                optimizer.zero_grad(
                    set_to_none=not getattr(strategy, "cpu_offload", False))

            state["global_step"] += 1
            state["epoch_step"] += 1
            state["running_loss"] += total_loss
            progress_bar.update(1)

            if state["global_step"] % args.log_freq == 0:
                ms_per_step = sum(t.avg_elapsed_ms() for t in timers.values())
                info = {
                    "global_step": state["global_step"],
                    "lr": lr_scheduler.get_last_lr()[0],
                    "running_loss": state["running_loss"] / args.log_freq,
                    "epoch": state["epoch"],
                    "epoch_progress": state["epoch_step"] / (n_batches // accum),
                    "num_batches_remaining": n_batches - i_step * accum,
                    **get_mem_stats(strategy.device),
                    "tokens_per_s": 1000 * tok_per_step / max(ms_per_step, 1e-9),
                    "time/total": ms_per_step,
                    **{f"time/{k}": t.avg_elapsed_ms()
                       for k, t in timers.items()},
                }
                LOGGER.info(info)
                if wandb_run is not None:
                    wandb_run.log(info, step=state["global_step"])
                reset_peak_memory_stats(strategy.device)
                state["running_loss"] = 0
                for t in timers.values():
                    t.reset()

            if is_experiment and state["global_step"] % args.ckpt_freq == 0:
                LOGGER.info("Saving checkpoint.")
                strategy.save_checkpoint(exp_dir, model, optimizer,
                                         lr_scheduler, dict(state))
                if getattr(args, "deterministic", False):
                    save_rng(exp_dir, strategy.rank)

            if max_steps and state["global_step"] >= max_steps:
                done = True
                break

        if not done:
            state["epoch_step"] = 0

    if wandb_run is not None:
        wandb_run.finish()
    return state


def _tqdm(iterable, disable=False, initial=0):
    try:
        import tqdm

        return tqdm.tqdm(iterable, disable=disable, initial=initial)
    except ImportError:  # progress display is never load-bearing

        class _Noop:
            def update(self, *_):
                pass

        return _Noop()


def _maybe_init_wandb(args, strategy, resumed):
    """wandb topologies per the reference recipe
    (related-topics/wandb-configurations/README.md:10-63)."""
    if not getattr(args, "wandb", False):
        return None
    mode = getattr(args, "wandb_mode", "rank0")
    local_rank = getattr(strategy, "local_rank", 0)
    should = (strategy.rank == 0 if mode == "rank0" else
              local_rank == 0 if mode == "node0" else True)
    if not should:
        return None
    try:
        import wandb
    except ImportError:
        LOGGER.warning("wandb not installed; --wandb ignored")
        return None
    kwargs = dict(project="distributed-training-guide-amd",
                  config=vars(args))
    if args.experiment_name:
        suffix = "" if mode == "rank0" else f"-rank{strategy.rank}"
        kwargs.update(id=args.experiment_name + suffix,
                      resume="must" if resumed else None)
    if mode != "rank0":
        kwargs["group"] = args.experiment_name
    return wandb.init(**kwargs)
