"""ZeRO-1 optimizer-state sharding — re-implementation of what the
reference gets from torch's ZeroRedundancyOptimizer
(/root/reference/02-distributed-data-parallel/train_llm.py:87-89; rationale
and state-dict caveat 02-.../README.md:295-308).

Each rank keeps AdamW moments only for its greedy-balanced partition of the
parameters, steps that partition, then the updated parameter values are
exchanged with ONE flat all-gather over xGMI (params are packed into equal
padded shards — a single large collective instead of per-tensor broadcasts,
which is what the 7-link point-to-point fabric wants).
"""
import os

import torch
import torch.distributed as dist

from ..ops import FusedAdamW


def _force_collectives():
    """DTGA_FORCE_COLLECTIVES=1: run the post-step shard all-gather even at
    world=1 (see parallel/fsdp.py) so 1-GPU runs execute the exact
    multi-GPU call pattern (in-place all_gather_into_tensor)."""
    return os.environ.get("DTGA_FORCE_COLLECTIVES") == "1"


class ZeroRedundancyOptimizer(torch.optim.Optimizer):
    def __init__(self, params, optimizer_class=FusedAdamW,
                 process_group=None, **kwargs):
        params = [p for p in params if p.requires_grad]
        if not params:
            raise ValueError("no parameters")
        self.group = process_group
        self.world_size = dist.get_world_size(process_group)
        self.rank = dist.get_rank(process_group)

        # greedy balance by numel, deterministic across ranks
        order = sorted(range(len(params)), key=lambda i: -params[i].numel())
        loads = [0] * self.world_size
        self.partitions: list[list[torch.nn.Parameter]] = \
            [[] for _ in range(self.world_size)]
        for i in order:
            r = loads.index(min(loads))
            self.partitions[r].append(params[i])
            loads[r] += params[i].numel()

        self.local_opt = optimizer_class(self.partitions[self.rank], **kwargs)
        # delegate Optimizer surface to the local optimizer (schedulers
        # mutate param_groups[...]["lr"] in place)
        self.defaults = self.local_opt.defaults
        self.param_groups = self.local_opt.param_groups
        self.state = self.local_opt.state
        self._all_params = params

        if self.world_size > 1 or _force_collectives():
            self.shard_numel = max(loads)
            dev = params[0].device
            self.dtype = params[0].dtype
            if any(p.dtype != self.dtype for p in params):
                raise RuntimeError("ZeRO-1 requires uniform param dtype")
            self._gather_buf = torch.empty(
                self.world_size * self.shard_numel, dtype=self.dtype,
                device=dev)

    @torch.no_grad()
    def step(self, closure=None):
        loss = self.local_opt.step(closure)
        if self.world_size > 1 or (_force_collectives()
                                   and getattr(self, "_gather_buf",
                                               None) is not None):
            my = self._gather_buf[self.rank * self.shard_numel:
                                  (self.rank + 1) * self.shard_numel]
            off = 0
            for p in self.partitions[self.rank]:
                my[off: off + p.numel()].copy_(p.view(-1))
                off += p.numel()
            if self._gather_buf.is_cuda:
                dist.all_gather_into_tensor(self._gather_buf, my,
                                            group=self.group)
            else:  # gloo (CPU tests) lacks all_gather_into_tensor
                views = list(self._gather_buf.chunk(self.world_size))
                dist.all_gather(views, my.clone(), group=self.group)
            for r in range(self.world_size):
                if r == self.rank:
                    continue
                shard = self._gather_buf[r * self.shard_numel:
                                         (r + 1) * self.shard_numel]
                off = 0
                for p in self.partitions[r]:
                    p.view(-1).copy_(shard[off: off + p.numel()])
                    off += p.numel()
        return loss

    def zero_grad(self, set_to_none: bool = True):
        for p in self._all_params:
            if p.grad is not None:
                if set_to_none:
                    p.grad = None
                else:
                    p.grad.zero_()

    def state_dict(self):
        # local shard only — the reference DROPS optimizer checkpointing
        # under ZeRO-1 ("exorbitantly slow" full state dict, README 02:308);
        # our sharded checkpoint path saves this per-rank dict instead.
        return self.local_opt.state_dict()

    def load_state_dict(self, sd):
        self.local_opt.load_state_dict(sd)
