"""Bucketed data-parallel gradient synchronization — re-implementation of
what the reference gets from torch's DistributedDataParallel
(/root/reference/02-distributed-data-parallel/train_llm.py:66-68:
bucket_cap_mb, gradient_as_bucket_view; semantics grad/world then SUM per
02-.../README.md:179-185; param broadcast at construction per README:177;
no_sync gating per related-topics/gradient-accumulation/README.md:28-38).

Mechanics (ours, not torch's):
  * params are packed into flat bucket buffers in reverse registration
    order (≈ backward completion order) and p.grad aliases its bucket slice
    (gradient_as_bucket_view always on — one memset instead of per-param
    allocs, and the all-reduce runs on the flat buffer directly).
  * post-accumulate-grad hooks count arrivals per bucket; a complete bucket
    is pre-divided by world size and handed to an async all-reduce that
    overlaps the remaining backward.
  * an end-of-backward engine callback waits on all in-flight work.

xGMI note: each MI355X GPU has 7 point-to-point links at ~153 GB/s; RCCL's
ring all-reduce is per-link bound, so the default bucket is sized for link
saturation while still giving several overlap chunks per backward
(--bucket-cap-mb tunable; the reference's 500 MB was an NVLink choice).
"""
import logging
import os
from contextlib import contextmanager

import torch
import torch.distributed as dist
from torch import nn

LOGGER = logging.getLogger(__name__)


def _force_collectives():
    """DTGA_FORCE_COLLECTIVES=1: run the real RCCL collective branch even
    at world=1 (see parallel/fsdp.py) so 1-GPU runs execute the exact
    multi-GPU call pattern."""
    return os.environ.get("DTGA_FORCE_COLLECTIVES") == "1"


class _Bucket:
    __slots__ = ("params", "flat", "views", "pending", "work")

    def __init__(self):
        self.params = []
        self.flat = None
        self.views = {}
        self.pending = 0
        self.work = None


class DistributedDataParallel(nn.Module):
    def __init__(self, module: nn.Module, bucket_cap_mb: int = 128,
                 process_group=None, broadcast_params: bool = True):
        super().__init__()
        self.module = module
        self.group = process_group
        self.world_size = dist.get_world_size(process_group)
        self.sync_enabled = True
        self._works = []
        self._final_cb_armed = False

        if broadcast_params:
            self._broadcast_module_states()

        # ---- build buckets (reverse registration order) ----
        cap = bucket_cap_mb * 1024 * 1024
        params = [p for p in module.parameters() if p.requires_grad]
        self.buckets: list[_Bucket] = []
        self._param_bucket = {}
        cur = _Bucket()
        size = 0
        for p in reversed(params):
            nbytes = p.numel() * p.element_size()
            if cur.params and size + nbytes > cap:
                self.buckets.append(cur)
                cur = _Bucket()
                size = 0
            cur.params.append(p)
            size += nbytes
        if cur.params:
            self.buckets.append(cur)

        for b in self.buckets:
            total = sum(p.numel() for p in b.params)
            # one dtype per bucket (split if models mix dtypes)
            dtypes = {p.dtype for p in b.params}
            if len(dtypes) != 1:
                raise RuntimeError("mixed dtypes within a bucket")
            b.flat = torch.zeros(total, dtype=b.params[0].dtype,
                                 device=b.params[0].device)
            off = 0
            for p in b.params:
                b.views[p] = b.flat[off: off + p.numel()].view_as(p)
                off += p.numel()
                self._param_bucket[p] = b
            b.pending = len(b.params)

        self._install_grad_views()
        self._hooks = [
            p.register_post_accumulate_grad_hook(self._make_hook(p))
            for b in self.buckets for p in b.params
        ]
        LOGGER.debug(f"DDP: {len(self.buckets)} buckets over {len(params)} "
                     f"params")

    # ---- construction-time state broadcast (reference README 02:177) ----
    def _broadcast_module_states(self):
        if self.world_size == 1 and not _force_collectives():
            return
        with torch.no_grad():
            for t in list(self.module.parameters()) + \
                    list(self.module.buffers()):
                dist.broadcast(t.data, group_src=0, group=self.group)

    def _install_grad_views(self):
        for b in self.buckets:
            for p in b.params:
                if p.grad is None or p.grad.data_ptr() != b.views[p].data_ptr():
                    v = b.views[p]
                    if p.grad is not None:
                        v.copy_(p.grad)
                    else:
                        v.zero_()
                    p.grad = v

    def _make_hook(self, p):
        def hook(param):
            b = self._param_bucket[param]
            if param.grad is not None and \
                    param.grad.data_ptr() != b.views[param].data_ptr():
                # autograd allocated a fresh grad (zero_grad(set_to_none));
                # fold it into the bucket view and re-alias.
                b.views[param].copy_(param.grad)
                param.grad = b.views[param]
            if not self.sync_enabled or (self.world_size == 1
                                         and not _force_collectives()):
                return
            if not self._final_cb_armed:
                self._final_cb_armed = True
                torch.autograd.Variable._execution_engine.queue_callback(
                    self._finalize_backward)
            b.pending -= 1
            if b.pending == 0:
                b.flat.div_(self.world_size)
                from . import xgmi

                if b.flat.is_cuda and xgmi.algo() == "direct":
                    # fully-connected xGMI RS+AG (parallel/xgmi.py)
                    b.work = xgmi.direct_all_reduce(b.flat,
                                                    group=self.group,
                                                    async_op=True)
                else:
                    b.work = dist.all_reduce(b.flat, group=self.group,
                                             async_op=True)
                self._works.append(b.work)
        return hook

    def _finalize_backward(self):
        for w in self._works:
            w.wait()
        self._works.clear()
        for b in self.buckets:
            b.pending = len(b.params)
            b.work = None
        self._final_cb_armed = False

    @contextmanager
    def no_sync(self):
        """Skip gradient all-reduce (gradient-accumulation recipe)."""
        prev = self.sync_enabled
        self.sync_enabled = False
        try:
            yield
        finally:
            self.sync_enabled = prev

    def forward(self, *args, **kwargs):
        # re-alias any grads dropped by zero_grad(set_to_none=True)
        if torch.is_grad_enabled():
            self._install_grad_views()
        return self.module(*args, **kwargs)

    # passthroughs so checkpointing sees the bare module
    def state_dict(self, *a, **kw):
        return self.module.state_dict(*a, **kw)

    def load_state_dict(self, sd, *a, **kw):
        return self.module.load_state_dict(sd, *a, **kw)
