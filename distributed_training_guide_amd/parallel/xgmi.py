"""Direct (fully-connected) collectives for the MI355X xGMI mesh.

Intra-node topology is 7 point-to-point xGMI links per GPU at ~153 GB/s
(SURVEY.md §2b): RCCL's default ring pushes the whole message over ONE
link per step, so a ring all-gather of N bytes takes ~(w-1)/w * N / link.
On a fully-connected mesh every rank can instead exchange its shard with
all w-1 peers SIMULTANEOUSLY — one batched isend/irecv group (RCCL lowers
it to concurrent p2p over the per-peer links), moving (w-1)/w * N total
but spread across 7 links: up to ~7x the per-link-bound ring for large
messages.  Reduce-scatter is the mirror (scatter chunks, reduce locally);
all-reduce composes the two (RS + AG), the same decomposition RCCL's ring
uses but with both phases direct.

Selection: DTGA_XGMI_ALGO=direct switches the DDP bucket all-reduce and
the FSDP unshard/reduce-scatter to these; default stays RCCL's built-ins
(measured choice — tools/bench_collectives.py --algo compares both at
N=2..8).  All functions are synchronous within the call (the engines'
async overlap wraps them at a higher level via worker streams when
needed) and fall back to the stock collective at world==1.
"""
import os

import torch
import torch.distributed as dist


def algo() -> str:
    return os.environ.get("DTGA_XGMI_ALGO", "rccl")


class _DirectWork:
    """Work-like handle over a batched p2p group: wait() drains the
    requests (stream-ordered for NCCL/RCCL) and runs the finish stage
    (local copies/reductions, possibly a follow-on phase)."""

    __slots__ = ("reqs", "_finish", "_done")

    def __init__(self, reqs, finish=None):
        self.reqs = reqs
        self._finish = finish
        self._done = False

    def wait(self):
        if self._done:
            return True
        for r in self.reqs:
            r.wait()
        if self._finish is not None:
            self._finish()
        self._done = True
        return True


def direct_all_gather_into(out: torch.Tensor, shard: torch.Tensor,
                           group=None, async_op: bool = False):
    """out [w*n] <- gather of per-rank shard [n]; every peer pair
    exchanges simultaneously (one batched p2p group)."""
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    assert out.numel() == shard.numel() * world, \
        "direct_all_gather_into: out must be world x shard"
    chunks = list(out.chunk(world))
    if world == 1:
        chunks[0].copy_(shard)
        return _DirectWork([]) if async_op else None
    ops = []
    for off in range(1, world):
        dst = (rank + off) % world
        src = (rank - off) % world
        ops.append(dist.P2POp(dist.isend, shard, group_peer=dst,
                              group=group))
        ops.append(dist.P2POp(dist.irecv, chunks[src], group_peer=src,
                              group=group))
    reqs = dist.batch_isend_irecv(ops)
    chunks[rank].copy_(shard)
    w = _DirectWork(reqs)
    if async_op:
        return w
    w.wait()


def direct_reduce_scatter(out: torch.Tensor, inp: torch.Tensor, group=None,
                          async_op: bool = False):
    """out [n] <- sum over ranks of inp chunk `rank`; chunks scatter
    directly to their owners, reduction is local adds after the
    exchange."""
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    assert inp.numel() == out.numel() * world, \
        "direct_reduce_scatter: inp must be world x out"
    chunks = list(inp.chunk(world))
    if world == 1:
        out.copy_(chunks[0])
        return _DirectWork([]) if async_op else None
    recv = [torch.empty_like(out) for _ in range(world - 1)]
    ops = []
    for i, off in enumerate(range(1, world)):
        dst = (rank + off) % world
        src = (rank - off) % world
        ops.append(dist.P2POp(dist.isend, chunks[dst].contiguous(),
                              group_peer=dst, group=group))
        ops.append(dist.P2POp(dist.irecv, recv[i], group_peer=src,
                              group=group))
    reqs = dist.batch_isend_irecv(ops)
    out.copy_(chunks[rank])

    def finish():
        for r in recv:
            out.add_(r)

    w = _DirectWork(reqs, finish)
    if async_op:
        return w
    w.wait()


def direct_all_reduce(t: torch.Tensor, group=None, async_op: bool = False):
    """In-place sum all-reduce as direct RS + direct AG (pads to a
    world-divisible chunking internally).  On CUDA, Work.wait() is only a
    stream dependency, so the WHOLE RS+reduce+AG pipeline is enqueued
    eagerly here and overlaps whatever the caller does next (the DDP
    engine's remaining backward); on gloo wait() host-blocks, so the AG
    phase is deferred into the returned handle's wait()."""
    world = dist.get_world_size(group)
    if world == 1:
        return _DirectWork([]) if async_op else None
    n = t.numel()
    flat = t.reshape(-1)
    pad = (n + world - 1) // world * world - n
    if pad:
        buf = torch.empty(n + pad, dtype=t.dtype, device=t.device)
        buf[:n].copy_(flat)
        buf[n:].zero_()
    else:
        buf = flat
    shard = torch.empty(buf.numel() // world, dtype=t.dtype,
                        device=t.device)
    rs = direct_reduce_scatter(shard, buf, group, async_op=True)

    if t.is_cuda:
        rs.wait()  # stream-ordered only: local adds enqueue now
        ag = direct_all_gather_into(buf, shard, group, async_op=True)

        def finish():
            ag.wait()
            if pad:
                flat.copy_(buf[:n])
    else:
        def finish():
            rs.wait()
            direct_all_gather_into(buf, shard, group)
            if pad:
                flat.copy_(buf[:n])

    w = _DirectWork([], finish)
    if async_op:
        return w
    w.wait()
