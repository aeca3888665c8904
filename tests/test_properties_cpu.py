"""Property-based tests (hypothesis) for the pure partitioning/packing
logic: sampler coverage, label shifting, TP shard/reconstruct inverses."""
import torch
from hypothesis import given, settings
from hypothesis import strategies as st


@settings(max_examples=30, deadline=None)
@given(n=st.integers(4, 200), world=st.integers(1, 5),
       seed=st.integers(0, 3), epoch=st.integers(0, 3))
def test_sampler_partition_properties(n, world, seed, epoch):
    """Ranks' index sets are disjoint, equal-sized (drop_last), within
    range, and epoch-stable across re-iteration."""
    from distributed_training_guide_amd.data.sampler import \
        DistributedSampler

    ds = list(range(n))
    all_idx = []
    per_rank = None
    for r in range(world):
        s = DistributedSampler(ds, num_replicas=world, rank=r, seed=seed)
        s.set_epoch(epoch)
        idx = list(iter(s))
        assert len(idx) == len(s) == n // world
        assert idx == list(iter(s))  # deterministic per epoch
        if per_rank is None:
            per_rank = len(idx)
        all_idx += idx
    assert len(set(all_idx)) == len(all_idx)  # disjoint
    assert all(0 <= i < n for i in all_idx)


@settings(max_examples=30, deadline=None)
@given(B=st.integers(1, 5), S=st.integers(2, 33))
def test_shifted_labels_property(B, S):
    from distributed_training_guide_amd.ops.cross_entropy import IGNORE_INDEX
    from distributed_training_guide_amd.ops.fused_linear_ce import \
        _shifted_flat_labels

    labels = torch.randint(0, 1000, (B, S))
    out = _shifted_flat_labels(labels).view(B, S)
    assert torch.equal(out[:, : S - 1], labels[:, 1:])
    assert (out[:, S - 1] == IGNORE_INDEX).all()


@settings(max_examples=20, deadline=None)
@given(tp_old=st.sampled_from([1, 2, 4]), tp_new=st.sampled_from([1, 2, 4]),
       segs=st.sampled_from([(16,), (16, 8, 8), (32, 32)]),
       in_f=st.sampled_from([8, 24]))
def test_colwise_shard_reconstruct_roundtrip(tp_old, tp_new, segs, in_f):
    """_shard_rows at tp_old for every rank -> reconstruct -> original;
    then re-sharding at tp_new is a partition of the original rows."""
    from distributed_training_guide_amd.parallel.tp import _shard_rows

    out_f = sum(segs)
    full = torch.arange(out_f * in_f, dtype=torch.float32).view(out_f, in_f)
    parts = [_shard_rows(full, list(segs), tp_old, r) for r in range(tp_old)]
    # reconstruct (mirror of TPLlamaForCausalLM.reconstruct_full_tensor)
    full_segs = []
    off = 0
    for seg in segs:
        loc = seg // tp_old
        full_segs.append(torch.cat([p[off: off + loc] for p in parts]))
        off += loc
    rebuilt = torch.cat(full_segs)
    assert torch.equal(rebuilt, full)
    new_parts = [_shard_rows(rebuilt, list(segs), tp_new, r)
                 for r in range(tp_new)]
    assert sum(p.shape[0] for p in new_parts) == out_f
