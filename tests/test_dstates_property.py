"""Property-based tests of the DistributedStates algebra (hypothesis):
partition completeness, index-map consistency, shape roundtrips —
invariants the reference's distributed_states.cc relies on everywhere."""
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from hetu_amd.parallel.dstates import DistributedStates, ds_from_index_table


def _ds_strategy():
    """Random valid ds: device_num = product of split factors x dup."""

    @st.composite
    def build(draw):
        factors = draw(st.lists(st.sampled_from([2, 2, 3, 4]),
                                min_size=0, max_size=3))
        dims = draw(st.permutations([0, 1, 2]))
        states = {}
        order = []
        for f, d in zip(factors, dims):
            states[d] = f
            order.append(d)
        dup = draw(st.sampled_from([1, 2, 3]))
        if dup > 1 or not states:
            states[-1] = max(dup, 1) if (dup > 1 or not states) else 1
            if -1 in states and states[-1] > 0:
                order.append(-1)
        n = 1
        for v in states.values():
            n *= v
        return DistributedStates(n, states, order)
    return build()


@settings(max_examples=200, deadline=None)
@given(_ds_strategy())
def test_partition_covers_global_exactly(ds):
    """Union of every device's local_slice = the global tensor; slices of
    devices in the same dup group coincide, others tile disjointly."""
    shape = (12, 12, 12)
    full = torch.arange(12 ** 3).reshape(shape)
    counts = torch.zeros(shape, dtype=torch.int64)
    for dev in range(ds.device_num):
        sl = ds.local_slice(shape, dev)
        counts[sl] += 1
    # every element covered exactly dup times
    assert (counts == ds.dup).all(), (ds, counts.unique())


@settings(max_examples=200, deadline=None)
@given(_ds_strategy())
def test_local_global_shape_roundtrip(ds):
    shape = (12, 12, 12)
    loc = ds.local_shape(shape)
    assert tuple(ds.global_shape(loc)) == shape


@settings(max_examples=200, deadline=None)
@given(_ds_strategy())
def test_index_table_roundtrip(ds):
    """ds -> per-device state index table -> ds reconstructs the same
    partition (same local slices for every device)."""
    table = [ds.map_device_to_state_index(i) for i in range(ds.device_num)]
    counts = {d: ds.get_dim(d) for d in ds.split_dims()}
    ds2 = ds_from_index_table(ds.device_num, [
        {d: t.get(d, 0) for d in counts} for t in table], counts)
    shape = (12, 12, 12)
    for dev in range(ds.device_num):
        assert ds.local_slice(shape, dev) == ds2.local_slice(shape, dev), \
            (ds, ds2, dev)


@settings(max_examples=100, deadline=None)
@given(_ds_strategy())
def test_group_devices_along_partitions_devices(ds):
    for d in ds.split_dims():
        groups = ds.group_devices_along(d)
        flat = sorted(x for g in groups for x in g)
        assert flat == list(range(ds.device_num))
        assert all(len(g) == ds.get_dim(d) for g in groups)


@settings(max_examples=150, deadline=None)
@given(_ds_strategy(), st.integers(0, 2), st.sampled_from(
    ["allgather", "reducescatter", "slice_dup", "allreduce"]))
def test_comm_kind_deduction_consistency(ds, dim, move):
    """deduce_comm_kind must recognize the canonical transitions built
    from any source layout (reference Communication.h DoDeduceStates)."""
    from hetu_amd.graph.ops.comm import (_rs_positions_ok,
                                         _slice_contained,
                                         _stable_dims_match,
                                         deduce_comm_kind)
    n = ds.device_num
    states = dict(ds.states)
    order = list(ds.order)

    def mk(st_, od_):
        return DistributedStates(n, st_, od_)

    if move == "allreduce":
        if -2 in states or ds.dup == n or n == 1:
            return
        # partial over everything -> pure dup
        src = mk({-2: n}, [-2])
        dst = mk({-1: n}, [-1])
        assert deduce_comm_kind(src, dst)[0] == "allreduce"
    elif move == "allgather":
        if ds.get_dim(dim) <= 1:
            return
        # split(dim) folds into dup
        k = states.pop(dim)
        states[-1] = states.get(-1, 1) * k
        od = [-1 if d == dim else d for d in order]
        dedup = []
        for d in od:
            if d not in dedup:
                dedup.append(d)
        dst = mk(states, dedup)
        kind, d = deduce_comm_kind(ds, dst)
        if _stable_dims_match(ds, dst):
            assert kind == "allgather" and d == dim, (ds, dst, kind)
        else:
            # order change moved another dim's device mapping: the fast
            # kind would place shards wrongly — generic is required
            assert kind == "generic", (ds, dst, kind)
    elif move == "reducescatter":
        if -2 in states or ds.dup <= 1 or dim in states:
            return
        # build a partial source, dst moves partial into split(dim)
        k = states.pop(-1)
        src_states = dict(states)
        src_states[-2] = k
        src = mk(src_states, [d for d in order if d != -1] + [-2])
        dst_states = dict(states)
        dst_states[dim] = k
        dst = mk(dst_states, [d for d in order if d != -1] + [dim])
        kind, d = deduce_comm_kind(src, dst)
        if _stable_dims_match(src, dst) and _rs_positions_ok(src, dst, dim):
            assert kind == "reducescatter" and d == dim, (src, dst, kind)
        else:
            assert kind == "generic", (src, dst, kind)
    else:  # slice_dup: dup splits into a new dim
        if ds.dup <= 1 or dim in states:
            return
        k = states.pop(-1)
        dst_states = dict(states)
        dst_states[dim] = k
        dst = mk(dst_states, [d for d in order if d != -1] + [dim])
        kind, _ = deduce_comm_kind(ds, dst)
        if _stable_dims_match(ds, dst) and _slice_contained(ds, dst):
            assert kind == "slice", (ds, dst, kind)
        else:
            assert kind == "generic", (ds, dst, kind)


@settings(max_examples=100, deadline=None)
@given(st.lists(st.integers(1, 120), min_size=1, max_size=40),
       st.integers(0, 10_000))
def test_pack_data_preserves_tokens(lengths, seed):
    """Packing invariants (reference Bucket.pack_data): every token lands
    in exactly one bin at the cu_seqlens offsets, no segment crosses
    max_seqlen, alignment respected."""
    from hetu_amd.data.bucket import Bucket
    torch.manual_seed(seed)
    b = Bucket(max_seqlen=128, pad_token=-1, alignment=16)
    seqs = [torch.randint(0, 1000, (n,)) for n in lengths]
    for s in seqs:
        b.add(s)
    tokens, cus = b.pack_data()
    assert tokens.shape[1] == 128
    # collect back every packed segment
    seen = []
    for bin_i, cu in enumerate(cus):
        cu = cu.tolist()
        assert cu[-1] <= 128
        for s0, s1 in zip(cu[:-1], cu[1:]):
            assert s0 % 16 == 0
            seg = tokens[bin_i, s0:s1]
            real = seg[seg != -1]
            seen.append(real)
    # multiset of sequences must match (order-independent)
    got = sorted([tuple(t.tolist()) for t in seen])
    want = sorted([tuple(s.tolist()) for s in seqs])
    assert got == want
