"""DistributedStates — the SPMD tensor-layout algebra.

Semantics follow the reference's DistributedStates
(/root/reference/hetu/graph/distributed_states.h:13-200): a tensor placed on a
device group of `device_num` devices is described by

  * ``states``: {dim: split_count} where dim >= 0 splits that tensor dim,
    dim == -1 is the replication (duplicate) count, dim == -2 is the
    partial-reduction count (the tensor is a partial sum over that many
    devices).
  * ``order``: the sequence of dims (from slowest- to fastest-varying) that
    maps a device's index within the group to its (split-index per dim) tuple.

The product of all state counts must equal ``device_num``.

Conversion predicates (check_allreduce / check_allgather / check_reducescatter
/ check_split / check_scatter) mirror distributed_states.h:110-116 and drive
comm-op type deduction (see hetu_amd/graph/ops/comm.py).

This is a from-scratch implementation for the MI355X runtime: the execution
side maps each (device group, participating dims) to a torch.distributed
process group over RCCL.
"""
from __future__ import annotations

from typing import Dict, List, Optional


class DistributedStates:
    __slots__ = ("device_num", "states", "order", "zero")

    def __init__(self, device_num: int, states: Dict[int, int],
                 order: Optional[List[int]] = None, zero: bool = False):
        states = {int(k): int(v) for k, v in states.items() if int(v) > 1}
        prod = 1
        for v in states.values():
            prod *= v
        if prod != device_num:
            raise ValueError(
                f"states {states} product {prod} != device_num {device_num}")
        if order is None:
            # default order: partial, dup, then splits ascending
            order = sorted(states.keys())
        order = [d for d in order if states.get(d, 1) > 1]
        missing = [d for d in sorted(states.keys()) if d not in order]
        order = missing + order if missing else order
        self.device_num = device_num
        self.states = states
        self.order = list(order)
        self.zero = zero

    # ---- accessors -------------------------------------------------------
    def get_dim(self, dim: int) -> int:
        return self.states.get(dim, 1)

    @property
    def partial(self) -> int:
        return self.get_dim(-2)

    @property
    def dup(self) -> int:
        return self.get_dim(-1)

    def split_dims(self) -> List[int]:
        return sorted(d for d in self.states if d >= 0)

    def is_pure_dup(self) -> bool:
        return all(d == -1 for d in self.states)

    # ---- device index <-> state index ------------------------------------
    def map_device_to_state_index(self, device_index: int) -> Dict[int, int]:
        """Index of this device along each state dim (all dims, default 0)."""
        if not 0 <= device_index < self.device_num:
            raise ValueError(f"device_index {device_index} out of range")
        idx: Dict[int, int] = {}
        rem = device_index
        for dim in reversed(self.order):
            n = self.states[dim]
            idx[dim] = rem % n
            rem //= n
        return idx

    def get_dup_group_index(self, device_index: int) -> int:
        """Linear index over the non-(-1) dims — identifies which unique data
        shard this device holds (devices sharing it are replicas)."""
        st = self.map_device_to_state_index(device_index)
        idx = 0
        for dim in self.order:
            if dim == -1:
                continue
            idx = idx * self.states[dim] + st.get(dim, 0)
        return idx

    def group_devices_along(self, dim: int) -> List[List[int]]:
        """Partition the device group's local indices into groups that vary
        only along `dim` (used to build RCCL subgroups for a collective
        over that state dim)."""
        n = self.get_dim(dim)
        if n == 1:
            return [[i] for i in range(self.device_num)]
        buckets: Dict[tuple, List[int]] = {}
        for i in range(self.device_num):
            st = self.map_device_to_state_index(i)
            key = tuple((d, st.get(d, 0)) for d in sorted(self.states)
                        if d != dim)
            buckets.setdefault(key, []).append(i)
        return list(buckets.values())

    # ---- conversion predicates (distributed_states.h:110-116) ------------
    def _same_but(self, other: "DistributedStates", src_dim: int,
                  dst_dim: int) -> bool:
        """True if `other` equals self with the count moved src_dim->dst_dim."""
        if self.device_num != other.device_num:
            return False
        a = dict(self.states)
        n = a.pop(src_dim, 1)
        if n == 1:
            return False
        b = dict(other.states)
        m = b.pop(dst_dim, 1)
        if m % n != 0 and n % m != 0:
            return False
        # fold moved count into dst side and compare
        a[dst_dim] = self.get_dim(dst_dim) * n
        aa = {k: v for k, v in a.items() if v > 1}
        return aa == other.states

    def check_equal(self, other: "DistributedStates") -> bool:
        return (self.device_num == other.device_num
                and self.states == other.states)

    def check_allreduce(self, dst: "DistributedStates") -> bool:
        """partial k -> dup k (everything else unchanged)."""
        return self.partial > 1 and self._same_but(dst, -2, -1)

    def check_allgather(self, dst: "DistributedStates", gather_dim: int = 0
                        ) -> bool:
        """split(gather_dim) k -> dup k."""
        return self.get_dim(gather_dim) > 1 and self._same_but(dst, gather_dim, -1)

    def check_reducescatter(self, dst: "DistributedStates",
                            scatter_dim: int = 0) -> bool:
        """partial k -> split(scatter_dim) k."""
        return self.partial > 1 and self._same_but(dst, -2, scatter_dim)

    def check_scatter(self, dst: "DistributedStates", dim: int = 0) -> bool:
        """dup k -> split(dim) k (no communication if data already present:
        each replica keeps its slice)."""
        return self.dup > 1 and self._same_but(dst, -1, dim)

    def check_split(self, dst: "DistributedStates") -> bool:
        """src dup covers some new split in dst: local slicing only."""
        if self.device_num != dst.device_num:
            return False
        if self.partial != dst.partial:
            return False
        for d in dst.split_dims():
            if dst.get_dim(d) % self.get_dim(d) != 0:
                return False
        # every extra split in dst must come out of src's dup
        extra = 1
        for d in dst.split_dims():
            extra *= dst.get_dim(d) // self.get_dim(d)
        for d in self.split_dims():
            if self.get_dim(d) > dst.get_dim(d):
                return False
        return extra > 1 and self.dup == dst.dup * extra

    # ---- helpers ---------------------------------------------------------
    def local_shape(self, global_shape) -> tuple:
        out = list(global_shape)
        for d, n in self.states.items():
            if d >= 0:
                if out[d] % n != 0:
                    raise ValueError(
                        f"dim {d} size {out[d]} not divisible by split {n}")
                out[d] //= n
        return tuple(out)

    def global_shape(self, local_shape) -> tuple:
        out = list(local_shape)
        for d, n in self.states.items():
            if d >= 0:
                out[d] *= n
        return tuple(out)

    def local_slice(self, global_shape, device_index: int):
        """List of python slices selecting this device's shard."""
        st = self.map_device_to_state_index(device_index)
        slices = []
        for d, size in enumerate(global_shape):
            n = self.get_dim(d)
            if n > 1:
                blk = size // n
                i = st.get(d, 0)
                slices.append(slice(i * blk, (i + 1) * blk))
            else:
                slices.append(slice(None))
        return tuple(slices)

    def __eq__(self, other):
        return (isinstance(other, DistributedStates)
                and self.check_equal(other) and self.order == other.order)

    def __hash__(self):
        return hash((self.device_num, tuple(sorted(self.states.items())),
                     tuple(self.order)))

    def __repr__(self):
        return (f"DS(n={self.device_num}, states={self.states}, "
                f"order={self.order})")


NULL_HETERO_DIM = -3


class DistributedStatesUnion:
    """Per-pipeline heterogeneous layouts (reference
    /root/reference/hetu/graph/distributed_states.h:158-233).

    A union holds one DistributedStates per heterogeneous pipeline plus a
    ``hetero_dim``: the dim along which the pipelines partition the global
    tensor (>=0 a tensor dim, -1 duplicate, -2 partial; NULL_HETERO_DIM=-3
    means homogeneous, union size 1).  Each entry is a FULL-device-count ds
    that includes the union factor on the hetero dim; ``get_local(i)``
    strips it, yielding pipeline i's own layout over its own device count.

    This is the algebra behind Malleus hetero parallel: e.g. a weight
    trained by two pipelines with tp2 and tp1 has a union of two entries
    whose local layouts differ; grads reduce across pipelines with
    split-allreduce groups (parallel/hetero.py)."""

    __slots__ = ("union", "hetero_dim")

    def __init__(self, ds_list: List[DistributedStates],
                 hetero_dim: int = NULL_HETERO_DIM):
        if len(ds_list) > 1 and hetero_dim == NULL_HETERO_DIM:
            raise ValueError("hetero_dim required for union size > 1")
        if len(ds_list) <= 1 and hetero_dim != NULL_HETERO_DIM:
            raise ValueError("hetero_dim must be NULL for union size <= 1")
        self.union = list(ds_list)
        self.hetero_dim = hetero_dim

    def is_hetero(self) -> bool:
        return self.hetero_dim != NULL_HETERO_DIM

    def size(self) -> int:
        return len(self.union)

    def get(self, i: int) -> DistributedStates:
        return self.union[i]

    def get_default_ds(self) -> DistributedStates:
        return self.union[0]

    def get_local(self, i: int) -> DistributedStates:
        """Pipeline i's layout with the union factor stripped off the
        hetero dim (reference get_local)."""
        if not self.is_hetero():
            return self.union[0]
        ds = self.union[i]
        n = len(self.union)
        if ds.get_dim(self.hetero_dim) % n != 0:
            raise ValueError(
                f"hetero dim {self.hetero_dim} count "
                f"{ds.get_dim(self.hetero_dim)} not divisible by union {n}")
        states = dict(ds.states)
        states[self.hetero_dim] = ds.get_dim(self.hetero_dim) // n
        order = list(ds.order)
        if states[self.hetero_dim] == 1:
            states.pop(self.hetero_dim)
            order = [d for d in order if d != self.hetero_dim]
        return DistributedStates(ds.device_num // n, states, order,
                                 zero=ds.zero)

    @classmethod
    def to_hetero(cls, ds: DistributedStates, dim: int, num: int
                  ) -> "DistributedStatesUnion":
        """Lift a homogeneous ds into a hetero union of `num` identical
        entries split along `dim` (reference to_hetero)."""
        if ds.get_dim(dim) % num != 0:
            raise ValueError(f"dim {dim} count {ds.get_dim(dim)} not "
                             f"divisible by {num}")
        return cls([ds] * num, hetero_dim=dim)

    def check_equal(self, other: "DistributedStatesUnion") -> bool:
        if (self.is_hetero() != other.is_hetero()
                or self.hetero_dim != other.hetero_dim
                or len(self.union) != len(other.union)):
            return False
        return all(a.check_equal(b)
                   for a, b in zip(self.union, other.union))

    def __repr__(self):
        return (f"DSUnion(hetero_dim={self.hetero_dim}, "
                f"union={self.union})")


def ds_from_index_table(device_num: int, table: List[Dict[int, int]],
                        counts: Dict[int, int]) -> DistributedStates:
    """Infer (states, order) from a per-device {dim: state_index} table.

    Used by op-level DoDeduceStates implementations (e.g. matmul): the
    producer computes, for every device in the group, which shard indices its
    local output holds, and this reconstructs the DistributedStates that
    describes that layout. Raises if the table is not a regular grid (then
    the layout is not expressible as a DistributedStates).
    """
    counts = {d: n for d, n in counts.items() if n > 1}
    prod = 1
    for n in counts.values():
        prod *= n
    if prod != device_num:
        # remaining degrees of freedom are replicas
        if device_num % prod != 0:
            raise ValueError(f"counts {counts} do not divide {device_num}")
        counts[-1] = counts.get(-1, 1) * (device_num // prod)
    # find each dim's period: smallest stride at which its index changes
    periods: Dict[int, int] = {}
    for d in counts:
        if d == -1 and -1 not in (table[0] if table else {}):
            continue
        p = device_num
        for i in range(1, device_num):
            if table[i].get(d, 0) != table[0].get(d, 0):
                p = i
                break
        periods[d] = p
    # dup dims absent from the table get the leftover periods; fall back to
    # ordering by period descending (slowest-varying first)
    dims = sorted(counts.keys(), key=lambda d: -periods.get(d, 1))
    ds = DistributedStates(device_num, counts, order=dims)
    # validate round-trip for the dims present in the table
    for i in range(device_num):
        st = ds.map_device_to_state_index(i)
        for d in table[i]:
            if d in counts and st.get(d, 0) != table[i][d]:
                raise ValueError(
                    f"layout not expressible as DistributedStates: device {i}"
                    f" dim {d}: table={table[i][d]} ds={st.get(d, 0)}")
    return ds


def ds_dup(device_num: int) -> DistributedStates:
    return DistributedStates(device_num, {-1: device_num} if device_num > 1 else {})


def ds_split(device_num: int, dim: int = 0) -> DistributedStates:
    return DistributedStates(device_num, {dim: device_num} if device_num > 1 else {})


def ds_partial(device_num: int) -> DistributedStates:
    return DistributedStates(device_num, {-2: device_num} if device_num > 1 else {})
