"""DistributedStates algebra tests (reference distributed_states.h
semantics: states/order device mapping, conversion predicates)."""
import pytest

from hetu_amd.parallel.dstates import (DistributedStates, ds_dup,
                                       ds_from_index_table, ds_partial,
                                       ds_split)


class TestMapping:
    def test_dp_tp_order(self):
        # 8 devices: dp=2 (slow), tp=4 (fast); weight split dim0 over tp
        ds = DistributedStates(8, {-1: 2, 0: 4}, order=[-1, 0])
        assert ds.map_device_to_state_index(0) == {-1: 0, 0: 0}
        assert ds.map_device_to_state_index(3) == {-1: 0, 0: 3}
        assert ds.map_device_to_state_index(5) == {-1: 1, 0: 1}

    def test_groups_along(self):
        ds = DistributedStates(8, {-1: 2, 0: 4}, order=[-1, 0])
        groups = ds.group_devices_along(0)   # tp groups
        assert sorted(map(sorted, groups)) == [[0, 1, 2, 3], [4, 5, 6, 7]]
        groups = ds.group_devices_along(-1)  # dp (replica) groups
        assert sorted(map(sorted, groups)) == [[0, 4], [1, 5], [2, 6], [3, 7]]

    def test_local_slice(self):
        ds = DistributedStates(4, {0: 2, 1: 2}, order=[0, 1])
        sl = ds.local_slice((8, 6), 3)
        assert sl[0] == slice(4, 8) and sl[1] == slice(3, 6)

    def test_local_global_shape(self):
        ds = DistributedStates(4, {0: 4})
        assert ds.local_shape((8, 3)) == (2, 3)
        assert ds.global_shape((2, 3)) == (8, 3)


class TestPredicates:
    def test_allreduce(self):
        src = ds_partial(4)
        dst = ds_dup(4)
        assert src.check_allreduce(dst)
        assert not dst.check_allreduce(src)

    def test_allgather(self):
        src = ds_split(4, 0)
        dst = ds_dup(4)
        assert src.check_allgather(dst, 0)
        assert not src.check_allgather(dst, 1)

    def test_reducescatter(self):
        src = ds_partial(4)
        dst = ds_split(4, 0)
        assert src.check_reducescatter(dst, 0)

    def test_mixed_dims(self):
        # dp2 x tp4: partial over tp -> dup over tp, dp split kept
        src = DistributedStates(8, {0: 2, -2: 4}, order=[0, -2])
        dst = DistributedStates(8, {0: 2, -1: 4}, order=[0, -1])
        assert src.check_allreduce(dst)

    def test_split_pred(self):
        src = ds_dup(4)
        dst = ds_split(4, 1)
        assert src.check_split(dst)


class TestIndexTable:
    def test_roundtrip(self):
        ds = DistributedStates(8, {-1: 2, 0: 2, 1: 2}, order=[0, -1, 1])
        table = [ds.map_device_to_state_index(i) for i in range(8)]
        counts = dict(ds.states)
        ds2 = ds_from_index_table(8, table, counts)
        for i in range(8):
            assert ds2.map_device_to_state_index(i) == table[i]

    def test_matmul_colparallel(self):
        """Column-parallel linear: x {0:dp,-1:tp} @ w^T {0(N):tp,-1:dp}
        -> y {0:dp, last:tp}."""
        import torch
        import hetu_amd as ht
        dp, tp = 2, 2
        n = dp * tp
        ds_x = DistributedStates(n, {0: dp, -1: tp}, order=[0, -1])
        ds_w = DistributedStates(n, {0: tp, -1: dp}, order=[-1, 0])
        with ht.graph("define_and_run"):
            x = ht.placeholder([8, 16], ds=ds_x, device_group=list(range(n)))
            w = ht.variable(torch.randn(8, 16), ds=ds_w,
                            device_group=list(range(n)))
            y = ht.linear(x, w)
        assert y.ds.get_dim(0) == dp
        assert y.ds.get_dim(1) == tp
        assert y.ds.partial == 1

    def test_matmul_rowparallel(self):
        """Row-parallel: x split on K over tp -> y partial over tp."""
        import torch
        import hetu_amd as ht
        dp, tp = 2, 2
        n = dp * tp
        ds_x = DistributedStates(n, {0: dp, 1: tp}, order=[0, 1])
        ds_w = DistributedStates(n, {1: tp, -1: dp}, order=[-1, 1])
        with ht.graph("define_and_run"):
            x = ht.placeholder([8, 16], ds=ds_x, device_group=list(range(n)))
            w = ht.variable(torch.randn(32, 16), ds=ds_w,
                            device_group=list(range(n)))
            y = ht.linear(x, w)
        assert y.ds.get_dim(0) == dp
        assert y.ds.partial == tp
