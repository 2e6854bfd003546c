"""SYM (zigzag) ring-attention split: causal load balance + 2-rank exact
parity vs full-sequence attention (reference STRIPE/SYM split patterns,
ParallelAttention.cc:196-204)."""
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_sym_subblock_load_balance():
    """Every ring rank computes exactly 2n+1 chunk-pairs per full ring
    pass (NORMAL gives rank p only p+1 of n blocks)."""
    from hetu_amd.parallel.ring_attention import _sym_subblocks
    for n in (2, 4, 8):
        for my in range(n):
            total = 0
            for src in range(n):
                total += len(list(_sym_subblocks(my, src, n)))
            assert total == 2 * n + 1, (n, my, total)


SYM_WORKER = r"""
import os, sys, torch
sys.path.insert(0, os.environ["HETU_REPO"])
from hetu_amd.parallel.comm import comm_backend
from hetu_amd.parallel.ring_attention import (ring_attn_fwd_sym,
                                              ring_attn_bwd_sym)
import hetu_amd.ops.functional as F
comm = comm_backend()
rank, n = comm.rank, comm.world_size
torch.manual_seed(0)
B, H, S, D = 2, 2, 32, 16
q = torch.randn(B, H, S, D)
k = torch.randn(B, H, S, D)
v = torch.randn(B, H, S, D)
dout = torch.randn(B, H, S, D)
half = S // (2 * n)


def sym_slice(t):
    head = t[:, :, rank * half:(rank + 1) * half]
    tail = t[:, :, (2 * n - 1 - rank) * half:(2 * n - rank) * half]
    return torch.cat([head, tail], dim=2).contiguous()


o_ref, lse_ref = F.flash_attn_fwd(q, k, v, True, None)
dq_ref, dk_ref, dv_ref = F.flash_attn_bwd(dout, q, k, v, o_ref, lse_ref,
                                          True, None)
ql, kl, vl = sym_slice(q), sym_slice(k), sym_slice(v)
ranks = list(range(n))
o, lse = ring_attn_fwd_sym(ql, kl, vl, comm, ranks)
assert torch.allclose(o, sym_slice(o_ref), atol=1e-4), \
    (o - sym_slice(o_ref)).abs().max()
assert torch.allclose(lse, sym_slice(lse_ref.unsqueeze(-1)).squeeze(-1),
                      atol=1e-4)
dq, dk, dv = ring_attn_bwd_sym(sym_slice(dout), ql, kl, vl, o, lse,
                               comm, ranks)
assert torch.allclose(dq, sym_slice(dq_ref), atol=1e-4), \
    (dq - sym_slice(dq_ref)).abs().max()
assert torch.allclose(dk, sym_slice(dk_ref), atol=1e-4)
assert torch.allclose(dv, sym_slice(dv_ref), atol=1e-4)
print("SYMOK")
"""


def _run_sym(ws, port):
    env0 = {**os.environ, "HETU_REPO": REPO, "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port), "GLOO_SOCKET_IFNAME": "lo"}
    procs = []
    for r in range(ws):
        env = dict(env0, RANK=str(r), WORLD_SIZE=str(ws),
                   LOCAL_RANK=str(r))
        procs.append(subprocess.Popen([sys.executable, "-c", SYM_WORKER],
                                      env=env, stdout=subprocess.PIPE,
                                      stderr=subprocess.PIPE, text=True))
    for r, p in enumerate(procs):
        out, err = p.communicate(timeout=300)
        ok = (p.returncode in (0, -6)) and "SYMOK" in out
        assert ok, f"rank {r}: rc={p.returncode}\n{out}\n{err}"


def test_sym_ring_two_ranks_parity():
    _run_sym(2, 29707)


def test_sym_ring_four_ranks_parity():
    _run_sym(4, 29727)


HET_WORKER = r"""
import os, sys, torch
sys.path.insert(0, os.environ["HETU_REPO"])
from hetu_amd.parallel.comm import comm_backend
from hetu_amd.parallel.ring_attention import (ring_attn_fwd_hetero,
                                              ring_attn_bwd_hetero)
import hetu_amd.ops.functional as F
comm = comm_backend()
rank, n = comm.rank, comm.world_size
torch.manual_seed(0)
B, H, D = 2, 2, 16
lens = [8, 24]                      # UNEQUAL shards (hetero CP)
S = sum(lens)
q = torch.randn(B, H, S, D)
k = torch.randn(B, H, S, D)
v = torch.randn(B, H, S, D)
dout = torch.randn(B, H, S, D)
offs = [0, lens[0], S]
sl = slice(offs[rank], offs[rank + 1])
o_ref, lse_ref = F.flash_attn_fwd(q, k, v, True, None)
dq_ref, dk_ref, dv_ref = F.flash_attn_bwd(dout, q, k, v, o_ref, lse_ref,
                                          True, None)
ranks = list(range(n))
ql, kl, vl = (t[:, :, sl].contiguous() for t in (q, k, v))
o, lse = ring_attn_fwd_hetero(ql, kl, vl, comm, ranks, lens)
assert torch.allclose(o, o_ref[:, :, sl], atol=1e-4), \
    (o - o_ref[:, :, sl]).abs().max()
dq, dk, dv = ring_attn_bwd_hetero(dout[:, :, sl].contiguous(), ql, kl, vl,
                                  o, lse, comm, ranks, lens)
assert torch.allclose(dq, dq_ref[:, :, sl], atol=1e-4)
assert torch.allclose(dk, dk_ref[:, :, sl], atol=1e-4)
assert torch.allclose(dv, dv_ref[:, :, sl], atol=1e-4)
print("HETOK")
"""


def test_hetero_cp_unequal_shards():
    """Hetero CP: NORMAL ring with unequal per-rank seq shards
    (reference trainer.py:255-259) matches full-sequence attention."""
    env0 = {**os.environ, "HETU_REPO": REPO, "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": "29717", "GLOO_SOCKET_IFNAME": "lo"}
    procs = []
    for r in range(2):
        env = dict(env0, RANK=str(r), WORLD_SIZE="2", LOCAL_RANK=str(r))
        procs.append(subprocess.Popen([sys.executable, "-c", HET_WORKER],
                                      env=env, stdout=subprocess.PIPE,
                                      stderr=subprocess.PIPE, text=True))
    for r, p in enumerate(procs):
        out, err = p.communicate(timeout=300)
        ok = (p.returncode in (0, -6)) and "HETOK" in out
        assert ok, f"rank {r}: rc={p.returncode}\n{out}\n{err}"
