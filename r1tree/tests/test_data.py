"""Data pipeline: packing invariants, CP-symmetric chunking, samplers,
prefetch loader (reference data/bucket.py + dataloader.py parity)."""
import torch

from hetu_amd.data.bucket import Bucket, bucketize
from hetu_amd.data.dataloader import (PrefetchLoader, SampleBatchSampler,
                                      SyntheticLMDataset, TokenBatchSampler,
                                      lm_collate)


def test_pack_data_invariants():
    torch.manual_seed(0)
    b = Bucket(max_seqlen=128, pad_token=0, alignment=16)
    lens = [5, 17, 33, 64, 100, 128, 16, 48, 80, 31]
    for n in lens:
        b.add(torch.randint(1, 100, (n,)))
    tokens, cus = b.pack_data()
    assert tokens.shape[1] == 128
    total = 0
    for bin_i, cu in enumerate(cus):
        assert cu[-1] <= 128
        assert (cu[1:] > cu[:-1]).all()
        total += len(cu) - 1
    assert total == len(lens)           # every sequence placed exactly once
    # aligned starts
    for cu in cus:
        assert all(int(c) % 16 == 0 for c in cu)


def test_cp_pack_symmetric_roundtrip():
    b = Bucket(max_seqlen=64, pad_token=0, alignment=16)
    for n in (64, 32, 16):
        b.add(torch.arange(1, n + 1))
    cp = 2
    out, rank_cus = b.generate_cp_pack_data(cp)
    assert out.shape[0] == cp and out.shape[2] == 32
    # every rank got an equal share of every sequence
    tokens, cus = b.pack_data()
    for bin_i in range(tokens.shape[0]):
        for si in range(len(cus[bin_i]) - 1):
            s0, s1 = int(cus[bin_i][si]), int(cus[bin_i][si + 1])
            seg = tokens[bin_i, s0:s1]
            c = len(seg) // (2 * cp)
            got = torch.cat([out[r, bin_i] for r in range(cp)])
            for r in range(cp):
                head = seg[r * c:(r + 1) * c]
                # head chunk must appear in rank r's row
                row = out[r, bin_i]
                found = any(torch.equal(row[i:i + c], head)
                            for i in range(0, row.shape[0] - c + 1))
                assert found


def test_sample_sampler_shards():
    s0 = list(SampleBatchSampler(32, 8, dp=2, dp_rank=0, shuffle=False))
    s1 = list(SampleBatchSampler(32, 8, dp=2, dp_rank=1, shuffle=False))
    assert len(s0) == len(s1) == 4
    for a, b in zip(s0, s1):
        assert len(a) == len(b) == 4
        assert not set(a) & set(b)


def test_token_sampler_budget():
    lens = [10, 20, 30, 40, 50, 60]
    batches = list(TokenBatchSampler(lens, max_tokens=64, sort=True))
    for b in batches:
        assert sum(lens[i] for i in b) <= 64


def test_prefetch_loader():
    ds = SyntheticLMDataset(vocab=100, seq_len=16, n=32)
    sampler = SampleBatchSampler(32, 8, dp=1, dp_rank=0, shuffle=False)
    out = list(PrefetchLoader(ds, sampler, lm_collate))
    assert len(out) == 4
    x, y = out[0]
    assert x.shape == (8, 16) and y.shape == (8 * 16,)


def test_bucketize():
    seqs = [torch.zeros(n, dtype=torch.int64) for n in (10, 100, 300, 600)]
    bk = bucketize(seqs, [128, 512, 1024])
    assert len(bk[128]) == 2 and len(bk[512]) == 1 and len(bk[1024]) == 1
