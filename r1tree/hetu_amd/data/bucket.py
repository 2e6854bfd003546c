"""Sequence bucketing / packing for variable-length training.

Reference parity: python/hetu/data/bucket.py — `pack_data:86` (greedy
first-fit packing of length-sorted sequences into max_seqlen bins with
alignment padding) and `generate_cp_pack_data:193` (symmetric context-
parallel chunking: every cp rank gets a (head, tail) slice of each packed
sequence so causal-attention work balances around the ring; emits per-rank
cu_seqlens).
"""
from __future__ import annotations

from typing import Dict, List, Sequence, Tuple

import torch


class Bucket:
    """Holds sequences (1-D LongTensors) padded/packed to max_seqlen."""

    def __init__(self, max_seqlen: int, pad_token: int = 0,
                 alignment: int = 16):
        self.max_seqlen = max_seqlen
        self.pad_token = pad_token
        self.alignment = alignment
        self.seqs: List[torch.Tensor] = []

    def add(self, seq: torch.Tensor):
        self.seqs.append(seq[:self.max_seqlen])

    def __len__(self):
        return len(self.seqs)

    def _aligned(self, n: int) -> int:
        a = self.alignment
        return (n + a - 1) // a * a

    def pad_data(self) -> torch.Tensor:
        """[num_seqs, max_seqlen] padded batch."""
        out = torch.full((len(self.seqs), self.max_seqlen), self.pad_token,
                         dtype=torch.int64)
        for i, s in enumerate(self.seqs):
            out[i, :len(s)] = s
        return out

    def pack_data(self) -> Tuple[torch.Tensor, List[torch.Tensor]]:
        """Greedy first-fit-decreasing packing (reference pack_data:86):
        sequences sorted by length descending drop into the first bin with
        room (aligned lengths); returns ([num_bins, max_seqlen] tokens,
        per-bin cu_seqlens)."""
        order = sorted(range(len(self.seqs)),
                       key=lambda i: -len(self.seqs[i]))
        bins: List[List[int]] = []
        used: List[int] = []
        for i in order:
            n = self._aligned(len(self.seqs[i]))
            placed = False
            for b, u in enumerate(used):
                if u + n <= self.max_seqlen:
                    bins[b].append(i)
                    used[b] += n
                    placed = True
                    break
            if not placed:
                bins.append([i])
                used.append(n)
        tokens = torch.full((len(bins), self.max_seqlen), self.pad_token,
                            dtype=torch.int64)
        cu_seqlens: List[torch.Tensor] = []
        for b, idxs in enumerate(bins):
            cu = [0]
            off = 0
            for i in idxs:
                s = self.seqs[i]
                tokens[b, off:off + len(s)] = s
                off += self._aligned(len(s))
                cu.append(off)
            cu_seqlens.append(torch.tensor(cu, dtype=torch.int32))
        return tokens, cu_seqlens

    def generate_cp_pack_data(self, cp: int
                              ) -> Tuple[torch.Tensor, List[List[torch.Tensor]]]:
        """Symmetric CP split (reference generate_cp_pack_data:193,
        HETU_PARALLEL_ATTN_SPLIT_PATTERN=SYM): each packed sequence of
        aligned length L splits into 2*cp equal chunks; cp rank r takes
        chunk r (head) and chunk 2*cp-1-r (tail), so causal work is equal
        around the ring.  Returns ([cp, num_bins, max_seqlen/cp] tokens,
        per-rank per-bin cu_seqlens)."""
        tokens, cus = self.pack_data()
        nb, L = tokens.shape
        assert L % (2 * cp) == 0, "max_seqlen must divide 2*cp"
        per = L // cp
        out = torch.full((cp, nb, per), self.pad_token, dtype=torch.int64)
        rank_cus: List[List[torch.Tensor]] = [[] for _ in range(cp)]
        for b in range(nb):
            cu = cus[b]
            for r in range(cp):
                parts = []
                rcu = [0]
                off = 0
                for si in range(len(cu) - 1):
                    s0, s1 = int(cu[si]), int(cu[si + 1])
                    seg = tokens[b, s0:s1]
                    c = len(seg) // (2 * cp)
                    head = seg[r * c:(r + 1) * c]
                    tail = seg[(2 * cp - 1 - r) * c:(2 * cp - r) * c]
                    parts.append(head)
                    parts.append(tail)
                    off += 2 * c
                    rcu.append(off)
                row = torch.cat(parts)
                out[r, b, :len(row)] = row
                rank_cus[r].append(torch.tensor(rcu, dtype=torch.int32))
        return out, rank_cus


def bucketize(seqs: Sequence[torch.Tensor], boundaries: Sequence[int],
              pad_token: int = 0, alignment: int = 16
              ) -> Dict[int, Bucket]:
    """Distribute sequences into per-boundary buckets (hotspa seq-len
    bucket dispatch feeds each bucket to its own strategy)."""
    out: Dict[int, Bucket] = {b: Bucket(b, pad_token, alignment)
                              for b in sorted(boundaries)}
    bs = sorted(boundaries)
    for s in seqs:
        for b in bs:
            if len(s) <= b:
                out[b].add(s)
                break
        else:
            out[bs[-1]].add(s[:bs[-1]])
    return out
