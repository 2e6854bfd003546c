"""Datasets + samplers + prefetching loader.

Reference parity: python/hetu/data — JsonDataset, sample-/token-level batch
samplers (dataloader.py:162,244), and the C++ prefetching dataloader
(hetu/graph/data/dataloader.h:18) realized here as a double-buffered
background-thread loader feeding pinned host tensors.
"""
from __future__ import annotations

import json
import queue
import threading
from typing import Callable, Iterator, List, Optional, Sequence

import torch


class JsonDataset:
    """jsonl corpus; each line {"text": ...}; tokenizer: str -> List[int]
    (any callable — HF `tokenizers`, tiktoken, sentencepiece wrappers)."""

    def __init__(self, path: str, tokenizer: Callable[[str], List[int]],
                 key: str = "text", max_seq_len: Optional[int] = None):
        self.samples: List[torch.Tensor] = []
        with open(path) as f:
            for line in f:
                line = line.strip()
                if not line:
                    continue
                text = json.loads(line)[key]
                ids = tokenizer(text)
                if max_seq_len:
                    ids = ids[:max_seq_len]
                self.samples.append(torch.tensor(ids, dtype=torch.int64))

    def __len__(self):
        return len(self.samples)

    def __getitem__(self, i):
        return self.samples[i]


class SyntheticLMDataset:
    """Random-token dataset for benchmarks (no network for corpora)."""

    def __init__(self, vocab: int, seq_len: int, n: int = 1024,
                 seed: int = 0):
        g = torch.Generator().manual_seed(seed)
        self.data = torch.randint(0, vocab, (n, seq_len + 1), generator=g)

    def __len__(self):
        return self.data.shape[0]

    def __getitem__(self, i):
        row = self.data[i]
        return row[:-1], row[1:]


class SampleBatchSampler:
    """Fixed number of SAMPLES per global batch, sharded over dp ranks
    (reference dataloader.py:162)."""

    def __init__(self, n: int, global_batch: int, dp: int, dp_rank: int,
                 shuffle: bool = True, seed: int = 0, drop_last: bool = True):
        self.n, self.gb = n, global_batch
        self.dp, self.dp_rank = dp, dp_rank
        self.shuffle, self.seed, self.drop_last = shuffle, seed, drop_last
        self.epoch = 0

    def set_epoch(self, e: int):
        self.epoch = e

    def __iter__(self) -> Iterator[List[int]]:
        idx = list(range(self.n))
        if self.shuffle:
            g = torch.Generator().manual_seed(self.seed + self.epoch)
            idx = torch.randperm(self.n, generator=g).tolist()
        per = self.gb // self.dp
        for i in range(0, self.n - (self.gb if self.drop_last else 0) + 1,
                       self.gb):
            chunk = idx[i:i + self.gb]
            if len(chunk) < self.gb and self.drop_last:
                break
            yield chunk[self.dp_rank * per:(self.dp_rank + 1) * per]


class TokenBatchSampler:
    """Greedy batches capped by a TOKEN budget (reference :244): sequences
    accumulate until adding one would exceed max_tokens."""

    def __init__(self, lengths: Sequence[int], max_tokens: int,
                 dp: int = 1, dp_rank: int = 0, sort: bool = True):
        self.lengths = list(lengths)
        self.max_tokens = max_tokens
        self.dp, self.dp_rank = dp, dp_rank
        self.sort = sort

    def __iter__(self) -> Iterator[List[int]]:
        order = sorted(range(len(self.lengths)),
                       key=lambda i: self.lengths[i]) if self.sort \
            else list(range(len(self.lengths)))
        batches: List[List[int]] = []
        cur: List[int] = []
        tok = 0
        for i in order:
            li = self.lengths[i]
            if cur and tok + li > self.max_tokens:
                batches.append(cur)
                cur, tok = [], 0
            cur.append(i)
            tok += li
        if cur:
            batches.append(cur)
        for j, b in enumerate(batches):
            if j % self.dp == self.dp_rank:
                yield b


class PrefetchLoader:
    """Background-thread prefetcher (C++ dataloader parity): collate on a
    worker thread into (optionally pinned) tensors, depth-2 queue."""

    def __init__(self, dataset, sampler, collate: Callable,
                 pin_memory: bool = False, depth: int = 2):
        self.dataset = dataset
        self.sampler = sampler
        self.collate = collate
        self.pin = pin_memory
        self.depth = depth

    def __iter__(self):
        q: "queue.Queue" = queue.Queue(maxsize=self.depth)
        stop = object()

        def worker():
            for idxs in self.sampler:
                items = [self.dataset[i] for i in idxs]
                batch = self.collate(items)
                if self.pin:
                    batch = tuple(t.pin_memory() if torch.is_tensor(t) else t
                                  for t in batch)
                q.put(batch)
            q.put(stop)

        t = threading.Thread(target=worker, daemon=True)
        t.start()
        while True:
            item = q.get()
            if item is stop:
                break
            yield item


def lm_collate(items):
    xs = torch.stack([x for x, _ in items])
    ys = torch.stack([y for _, y in items])
    return xs, ys.reshape(-1)
