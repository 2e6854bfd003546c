"""Python wrapper over the native C++ TokenBinLoader
(hetu_amd/ops/hip/dataloader.cpp — reference graph/data/dataloader.h)."""
from __future__ import annotations

from typing import Iterator, Tuple

import numpy as np
import torch


class NativeTokenDataset:
    """Iterates (input_ids [B,S], labels [B*S]) batches from a flat token
    .bin file via the C++ prefetching loader."""

    def __init__(self, path: str, batch: int, seq_len: int,
                 dtype_bytes: int = 2, prefetch: int = 4, seed: int = 0,
                 pin: bool = True, drop_last: bool = True):
        from ..ops.functional import ext
        self._ldr = ext().TokenBinLoader(path, batch, seq_len, dtype_bytes,
                                         prefetch, seed, pin, drop_last)
        self.batch, self.seq_len = batch, seq_len
        self._epoch = 0

    def __len__(self) -> int:
        return self._ldr.num_batches()

    def __iter__(self) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
        while True:
            t = self._ldr.next()
            if t.numel() == 0:
                self._epoch += 1
                self._ldr.start_epoch(self._epoch)
                return
            yield t[:, :-1].contiguous(), t[:, 1:].reshape(-1).contiguous()


def write_token_bin(path: str, tokens, dtype_bytes: int = 2) -> None:
    """Helper: dump a token id sequence to the flat .bin format."""
    arr = np.asarray(tokens, dtype=np.uint16 if dtype_bytes == 2
                     else np.int32)
    arr.tofile(path)
