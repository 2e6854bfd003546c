"""Activation CPU offload for pipeline stages.

Reference parity: graph/offload/activation_cpu_offload.cc — per-block
offload flags insert D2H copies after forward and H2D prefetches before
backward on a dedicated offload stream.  MI355X-native shape: the unit of
offload is a pipeline micro-batch's cached activation set (the
`keep_values` dict between the stage's fwd and bwd); copies ride a side
HIP stream into a reusable pinned-buffer pool and overlap with compute,
paying one PCIe round trip to free HBM while 1F1B holds up to `pp`
micro-batches in flight.
"""
from __future__ import annotations

from typing import Dict, List, Optional

import torch


class _Handle:
    __slots__ = ("cpu", "meta", "ev_off", "ev_pre", "gpu", "pool_slot")

    def __init__(self):
        self.cpu: Dict[int, torch.Tensor] = {}
        self.meta: Dict[int, tuple] = {}
        self.ev_off: Optional[torch.cuda.Event] = None
        self.ev_pre: Optional[torch.cuda.Event] = None
        self.gpu: Optional[Dict[int, torch.Tensor]] = None
        self.pool_slot: Optional[int] = None


class ActOffloader:
    """Offload/prefetch the activation cache of a micro-batch."""

    def __init__(self, device: torch.device, slots: int = 4):
        self.device = device
        self.use_cuda = device.type == "cuda" and torch.cuda.is_available()
        self.stream = torch.cuda.Stream(device) if self.use_cuda else None
        # pinned buffer pool: slot -> {key: pinned tensor}; shapes repeat
        # every micro-batch so buffers are allocated once
        self._pool: List[Dict[int, torch.Tensor]] = [
            {} for _ in range(slots)]
        self._free: List[int] = list(range(slots))

    def _pinned(self, slot: int, key: int, t: torch.Tensor) -> torch.Tensor:
        buf = self._pool[slot].get(key)
        if buf is None or buf.shape != t.shape or buf.dtype != t.dtype:
            buf = torch.empty_like(t, device="cpu",
                                   pin_memory=self.use_cuda)
            self._pool[slot][key] = buf
        return buf

    def offload(self, cache: Dict[int, torch.Tensor]) -> _Handle:
        """Copy every cached activation to pinned CPU; the caller drops the
        GPU refs afterwards (freeing HBM once the copies land)."""
        h = _Handle()
        assert self._free, "offload slots exhausted (raise slots=)"
        h.pool_slot = self._free.pop()
        if self.use_cuda:
            self.stream.wait_stream(torch.cuda.current_stream(self.device))
            with torch.cuda.stream(self.stream):
                for k, t in cache.items():
                    if not isinstance(t, torch.Tensor):
                        continue
                    buf = self._pinned(h.pool_slot, k, t)
                    buf.copy_(t, non_blocking=True)
                    h.cpu[k] = buf
                    h.meta[k] = (t.shape, t.dtype)
            h.ev_off = torch.cuda.Event()
            h.ev_off.record(self.stream)
        else:
            for k, t in cache.items():
                if isinstance(t, torch.Tensor):
                    h.cpu[k] = t.detach().cpu().clone()
                    h.meta[k] = (t.shape, t.dtype)
        return h

    def prefetch(self, h: _Handle) -> None:
        """Start async H2D for a handle (idempotent)."""
        if h.gpu is not None:
            return
        h.gpu = {}
        if self.use_cuda:
            # the offload copies must have landed before reuse
            if h.ev_off is not None:
                self.stream.wait_event(h.ev_off)
            with torch.cuda.stream(self.stream):
                for k, buf in h.cpu.items():
                    h.gpu[k] = buf.to(self.device, non_blocking=True)
            h.ev_pre = torch.cuda.Event()
            h.ev_pre.record(self.stream)
        else:
            for k, buf in h.cpu.items():
                h.gpu[k] = buf.clone()

    def fetch(self, h: _Handle) -> Dict[int, torch.Tensor]:
        """Block until the handle's activations are on-device; frees the
        pinned slot for reuse."""
        self.prefetch(h)
        if self.use_cuda and h.ev_pre is not None:
            torch.cuda.current_stream(self.device).wait_event(h.ev_pre)
        if h.pool_slot is not None:
            self._free.append(h.pool_slot)
            h.pool_slot = None
        return h.gpu
