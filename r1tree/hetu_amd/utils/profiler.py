"""Op profiler + memory snapshots.

Reference parity: hetu/impl/profiler/profiler.h:25 (per-op type/name/time
with context stack, exposed as the hetu.profiler context) and
graph/profiler.h:41 CUDAProfiler (GetCurrMemoryInfo, per-micro-batch
begin/end memory snapshots, executable_graph.cc:1524-1548).

MI355X-native: GPU timing via hipEvents (torch.cuda.Event), memory via the
torch-ROCm caching-allocator counters; deeper counters come from rocprofv3
(scripts/kbench.py, profiles/).
"""
from __future__ import annotations

import time
from collections import defaultdict
from typing import Dict, List, Optional

import torch


class OpProfiler:
    """Attach via ExecContext.profiler; the executor calls begin/end around
    every op compute."""

    def __init__(self, use_events: bool = True):
        self.records: List = []          # (op_type, op_name, seconds)
        self.use_events = use_events
        self._pending: List = []

    def begin(self, op):
        if self.use_events and torch.cuda.is_available():
            e0 = torch.cuda.Event(enable_timing=True)
            e0.record()
            return (op, e0, True)
        return (op, time.perf_counter(), False)

    def end(self, token):
        op, t0, is_event = token
        if is_event:
            e1 = torch.cuda.Event(enable_timing=True)
            e1.record()
            self._pending.append((op, t0, e1))
        else:
            self.records.append((op.type, op.name,
                                 time.perf_counter() - t0))

    def flush(self):
        if self._pending and torch.cuda.is_available():
            torch.cuda.synchronize()
            for op, e0, e1 in self._pending:
                self.records.append((op.type, op.name,
                                     e0.elapsed_time(e1) / 1e3))
            self._pending.clear()

    def summary(self, top: int = 20) -> str:
        self.flush()
        agg: Dict[str, List[float]] = defaultdict(list)
        for typ, _, sec in self.records:
            agg[typ].append(sec)
        rows = sorted(((sum(v), len(v), t) for t, v in agg.items()),
                      reverse=True)
        total = sum(r[0] for r in rows) or 1e-12
        out = [f"{'op type':24s} {'calls':>7s} {'total ms':>10s} "
               f"{'avg us':>9s} {'%':>6s}"]
        for tot, n, typ in rows[:top]:
            out.append(f"{typ:24s} {n:7d} {tot * 1e3:10.2f} "
                       f"{tot / n * 1e6:9.1f} {tot / total * 100:6.2f}")
        return "\n".join(out)

    def reset(self):
        self.records.clear()
        self._pending.clear()


def memory_info(device: Optional[torch.device] = None) -> Dict[str, float]:
    """Allocator snapshot in GiB (CUDAProfiler::GetCurrMemoryInfo)."""
    if not torch.cuda.is_available():
        return {"allocated": 0.0, "reserved": 0.0, "max_allocated": 0.0}
    g = 1 << 30
    return {
        "allocated": torch.cuda.memory_allocated(device) / g,
        "reserved": torch.cuda.memory_reserved(device) / g,
        "max_allocated": torch.cuda.max_memory_allocated(device) / g,
    }


class MemorySnapshots:
    """Per-micro-batch begin/end snapshots (MEMORY_PROFILE_LEVEL
    MICRO_BATCH parity)."""

    def __init__(self):
        self.snaps: List = []

    def mark(self, tag: str, device=None):
        self.snaps.append((tag, memory_info(device)))

    def report(self) -> str:
        return "\n".join(
            f"{tag:24s} alloc={m['allocated']:.2f}GiB "
            f"resv={m['reserved']:.2f}GiB peak={m['max_allocated']:.2f}GiB"
            for tag, m in self.snaps)
