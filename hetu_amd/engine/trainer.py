"""Trainer: owns the train graph, run context, and the hipGraph-captured
steady-state step.

Reference parity: python/hetu/engine/trainer.py:66-828 (build/feed-dict/train
loop). MI355X-native twist: instead of the reference's C++ run loop, the hot
step is captured once into a hipGraph (torch.cuda.graphs) after warmup and
replayed — removing per-op Python + launch overhead entirely; Adam bias
corrections stay correct via pinned-buffer re-reads (AdamStepOp).
"""
from __future__ import annotations

import os
from typing import Dict, Optional

import torch

from ..graph.executor import ExecContext
from ..graph.ops.optim import AdamStepOp
from ..parallel.comm import comm_backend
from .runner import prepare_run_context


class Trainer:
    def __init__(self, graph, handles: Dict, device: torch.device,
                 capture: Optional[bool] = None,
                 lr_schedule=None):
        """lr_schedule: callable step -> multiplier on the optimizer's
        base lr (engine.lr_schedule); works under capture — the
        multiplier rides the pinned bias-correction buffer."""
        self.graph = graph
        self.h = handles
        self.device = device
        self.lr_schedule = lr_schedule
        self.ctx = prepare_run_context(graph, device)
        env = os.environ.get("HETU_AMD_CAPTURE", "auto")
        if capture is None:
            capture = (env != "0") and device.type == "cuda"
        self.want_capture = capture
        self._cuda_graph = None
        self._static_feeds: Dict = {}
        self._loss_out = None
        # resume-aware: checkpoint load restores each Adam op's step
        # counter BEFORE the Trainer is built; starting _step at 0 would
        # make the first captured replay's set_replay_step clobber them
        # (and restart any LR schedule)
        self._step = max((op.interface.state.get("step", 0)
                          for op in graph.ops
                          if op.type in ("AdamStep", "ZeroAdamStep")
                          and op.interface.state), default=0)
        # host-side tracing (reference engine/trainer.py:22 wires
        # torch.profiler): HETU_AMD_TORCH_PROFILE=<dir> records steps
        # 2-4 (skip 0, warm 1) and writes a chrome trace there
        self._prof = None
        prof_dir = os.environ.get("HETU_AMD_TORCH_PROFILE", "")
        if prof_dir:
            acts = [torch.profiler.ProfilerActivity.CPU]
            if device.type == "cuda":
                acts.append(torch.profiler.ProfilerActivity.CUDA)
            self._prof = torch.profiler.profile(
                activities=acts,
                schedule=torch.profiler.schedule(wait=1, warmup=1,
                                                 active=3, repeat=1),
                on_trace_ready=torch.profiler.tensorboard_trace_handler(
                    prof_dir))
            self._prof.start()

    # ---- plain step ------------------------------------------------------
    def run_step(self, feed: Dict):
        if self.lr_schedule is not None:
            AdamStepOp.set_lr_scale(self.lr_schedule(self._step))
        loss, _ = self.graph.run([self.h["loss"], self.h["train_op"]],
                                 feed, ctx=self.ctx)
        self._step += 1
        return loss

    # ---- captured step ---------------------------------------------------
    def capture(self, feed: Dict):
        """Capture one full train step into a hipGraph. `feed` values become
        static device buffers (copy new data into `self.static(name)` before
        each replay)."""
        assert self.device.type == "cuda"
        for t, v in feed.items():
            self._static_feeds[t] = v.to(self.device).clone()
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        static_feed = dict(self._static_feeds)
        with torch.cuda.graph(g):
            loss, _ = self.graph.run(
                [self.h["loss"], self.h["train_op"]], static_feed,
                ctx=self.ctx)
            self._loss_out = loss
        self._cuda_graph = g
        # NOTE: stream capture only RECORDS the step — nothing executes —
        # so capture does not count as a training step; the caller replays
        # immediately after

    def replay(self):
        scale = self.lr_schedule(self._step) \
            if self.lr_schedule is not None else None
        self._step += 1
        AdamStepOp.set_replay_step(self._step, lr_scale=scale)
        self._cuda_graph.replay()
        return self._loss_out

    def _prof_tick(self):
        if self._prof is not None:
            self._prof.step()
            if self._step >= 5:            # schedule exhausted
                self._prof.stop()
                self._prof = None

    def step(self, feed: Dict):
        """Run one training step, transparently using capture when armed."""
        try:
            return self._step_impl(feed)
        finally:
            self._prof_tick()

    def _step_impl(self, feed: Dict):
        if self._cuda_graph is None:
            if self.want_capture and self._step >= 1:
                try:
                    self.capture(feed)
                    # run the recorded step for real (capture executed
                    # nothing; without this the capture call would
                    # silently skip one update and return stale loss)
                    return self.replay()
                except Exception as e:  # noqa: BLE001
                    from ..utils.logging import get_logger
                    get_logger().warning(
                        "hipGraph capture failed, running eager: %s", e)
                    self.want_capture = False
            for t, v in feed.items():
                feed[t] = v.to(self.device)
            return self.run_step(feed)
        for t, v in feed.items():
            self._static_feeds[t].copy_(v, non_blocking=True)
        return self.replay()
