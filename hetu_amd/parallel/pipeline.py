"""Pipeline parallelism: stage graphs + PipeDream-flush (1F1B) schedule.

Reference parity: hetu/graph/executable_graph.cc:803 (GPipe), :836-882
(GeneratePipedreamFlushSchedule), pipeline P2P batching (:996-1008), grad
accumulation buffers, and DeducePipeline (define_and_run_graph.cc:638).

MI355X-native design: instead of the reference's global graph with per-layer
device-group annotations compiled into one ExecutableGraph, each rank builds
its own stage subgraph (fwd / bwd / update fetch sets on one
DefineAndRunGraph) and a PipelineRunner drives the 1F1B schedule with
batched RCCL p2p (deadlock-free paired send/recv) over xGMI.  The forward
pass caches its activations (executor seed/keep values), the delayed
backward replays from the cache, gradients accumulate across micro-batches
into persistent fp32 buffers, and the data-parallel allreduce happens ONCE
per step inside the update graph — not per micro-batch.
"""
from __future__ import annotations

import dataclasses
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from .comm import CommBackend, comm_backend


@dataclasses.dataclass
class PipelineSpec:
    """pp x dp x tp grid; ranks laid out [pp][dp][tp] (tp fastest)."""
    pp: int = 1
    dp: int = 1
    tp: int = 1
    sequence_parallel: bool = False

    @property
    def world(self) -> int:
        return self.pp * self.dp * self.tp

    def stage_ranks(self, stage: int) -> List[int]:
        base = stage * self.dp * self.tp
        return list(range(base, base + self.dp * self.tp))

    def my_stage(self, rank: Optional[int] = None) -> int:
        rank = comm_backend().rank if rank is None else rank
        return rank // (self.dp * self.tp)

    def stage_spec(self, stage: int):
        """ParallelSpec (dp x tp) on this stage's rank group."""
        from ..nn.parallel import ParallelSpec
        return ParallelSpec(dp=self.dp, tp=self.tp,
                            device_group=self.stage_ranks(stage),
                            sequence_parallel=self.sequence_parallel)

    def peer(self, rank: int, to_stage: int) -> int:
        """Rank in `to_stage` with the same (dp, tp) coordinates."""
        return to_stage * self.dp * self.tp + rank % (self.dp * self.tp)

    def partition_layers(self, n_layer: int) -> List[List[int]]:
        """Contiguous layer split (reference DeducePipeline semantics;
        non-uniform splits may be passed to the model builders directly)."""
        per = n_layer // self.pp
        rem = n_layer % self.pp
        out, i = [], 0
        for s in range(self.pp):
            n = per + (1 if s < rem else 0)
            out.append(list(range(i, i + n)))
            i += n
        return out


class StageModule:
    """A stage subgraph and its handles:
    act_in (placeholder, non-first), act_out (non-last), loss (last),
    grad_in (placeholder, non-last), dx (grad of act_in, non-first),
    params, param_grads, grad_phs, train_op, act_shape, act_dtype."""

    def __init__(self, graph, handles: Dict):
        self.graph = graph
        self.h = handles


def _p2p(comm: CommBackend, sends, recvs):
    """One batched p2p group call (reference BatchedISendIRecv)."""
    ops = []
    for t, dst in sends:
        ops.append(dist.P2POp(dist.isend, t.contiguous(), dst))
    for t, src in recvs:
        ops.append(dist.P2POp(dist.irecv, t, src))
    if ops:
        for r in dist.batch_isend_irecv(ops):
            r.wait()


class PipelineRunner:
    """Drives one optimizer step = M micro-batches through 1F1B."""

    def __init__(self, spec: PipelineSpec, stage: StageModule,
                 device: torch.device, ctx=None, scaler=None,
                 recompute: bool = False, offload: bool = False,
                 schedule: str = "1f1b"):
        self.spec = spec
        self.stage = stage
        self.device = device
        self.schedule = schedule      # "1f1b" (PipeDream-flush) | "gpipe"
        # hipGraph capture of the per-micro-batch fwd/bwd bodies (rotating
        # pp+1 buffer slots); opt-in: HETU_AMD_PP_CAPTURE=1, GPU only
        import os as _os2
        # 1F1B only: GPipe keeps all M micro-batches in flight, which
        # would reuse a slot before its backward replays
        self._capture = (device.type == "cuda" and not recompute
                         and not offload and schedule == "1f1b"
                         and _os2.environ.get("HETU_AMD_PP_CAPTURE",
                                              "0") == "1")
        self._nslots = spec.pp + 1
        self._slots = [None] * self._nslots
        self._step_no = 0
        self.scaler = scaler          # engine.amp.GradScaler or None
        self.recompute = recompute    # rerun fwd in bwd instead of caching
        self.offloader = None
        if offload and not recompute:
            from ..engine.offload import ActOffloader
            # 1F1B holds up to pp in-flight micro-batches on stage 0
            self.offloader = ActOffloader(device, slots=spec.pp + 1)
        self.comm = comm_backend(device)
        self.rank = self.comm.rank
        self.sid = spec.my_stage(self.rank)
        self.is_first = self.sid == 0
        self.is_last = self.sid == spec.pp - 1
        self.prev = None if self.is_first else spec.peer(self.rank,
                                                         self.sid - 1)
        self.next = None if self.is_last else spec.peer(self.rank,
                                                        self.sid + 1)
        from ..engine.runner import prepare_run_context
        self.ctx = ctx or prepare_run_context(stage.graph, device)
        h = stage.h
        self.params: List = h["params"]
        self.grad_bufs = [
            torch.zeros(tuple(p.shape), dtype=torch.float32, device=device)
            for p in self.params]
        self._act_shape = tuple(h["act_shape"])
        self._act_dtype = h.get("act_dtype", torch.bfloat16)
        # per-micro-batch memory snapshots (reference CUDAProfiler
        # GetCurrMemoryInfo at MEMORY_PROFILE_LEVEL=MICRO_BATCH)
        self.mem_snapshots = None
        import os as _os
        if _os.environ.get("HETU_AMD_MEM_PROFILE", "0") == "1":
            from ..utils.profiler import MemorySnapshots
            self.mem_snapshots = MemorySnapshots()

    # ---- hipGraph-captured per-slot stage bodies -------------------------
    # (reference/VERDICT: each micro-batch re-ran the Python executor per
    # stage — ~30 ms host overhead per micro-batch; capturing the fwd and
    # bwd bodies into per-slot hipGraphs cuts that to a graph launch.  1F1B
    # keeps up to pp in-flight micro-batches, so pp+1 rotating buffer sets.)
    def _slot_capture(self, si, micro_batches, i):
        h = self.stage.h
        out_t = h["loss"] if self.is_last else h["act_out"]
        slot = {"feeds": {}, "kept": {}}
        for t, v in micro_batches[i].items():
            slot["feeds"][t] = v.to(self.device).clone()
        if not self.is_first:
            slot["feeds"][h["act_in"]] = torch.empty(
                self._act_shape, dtype=self._act_dtype, device=self.device)
        if not self.is_last:
            slot["gin"] = torch.empty(self._act_shape,
                                      dtype=self._act_dtype,
                                      device=self.device)
        if "loss_seed" in h and self.is_last:
            slot["seed"] = torch.ones((), dtype=torch.float32,
                                      device=self.device)
        torch.cuda.synchronize()
        gf = torch.cuda.CUDAGraph()
        with torch.cuda.graph(gf):
            slot["out"] = self.stage.graph.run(
                [out_t], dict(slot["feeds"]), ctx=self.ctx,
                keep_values=slot["kept"])[0]
        bwd_feed = {}
        if not self.is_last:
            bwd_feed[h["grad_in"]] = slot["gin"]
        elif "seed" in slot:
            bwd_feed[h["loss_seed"]] = slot["seed"]
        for t, v in slot["feeds"].items():
            bwd_feed.setdefault(t, v)
        fetches = ([] if self.is_first else [h["dx"]]) + h["param_grads"]
        gb = torch.cuda.CUDAGraph()
        with torch.cuda.graph(gb):
            slot["bwd_outs"] = self.stage.graph.run(
                fetches, bwd_feed, ctx=self.ctx,
                seed_values=dict(slot["kept"]))
        slot["fwd_g"], slot["bwd_g"] = gf, gb
        return slot

    def _fwd_captured(self, i, micro_batches, act):
        si = i % self._nslots
        if self._slots[si] is None:
            self._slots[si] = self._slot_capture(si, micro_batches, i)
        slot = self._slots[si]
        for t, v in micro_batches[i].items():
            slot["feeds"][t].copy_(v, non_blocking=True)
        if act is not None:
            slot["feeds"][self.stage.h["act_in"]].copy_(act)
        slot["fwd_g"].replay()
        return slot["out"], ("slot", si)

    def _bwd_captured(self, saved, gin):
        _, si = saved
        slot = self._slots[si]
        if gin is not None and "gin" in slot:
            slot["gin"].copy_(gin)
        if "seed" in slot:
            scale = self.scaler.scale if self.scaler is not None else 1.0
            slot["seed"].fill_(float(scale))
        slot["bwd_g"].replay()
        res = slot["bwd_outs"]
        dx = None
        if not self.is_first:
            dx, res = res[0], res[1:]
        for buf, g in zip(self.grad_bufs, res):
            if g is not None:
                buf += g.float()
        return dx

    # ---- fwd / bwd over the stage graph ---------------------------------
    def _fwd(self, i, micro_batches, act):
        if self.mem_snapshots is not None:
            self.mem_snapshots.mark(f"fwd_mb{i}", self.device)
        if self._capture and self._step_no >= 1:
            return self._fwd_captured(i, micro_batches, act)
        h = self.stage.h
        feed = dict(micro_batches[i])
        if not self.is_first:
            feed[h["act_in"]] = act
        out_t = h["loss"] if self.is_last else h["act_out"]
        if self.recompute:
            # activation recompute (reference recompute.cc semantics at
            # stage granularity): store only the feed; bwd reruns fwd
            out = self.stage.graph.run([out_t], feed, ctx=self.ctx)
            return out[0], (feed, None)
        cache: Dict[int, torch.Tensor] = {}
        out = self.stage.graph.run([out_t], feed, ctx=self.ctx,
                                   keep_values=cache)
        if self.offloader is not None:
            handle = self.offloader.offload(cache)
            cache.clear()                 # drop HBM refs
            return out[0], (feed, handle)
        return out[0], (feed, cache)

    def _bwd(self, saved, gin):
        if self.mem_snapshots is not None:
            self.mem_snapshots.mark("bwd_mb", self.device)
        if isinstance(saved, tuple) and len(saved) == 2 \
                and saved[0] == "slot":
            return self._bwd_captured(saved, gin)
        h = self.stage.h
        feed, cache = saved
        if self.offloader is not None and cache is not None \
                and not isinstance(cache, dict):
            cache = self.offloader.fetch(cache)
        if not self.is_last:
            feed = dict(feed)
            feed[h["grad_in"]] = gin
        elif "loss_seed" in h:
            # scale the backward seed (reference gradscaler.cc semantics):
            # fp16 grads flow through the stage graph multiplied by
            # scaler.scale so they stay above the fp16 underflow floor;
            # unscale_ divides the accumulated fp32 buffers back down.
            feed = dict(feed)
            scale = self.scaler.scale if self.scaler is not None else 1.0
            feed[h["loss_seed"]] = torch.tensor(
                float(scale), dtype=torch.float32, device=self.device)
        fetches = ([] if self.is_first else [h["dx"]]) + h["param_grads"]
        res = self.stage.graph.run(fetches, feed, ctx=self.ctx,
                                   seed_values=cache)
        dx = None
        if not self.is_first:
            dx, res = res[0], res[1:]
        for buf, g in zip(self.grad_bufs, res):
            if g is not None:
                buf += g.float()
        return dx

    def _recv_act(self):
        t = torch.empty(self._act_shape, dtype=self._act_dtype,
                        device=self.device)
        _p2p(self.comm, [], [(t, self.prev)])
        return t

    # ---- one training step (1F1B | GPipe) --------------------------------
    def step(self, micro_batches: List[Dict]):
        """Returns the mean micro-batch loss on the LAST stage, else None."""
        self._step_no += 1
        if self.schedule == "gpipe":
            return self._step_gpipe(micro_batches)
        M = len(micro_batches)
        h = self.stage.h
        for b in self.grad_bufs:
            b.zero_()
        losses = []
        pending: List = []      # FIFO of (feed, cache)

        W = min(self.spec.pp - self.sid - 1, M)   # warmup forwards
        R = M - W

        # ---- warmup ----
        for i in range(W):
            act = self._recv_act() if not self.is_first else None
            out, saved = self._fwd(i, micro_batches, act)
            pending.append(saved)
            if not self.is_last:
                _p2p(self.comm, [(out, self.next)], [])

        act = None
        if R > 0 and not self.is_first:
            act = self._recv_act()

        # ---- steady 1F1B ----
        for i in range(R):
            out, saved = self._fwd(W + i, micro_batches, act)
            pending.append(saved)
            gin = None
            if self.is_last:
                losses.append(out)
            else:
                gin = torch.empty(tuple(out.shape), dtype=out.dtype,
                                  device=self.device)
                _p2p(self.comm, [(out, self.next)], [(gin, self.next)])
            if self.offloader is not None and len(pending) > 1 \
                    and not isinstance(pending[1][1], dict):
                self.offloader.prefetch(pending[1][1])
            dx = self._bwd(pending.pop(0), gin)
            if self.is_last:
                losses[-1] = losses[-1].clone()   # loss survives cache free
            last = (i == R - 1)
            if self.is_first:
                act = None
            elif last:
                _p2p(self.comm, [(dx, self.prev)], [])
            else:
                nxt = torch.empty(self._act_shape, dtype=self._act_dtype,
                                  device=self.device)
                _p2p(self.comm, [(dx, self.prev)], [(nxt, self.prev)])
                act = nxt

        # ---- cooldown ----
        for _ in range(W):
            gin = None
            if not self.is_last:
                gin = torch.empty(self._act_shape, dtype=self._act_dtype,
                                  device=self.device)
                _p2p(self.comm, [], [(gin, self.next)])
            dx = self._bwd(pending.pop(0), gin)
            if not self.is_first:
                _p2p(self.comm, [(dx, self.prev)], [])

        return self._finish(M, losses)

    def _step_gpipe(self, micro_batches: List[Dict]):
        """GPipe schedule (reference executable_graph.cc:803): every
        forward first, then every backward in reverse order; peak memory
        holds all M activation caches."""
        M = len(micro_batches)
        for b in self.grad_bufs:
            b.zero_()
        losses = []
        pending: List = []
        for i in range(M):
            act = self._recv_act() if not self.is_first else None
            out, saved = self._fwd(i, micro_batches, act)
            pending.append(saved)
            if self.is_last:
                losses.append(out.clone())
            else:
                _p2p(self.comm, [(out, self.next)], [])
        for i in reversed(range(M)):
            gin = None
            if not self.is_last:
                gin = torch.empty(self._act_shape, dtype=self._act_dtype,
                                  device=self.device)
                _p2p(self.comm, [], [(gin, self.next)])
            dx = self._bwd(pending.pop(), gin)
            if not self.is_first:
                _p2p(self.comm, [(dx, self.prev)], [])
        return self._finish(M, losses)

    def _finish(self, M: int, losses):
        h = self.stage.h
        # ---- shared-weight p2p (tied wte/lm_head): first and last stage
        # exchange + sum the tied parameter's accumulated grads so both
        # copies update identically (reference executable_graph.cc:929-933)
        tied = self.stage.h.get("tied_name")
        if tied is not None and self.spec.pp > 1 \
                and (self.is_first or self.is_last):
            idx = next(i for i, p in enumerate(self.params)
                       if p.name == tied)
            peer = self.spec.peer(self.rank,
                                  self.spec.pp - 1 if self.is_first else 0)
            buf = self.grad_bufs[idx]
            other = torch.empty_like(buf)
            _p2p(self.comm, [(buf, peer)], [(other, peer)])
            buf += other

        # ---- update: grad allreduce over dp + optimizer, once ----
        # per-micro-batch losses are token means; the step optimizes their
        # mean, so the accumulated grads divide by M
        for b in self.grad_bufs:
            b /= M
        if self.scaler is not None:
            self.scaler.unscale_(self.grad_bufs)
            if not self.scaler.check_and_update(self.grad_bufs):
                # overflow: skip the update, keep the backed-off scale
                if self.is_last and losses:
                    return torch.stack([l.float() for l in losses]).mean()
                return None
        feed = {ph: buf for ph, buf in zip(h["grad_phs"], self.grad_bufs)}
        self.stage.graph.run([h["train_op"]], feed, ctx=self.ctx)

        if self.is_last and losses:
            return torch.stack([l.float() for l in losses]).mean()
        return None
