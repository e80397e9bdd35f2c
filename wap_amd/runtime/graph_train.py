"""hipGraph-captured training step.

The WAP train step launches ~6-7k kernels (decoder unroll: T timesteps x
~20 small kernels, forward and backward) — measured launch/gap overhead is
a double-digit share of the step on MI355X.  For fixed-shape steps (bench;
padded buckets) the whole step is captured once and replayed:

  graph A: grad-buffer zero -> autocast forward -> backward
  (world>1: eager bucketed RCCL all-reduce on the flat grad buffers)
  graph B: grad clip + optimizer step

The all-reduce stays OUTSIDE the graphs: capturing RCCL collectives into
hipGraphs is not load-bearing here (grad volume ~60 MB, sub-ms over xGMI)
and keeping it eager avoids capture-mode collective hazards.  Gradients
already live in the flat per-bucket buffers of BucketedDDP, so the
between-graph all-reduce is one call per bucket, no packing.
"""

from __future__ import annotations

from typing import Dict, Optional

import torch
import torch.distributed as dist


_SYNC_SCRATCH = None


def safe_stream_sync() -> None:
    """Drain the current stream WITHOUT hipStreamSynchronize /
    hipDeviceSynchronize: on ROCm 7.x those corrupt the next hipGraph
    replay's backward (measured, deterministic on most boxes).  A D2H
    scalar read (hipMemcpyWithStream under .item()) drains the stream and
    is measured safe.  No-op on CPU (nothing to drain)."""
    global _SYNC_SCRATCH
    if not torch.cuda.is_available():
        return
    if _SYNC_SCRATCH is None or not _SYNC_SCRATCH.is_cuda:
        _SYNC_SCRATCH = torch.zeros(1, device="cuda")
    _SYNC_SCRATCH.item()


def _capture_mode() -> str:
    """hipGraph capture mode: thread_local allows other host threads (the
    pinned-memory prefetcher) to touch HIP during capture; override with
    WAP_CAPTURE_MODE=global for A/B."""
    import os
    return os.environ.get("WAP_CAPTURE_MODE", "thread_local")



class GraphedTrainStep:
    """Capture-and-replay train step over static input buffers.

    Usage::

        gstep = GraphedTrainStep(model, ddp, opt, clip=100.0,
                                 autocast_bf16=True)
        gstep.initialize(batch)      # 2 eager warmups + capture
        loss_t = gstep.step(batch)   # replay (copies batch in)
    """

    def __init__(self, model, ddp, optimizer, clip: float = 100.0,
                 autocast_bf16: bool = True):
        self.model = model
        self.ddp = ddp
        self.opt = optimizer
        self.clip = clip
        self.autocast_bf16 = autocast_bf16
        self.buf: Dict[str, torch.Tensor] = {}
        self.loss: Optional[torch.Tensor] = None
        self.g_fwdbwd: Optional[torch.cuda.CUDAGraph] = None
        self.g_opt: Optional[torch.cuda.CUDAGraph] = None

    def _autocast(self):
        return torch.autocast("cuda", dtype=torch.bfloat16,
                              enabled=self.autocast_bf16)

    def _fwd_bwd(self):
        for b in self.ddp.buckets:
            b.buffer.zero_()
        with self._autocast():
            out = self.model(self.buf["x"], self.buf["x_mask"],
                             self.buf["y"], self.buf["y_mask"])
        out["loss"].backward()
        return out["loss"]

    def _clip_opt(self):
        import os
        if self.clip > 0 and os.environ.get("WAP_NO_CLIP", "0") != "1":
            self.ddp.clip_grad_norm_(self.clip)
        self.opt.step()

    def _allreduce(self):
        if self.ddp.world_size > 1:
            handles = [
                dist.all_reduce(b.buffer, op=dist.ReduceOp.SUM,
                                group=self.ddp.group, async_op=True)
                for b in self.ddp.buckets
            ]
            for h, b in zip(handles, self.ddp.buckets):
                h.wait()
                b.buffer.div_(self.ddp.world_size)

    def initialize(self, batch: Dict[str, torch.Tensor]) -> None:
        """Allocate static buffers, run eager warmups, capture both graphs.
        Must be called with hooks disabled for world>1 (we all-reduce
        manually between the graphs)."""
        self.ddp.sync_enabled = False  # we all-reduce between the graphs
        # torch optimizers require capturable=True (device-side step state)
        # to be captured into a graph
        for g in self.opt.param_groups:
            g["capturable"] = True
        for k in ("x", "x_mask", "y", "y_mask"):
            self.buf[k] = batch[k].clone()
        # Warmup (MIOpen algo search, allocator, optimizer state init) on a
        # side stream, the torch.cuda.graph idiom.
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self._fwd_bwd()
                self._allreduce()
                self._clip_opt()
        torch.cuda.current_stream().wait_stream(s)
        safe_stream_sync()

        # MEASURED HAZARD (ROCm 7.x): two back-to-back hipGraph replays on
        # one stream can RACE (the second reads grads before the first's
        # backward landed) -> exploding grads after a few steps.  For
        # world==1 capture the WHOLE step as ONE graph; for world>1 keep
        # two graphs with an explicit synchronize between the phases
        # (step() below).
        if self.ddp.world_size == 1:
            self.g_all = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.g_all, capture_error_mode=_capture_mode()):
                self.loss = self._fwd_bwd()
                self._clip_opt()
        else:
            self.g_all = None
            self.g_fwdbwd = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.g_fwdbwd,
                                  capture_error_mode=_capture_mode()):
                self.loss = self._fwd_bwd()
            self._allreduce()
            self.g_opt = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.g_opt, capture_error_mode=_capture_mode()):
                self._clip_opt()

    def step(self, batch: Optional[Dict[str, torch.Tensor]] = None) -> torch.Tensor:
        """One training step; returns the (device) loss tensor."""
        if batch is not None:
            for k in self.buf:
                self.buf[k].copy_(batch[k], non_blocking=True)
        if self.g_all is not None:
            self.g_all.replay()
            return self.loss
        self.g_fwdbwd.replay()
        safe_stream_sync()  # see docstring: never hipStreamSynchronize here
        self._allreduce()
        self.g_opt.replay()
        return self.loss


class _ShapeGraph:
    """A captured fwd+bwd for one batch-shape: static input buffers + graph."""

    def __init__(self, trainer: "MultiShapeGraphedTrainer",
                 batch: Dict[str, torch.Tensor]):
        self.t = trainer
        self.buf = {k: batch[k].clone() for k in ("x", "x_mask", "y", "y_mask")}
        # Warmup fwd+bwd only — NO collectives, NO optimizer: lazy capture
        # must not disturb the per-step all-reduce alignment across ranks.
        # 3 iterations: backward capture needs the autograd engine fully
        # warmed (fewer warmups segfaulted capture_end on MI355X).
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):
                self._fwd_bwd()
        torch.cuda.current_stream().wait_stream(s)
        safe_stream_sync()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph, capture_error_mode=_capture_mode()):
            self.loss = self._fwd_bwd()
            if trainer.ddp.world_size == 1:
                # single-rank: fold clip+opt into the same graph (avoids the
                # replay-replay race) once optimizer state exists
                trainer._clip_opt_ops()
                self.has_opt = True
            else:
                self.has_opt = False
        # NOTE: stream capture RECORDS the ops without executing them — the
        # caller must replay() to actually compute this batch.

    def _fwd_bwd(self):
        for b in self.t.ddp.buckets:
            b.buffer.zero_()
        with torch.autocast("cuda", dtype=torch.bfloat16,
                            enabled=self.t.autocast_bf16):
            out = self.t.model(self.buf["x"], self.buf["x_mask"],
                               self.buf["y"], self.buf["y_mask"])
        out["loss"].backward()
        return out["loss"]


class _EagerShapeStep:
    """Capture-free stand-in for _ShapeGraph (same interface): executes
    fwd+bwd eagerly on replay().  Used when graphs are disabled so the
    trainer's step() control flow — shape cache, capture branch, eager
    fallback, collective placement — runs (and is testable) on CPU."""

    def __init__(self, trainer: "MultiShapeGraphedTrainer",
                 batch: Dict[str, torch.Tensor]):
        self.t = trainer
        self.buf = {k: batch[k].clone() for k in ("x", "x_mask", "y", "y_mask")}
        self.loss: Optional[torch.Tensor] = None
        self.has_opt = False
        self.graph = self

    def replay(self) -> None:
        self.loss = self.t._eager_fwd_bwd(self.buf)


class MultiShapeGraphedTrainer:
    """Graphed training over VARIABLE bucket shapes (train.py).

    - fwd+bwd graphs are captured lazily per shape, with no collectives in
      the capture path, so ranks may capture at different times without
      breaking all-reduce alignment (each step still performs exactly one
      bucketed all-reduce sweep + one optimizer step on every rank);
    - the clip+optimizer graph is captured once, at the first step (all
      ranks reach it together);
    - shapes beyond ``max_graphs`` run an eager fwd+bwd fallback with the
      same collective pattern.

    Use together with ``prepare_batch(pad_to=...)`` so bucket shapes
    collapse onto a small grid.
    """

    def __init__(self, model, ddp, optimizer, clip: float = 100.0,
                 autocast_bf16: bool = True, max_graphs: int = 24,
                 graphs_enabled: Optional[bool] = None):
        self.model = model
        self.ddp = ddp
        self.opt = optimizer
        self.clip = clip
        self.autocast_bf16 = autocast_bf16
        self.max_graphs = max_graphs
        self.cache: Dict[tuple, _ShapeGraph] = {}
        self._failed: set = set()
        self.g_opt: Optional[torch.cuda.CUDAGraph] = None
        self.n_eager_fallback = 0
        # graphs_enabled=False runs the IDENTICAL step() control flow
        # (shape cache, capture branch, eager fallback, collective
        # placement) with eager execution standing in for capture/replay —
        # this is what the CPU world-4 collective-alignment test exercises
        # (tests/test_ddp_cpu.py), so the distributed step logic is covered
        # without a GPU.
        self.graphs_enabled = (torch.cuda.is_available()
                               if graphs_enabled is None else graphs_enabled)
        ddp.sync_enabled = False
        for g in self.opt.param_groups:
            g["capturable"] = self.graphs_enabled

    def _allreduce(self):
        if self.ddp.world_size > 1:
            handles = [
                dist.all_reduce(b.buffer, op=dist.ReduceOp.SUM,
                                group=self.ddp.group, async_op=True)
                for b in self.ddp.buckets
            ]
            for h, b in zip(handles, self.ddp.buckets):
                h.wait()
                b.buffer.div_(self.ddp.world_size)

    def _clip_opt_ops(self):
        if self.clip > 0:
            self.ddp.clip_grad_norm_(self.clip)
        self.opt.step()

    def _opt_step(self):
        if not self.graphs_enabled:
            self._clip_opt_ops()
            return
        if self.g_opt is not None:
            self.g_opt.replay()
            return
        # First use on every rank: capture the clip+opt graph, then replay
        # it to actually apply THIS step's update (capture only records).
        safe_stream_sync()
        self.g_opt = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.g_opt, capture_error_mode=_capture_mode()):
            self._clip_opt_ops()
        self.g_opt.replay()

    def _eager_fwd_bwd(self, batch):
        for b in self.ddp.buckets:
            b.buffer.zero_()
        with torch.autocast("cuda", dtype=torch.bfloat16,
                            enabled=self.autocast_bf16):
            out = self.model(batch["x"], batch["x_mask"],
                             batch["y"], batch["y_mask"])
        out["loss"].backward()
        return out["loss"]

    def step(self, batch: Dict[str, torch.Tensor]) -> torch.Tensor:
        if not self.opt.state:
            # Very first step: all-eager (initializes optimizer state so
            # every captured shape graph can include clip+opt at world==1).
            # Run it on a SIDE stream: the process's first backward creates
            # the per-param AccumulateGrad nodes bound to the current
            # stream, and default-stream-bound AccumulateGrad breaks later
            # graph captures (segfault in capture_end; torch warns about
            # exactly this).
            if self.graphs_enabled:
                s = torch.cuda.Stream()
                s.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(s):
                    loss = self._eager_fwd_bwd(batch)
                torch.cuda.current_stream().wait_stream(s)
            else:
                loss = self._eager_fwd_bwd(batch)
            self._allreduce()
            self._clip_opt_ops()
            return loss
        key = tuple(batch["x"].shape) + tuple(batch["y"].shape)
        sg = self.cache.get(key)
        if sg is not None:
            for k in sg.buf:
                sg.buf[k].copy_(batch[k], non_blocking=True)
            sg.graph.replay()
            if sg.has_opt:
                return sg.loss
            loss = sg.loss
        elif len(self.cache) < self.max_graphs and key not in self._failed:
            # Optimizer state must exist before its graph can be captured;
            # the Adadelta state dict is lazily created at the first eager
            # step, so the opt graph is only captured from step 2 on.
            try:
                sg = (_ShapeGraph(self, batch) if self.graphs_enabled
                      else _EagerShapeStep(self, batch))
                self.cache[key] = sg
                sg.graph.replay()  # capture records only; compute this batch
                if sg.has_opt:
                    return sg.loss
                loss = sg.loss
            except Exception as e:
                import sys
                print(f"[graph_train] capture failed for {key}: {e!r}; "
                      "eager fallback", file=sys.stderr)
                self._failed.add(key)
                loss = self._eager_fwd_bwd(batch)
        else:
            self.n_eager_fallback += 1
            loss = self._eager_fwd_bwd(batch)
        # split path (world>1, or a shape captured without opt):
        safe_stream_sync()
        self._allreduce()
        self._opt_step()
        return loss
