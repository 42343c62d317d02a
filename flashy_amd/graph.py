# Copyright (c) Flashy-AMD authors.
"""HIP-graph step capture: replace hundreds of per-step kernel launches with
one graph replay.

The MI355X-native answer to launch-bound small-batch training (a CIFAR-shaped
ResNet step at batch 64 is dominated by launch overhead on a 256-CU chip):
run the whole training step — forward, loss, backward, optimizer — once into
a ``hipGraph`` (``torch.cuda.CUDAGraph`` on ROCm) over static buffers, then
replay it each step.  Fresh data is copied into the static input buffers
before each replay.

Usage::

    static_x = torch.empty(bs, 3, 32, 32, device="cuda")
    static_y = torch.empty(bs, dtype=torch.long, device="cuda")

    def step():
        optimizer.zero_grad(set_to_none=False)
        loss = F.cross_entropy(model(static_x), static_y)
        loss.backward()
        optimizer.step()
        return loss

    graphed = CapturedStep(step).capture()
    for x, y in data:
        static_x.copy_(x, non_blocking=True)
        static_y.copy_(y, non_blocking=True)
        loss = graphed()          # one hipGraphLaunch

Notes:
 * warmup iterations run on a side stream so allocations (grads, optimizer
   state) exist before capture;
 * ``zero_grad(set_to_none=False)`` keeps gradient storage stable across
   replays;
 * on CPU (tests/CI) the same object degrades to calling ``fn`` eagerly.
"""
from __future__ import annotations

import os
import typing as tp

import torch


_capture_warmed = False


def _warm_capture_machinery() -> None:
    """Throwaway first capture (once per process): a tiny elementwise +
    GEMM + conv graph, captured and replayed, so lazy per-stream library
    state (BLAS workspaces, MIOpen descriptors, allocator graph-pool
    machinery) initializes OUTSIDE the first real graph's private pool.

    Defensive hygiene, not the NaN fix: the captured-step NaN chase
    (scripts/graph_nan_hunt.py) initially pointed here, but the
    per-process-stochastic culprit was MIOpen's implicit-GEMM solver class
    under replay (see capture() below).  Kept because first-capture lazy
    allocations landing in a training graph's pool remain a real hazard
    class and the cost is one tiny capture per process."""
    global _capture_warmed
    if _capture_warmed or not torch.cuda.is_available():
        return
    _capture_warmed = True
    a = torch.zeros(16, 16, device="cuda")
    b = torch.zeros(16, 16, device="cuda")
    img = torch.zeros(1, 8, 8, 8, device="cuda")
    wgt = torch.zeros(8, 8, 3, 3, device="cuda")
    side = torch.cuda.Stream()
    side.wait_stream(torch.cuda.current_stream())

    def tiny():
        c = a @ b
        d = torch.nn.functional.conv2d(img, wgt, padding=1)
        return c.sum() + d.sum()

    with torch.cuda.stream(side):
        tiny()
    torch.cuda.current_stream().wait_stream(side)
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        tiny()
    g.replay()
    torch.cuda.synchronize()


class CapturedStep:
    """Capture a closure over static tensors into a replayable HIP graph."""

    def __init__(self, fn: tp.Callable[[], tp.Any], warmup: int = 3,
                 pool: tp.Optional[tp.Any] = None):
        self.fn = fn
        self.warmup = warmup
        self.pool = pool
        self.graph: tp.Optional[torch.cuda.CUDAGraph] = None
        self.output: tp.Any = None

    @property
    def enabled(self) -> bool:
        return torch.cuda.is_available()

    def capture(self) -> "CapturedStep":
        if not self.enabled:
            return self
        # MIOpen's implicit-GEMM conv solvers mis-execute under hipGraph
        # REPLAY on this stack (ROCm 7.0/7.2, bf16): a torch-module model
        # captured with autocast replays NaN after 2-4 replays on ~50-70%
        # of processes, entering at the smallest conv.  Solver-class bisect
        # (scripts/graph_nan_hunt.py, 12-process matrix per class):
        # base 8/12 NaN, MIOPEN_DEBUG_CONV_IMPLICIT_GEMM=0 -> 0/12,
        # CONV_GEMM=0 (more shapes onto implicit-GEMM) -> 12/12.
        # Only affects torch-module (MIOpen) models — the native NHWC
        # kernels never touch MIOpen.  Set before the warmup below so the
        # find cache never selects the broken class for captured shapes.
        os.environ.setdefault("MIOPEN_DEBUG_CONV_IMPLICIT_GEMM", "0")
        # The autocast weight cache is incompatible with graph capture
        # (same rule as torch.cuda.make_graphed_callables): cached casts
        # allocated during capture are freed into the graph's private pool
        # at autocast exit, and later reuse of those blocks can race with
        # multi-stream library kernels on replay — observed as an
        # intermittent, box-dependent NaN after a few replays (GPUTEST_r01).
        # With the cache off, every cast is a recorded kernel whose output
        # block stays live for the whole graph.
        prev_cache = torch.is_autocast_cache_enabled()
        torch.set_autocast_cache_enabled(False)
        torch.clear_autocast_cache()
        try:
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(self.warmup):
                    self.fn()
            torch.cuda.current_stream().wait_stream(side)
            torch.cuda.synchronize()
            self.graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.graph, pool=self.pool):
                self.output = self.fn()
        finally:
            torch.set_autocast_cache_enabled(prev_cache)
        return self

    def replay(self) -> tp.Any:
        if self.graph is None:
            # eager fallback (CPU tests, or capture explicitly disabled)
            return self.fn()
        self.graph.replay()
        return self.output

    __call__ = replay
