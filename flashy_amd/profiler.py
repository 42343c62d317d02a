# Copyright (c) Flashy-AMD authors.
"""Lightweight tracing helpers (the reference has none — SURVEY.md §5.1; this
is the capability-add counterpart to the per-stage ``duration`` metric).

Two levels:

* :func:`trace` — a context manager around ``torch.profiler`` exporting a
  chrome trace (works on ROCm: kineto reads the HIP activity stream)::

      with flashy_amd.profiler.trace(xp.folder / "trace.json"):
          solver.run_stage("train", ...)

* :class:`StageTimer` — cheap named wall/device timers with an EMA summary,
  for always-on coarse timing without the profiler overhead.

For per-kernel counters use rocprofv3 externally (see profiles/README.md for
the recipe used on MI355X).
"""
from __future__ import annotations

import time
import typing as tp
from contextlib import contextmanager
from pathlib import Path

import torch


@contextmanager
def trace(out_path: tp.Union[str, Path], activities: tp.Optional[list] = None,
          record_shapes: bool = False):
    """Profile the enclosed block and export a chrome trace to ``out_path``."""
    from torch.profiler import ProfilerActivity, profile
    if activities is None:
        activities = [ProfilerActivity.CPU]
        if torch.cuda.is_available():
            activities.append(ProfilerActivity.CUDA)
    with profile(activities=activities, record_shapes=record_shapes) as prof:
        yield prof
    Path(out_path).parent.mkdir(parents=True, exist_ok=True)
    prof.export_chrome_trace(str(out_path))


class StageTimer:
    """Named timers with device-sync'ed boundaries and running means."""

    def __init__(self, sync: bool = True):
        self.sync = sync and torch.cuda.is_available()
        self.totals: tp.Dict[str, float] = {}
        self.counts: tp.Dict[str, int] = {}

    @contextmanager
    def __call__(self, name: str):
        if self.sync:
            torch.cuda.synchronize()
        begin = time.perf_counter()
        try:
            yield
        finally:
            if self.sync:
                torch.cuda.synchronize()
            self.totals[name] = self.totals.get(name, 0.0) + \
                (time.perf_counter() - begin)
            self.counts[name] = self.counts.get(name, 0) + 1

    def summary(self) -> tp.Dict[str, float]:
        """Mean seconds per named region."""
        return {k: self.totals[k] / self.counts[k] for k in self.totals}
