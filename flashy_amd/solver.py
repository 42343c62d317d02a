# Copyright (c) Flashy-AMD authors.
"""BaseSolver: the epoch/stage training-loop scaffolding.

API parity with the reference's ``flashy/solver.py:30-211`` (BaseSolver with
``register_stateful`` / ``run_stage`` / ``commit`` / ``restore`` / ``epoch``
/ ``history`` / ``log_*`` / ``init_tensorboard`` / ``init_wandb``), on the
in-house XP runtime instead of Dora.

Semantics kept exactly:

* ``epoch`` is 1-indexed: ``len(history) + 1``; resuming is implicit — after
  ``restore()`` repopulates history, ``for epoch in range(self.epoch, N+1)``
  continues where the run stopped.
* ``register_stateful`` walks dotted attribute paths with late binding (the
  attribute is resolved at save/load time).
* ``run_stage`` forbids nesting, injects ``metrics["duration"]`` (stage
  wall-clock seconds) and logs the returned metrics dict under the stage
  name; ``None`` becomes ``{}``.
* ``log_metrics`` raises if a stage is logged twice within one epoch.
* ``commit`` appends pending metrics to history on every rank, then rank 0
  persists history and atomically writes the checkpoint.
* ``restore`` loads the checkpoint to CPU on all ranks and restores in place.

Checkpoints self-describe: ``xp.cfg`` and ``xp.sig`` are registered
write-only, so they are saved but never restored over the live run.
"""
from __future__ import annotations

import logging
import time
import typing as tp
from abc import ABC, abstractmethod
from pathlib import Path

from . import checkpoint as _checkpoint
from . import distrib
from . import xp as _xp
from .formatter import Formatter
from .logging import LogProgressBar, ResultLogger
from .state import AttributeWrapper, StateManager, WriteOnlyWrapper

# drop-in alias kept from the reference API (flashy/solver.py):
# a stage method as passed to run_stage.
StageCallable = tp.Callable[..., tp.Optional[tp.Dict[str, tp.Any]]]

logger = logging.getLogger(__name__)


class BaseSolver(ABC):
    checkpoint_name = "checkpoint.th"

    def __init__(self):
        self.logger = logging.getLogger(self.__class__.__module__)
        self.stateful = StateManager()
        self.xp = _xp.get_xp()
        self.register_stateful("history")
        self.register_stateful("xp.cfg", "xp.sig", write_only=True)
        self.result_logger = ResultLogger(self.logger)
        self._current_stage: tp.Optional[str] = None
        self._current_formatter: tp.Optional[Formatter] = None
        self._pending_metrics: tp.Dict[str, tp.Dict[str, tp.Any]] = {}

    # -- identity ----------------------------------------------------------
    @property
    def folder(self) -> Path:
        return self.xp.folder

    @property
    def checkpoint_path(self) -> Path:
        return self.folder / self.checkpoint_name

    @property
    def history(self) -> tp.List[tp.Dict[str, tp.Any]]:
        # Not owned: proxies the XP link so the persisted history and the
        # in-memory one cannot diverge.
        return self.xp.link.history

    @property
    def epoch(self) -> int:
        """1-indexed current epoch = completed epochs + 1."""
        return len(self.history) + 1

    # -- stateful registry -------------------------------------------------
    def register_stateful(self, *names: str, write_only: bool = False) -> None:
        """Register dotted attribute paths as checkpointed state.

        ``'xp.cfg'`` registers attribute ``cfg`` on owner ``self.xp``.  The
        leaf attribute is looked up lazily at save/load time, so it may be
        (re)assigned after registration.  ``write_only`` state is saved but
        never restored.
        """
        klass = WriteOnlyWrapper if write_only else AttributeWrapper
        for name in names:
            owner = self
            *path, leaf = name.split(".")
            for part in path:
                owner = getattr(owner, part)
            self.stateful.register(name, klass(owner, leaf))

    # -- logging backends --------------------------------------------------
    def init_tensorboard(self, **kwargs) -> None:
        self.result_logger.init_tensorboard(**kwargs)

    def init_wandb(self, **kwargs) -> None:
        self.result_logger.init_wandb(**kwargs)

    # -- formatting / progress ---------------------------------------------
    def get_formatter(self, stage_name: str) -> Formatter:
        """Override to customize per-stage metric formatting."""
        del stage_name
        return Formatter()

    @property
    def current_stage(self) -> tp.Optional[str]:
        return self._current_stage

    def log_progress(self, stage_name: str, iterable: tp.Iterable,
                     updates: int = 5, total: tp.Optional[int] = None,
                     **kwargs) -> LogProgressBar:
        formatter = self._current_formatter if self._current_stage == stage_name \
            else self.get_formatter(stage_name)
        name = f"{stage_name.capitalize()} | Epoch {self.epoch}"
        return LogProgressBar(self.logger, iterable, updates=updates, total=total,
                              name=name, formatter=formatter, **kwargs)

    # -- metrics -----------------------------------------------------------
    def log_metrics(self, stage_name: str, metrics: tp.Mapping[str, tp.Any]) -> None:
        if stage_name in self._pending_metrics:
            raise RuntimeError(
                f"stage {stage_name!r} was already logged for epoch {self.epoch}")
        metrics = dict(metrics)
        self._pending_metrics[stage_name] = metrics
        formatter = self._current_formatter if self._current_stage == stage_name \
            else self.get_formatter(stage_name)
        self.result_logger.log_metrics(stage_name, metrics, step=self.epoch,
                                       step_name="epoch", formatter=formatter)

    def log_hyperparams(self, params: tp.Mapping[str, tp.Any],
                        metrics: tp.Optional[tp.Mapping[str, tp.Any]] = None) -> None:
        self.result_logger.log_hyperparams(params, metrics)

    def log_audio(self, stage_name: str, key: str, audio: tp.Any,
                  sample_rate: int, **kwargs) -> None:
        self.result_logger.log_audio(stage_name, key, audio, sample_rate,
                                     step=self.epoch, **kwargs)

    def log_image(self, stage_name: str, key: str, image: tp.Any, **kwargs) -> None:
        self.result_logger.log_image(stage_name, key, image, step=self.epoch, **kwargs)

    def log_text(self, stage_name: str, key: str, text: str, **kwargs) -> None:
        self.result_logger.log_text(stage_name, key, text, step=self.epoch, **kwargs)

    # -- stages ------------------------------------------------------------
    def run_stage(self, stage_name: str, method: tp.Callable, *args, **kwargs):
        """Run one named stage of the current epoch and log its metrics.

        The stage method returns a metrics dict (or None).  Wall-clock
        ``duration`` is injected.  Stages cannot nest.
        """
        if self._current_stage is not None:
            raise RuntimeError(
                f"cannot start stage {stage_name!r} inside stage {self._current_stage!r}")
        self._current_stage = stage_name
        self._current_formatter = self.get_formatter(stage_name)
        begin = time.time()
        try:
            metrics = method(*args, **kwargs)
            if metrics is None:
                metrics = {}
            metrics = dict(metrics)
            metrics["duration"] = time.time() - begin
            self.log_metrics(stage_name, metrics)
            return metrics
        finally:
            self._current_stage = None
            self._current_formatter = None

    # -- commit / restore --------------------------------------------------
    #: opt-in: overlap the pickle+disk half of the checkpoint with the next
    #: epoch (device-to-host staging still happens inside commit, so the
    #: state snapshot is consistent; durability is deferred one commit).
    async_checkpoint: bool = False
    _async_ckpt: tp.Optional[_checkpoint.AsyncCheckpointer] = None

    def commit(self, save_checkpoint: bool = True) -> None:
        """End the epoch: push pending metrics to history (every rank — the
        epoch counter must advance identically everywhere), then on rank 0
        persist history and atomically write the checkpoint."""
        self.history.append(self._pending_metrics)
        self._pending_metrics = {}
        if distrib.is_rank_zero():
            self.xp.link.update_history(self.history)
            if save_checkpoint:
                state = self.stateful.state_dict()
                if self.async_checkpoint:
                    if self._async_ckpt is None:
                        self._async_ckpt = _checkpoint.AsyncCheckpointer()
                    self._async_ckpt.save(state, self.checkpoint_path)
                else:
                    _checkpoint.save_state(state, self.checkpoint_path)
                self.logger.debug("checkpoint saved to %s", self.checkpoint_path)

    def finalize_checkpoint(self) -> None:
        """Join any in-flight async checkpoint write (call at run end)."""
        if self._async_ckpt is not None:
            self._async_ckpt.wait()

    def restore(self) -> bool:
        """Load + restore the checkpoint if one exists.  Returns True when a
        checkpoint was restored.  All ranks read the file (to CPU), restore
        in place, and the epoch property advances via the restored history."""
        if not self.checkpoint_path.exists():
            return False
        self.logger.info("restoring from %s", self.checkpoint_path)
        state = _checkpoint.load_state(self.checkpoint_path)
        self.stateful.load_state_dict(state)
        return True

    # -- entry -------------------------------------------------------------
    @abstractmethod
    def run(self) -> tp.Any:
        ...
