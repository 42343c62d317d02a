# Copyright (c) Flashy-AMD authors.
"""TensorBoard experiment logger (soft dependency).

Capability parity with the reference's ``flashy/loggers/tensorboard.py``:
``SummaryWriter`` in ``<xp>/tensorboard/``, scalars, hparams, audio, image,
text.  Import of tensorboard is deferred and failure raises a clear error
only when the backend is actually requested.
"""
from __future__ import annotations

import typing as tp
from pathlib import Path

import torch

from .. import distrib
from .base import ExperimentLogger
from .utils import _add_prefix, _flatten_dict, _sanitize_params


class TensorboardLogger(ExperimentLogger):
    def __init__(self, save_dir: Path, with_media_logging: bool = True, **writer_kwargs):
        super().__init__(with_media_logging, Path(save_dir))
        self._writer = None
        self._writer_kwargs = writer_kwargs

    @classmethod
    def from_xp(cls, with_media_logging: bool = True,
                sub_dir: str = "tensorboard", **kwargs) -> "TensorboardLogger":
        from .. import xp as _xp
        xp = _xp.get_xp()
        return cls(xp.folder / sub_dir, with_media_logging, **kwargs)

    @property
    def writer(self):
        if self._writer is None:
            try:
                from torch.utils.tensorboard import SummaryWriter
            except ImportError as exc:
                raise RuntimeError(
                    "tensorboard is not installed; `pip install tensorboard` "
                    "or skip init_tensorboard()") from exc
            assert self.save_dir is not None
            self.save_dir.mkdir(parents=True, exist_ok=True)
            self._writer = SummaryWriter(str(self.save_dir), **self._writer_kwargs)
        return self._writer

    @distrib.rank_zero_only
    def log_hyperparams(self, params: tp.Mapping[str, tp.Any],
                        metrics: tp.Optional[tp.Mapping[str, tp.Any]] = None) -> None:
        params = _sanitize_params(_flatten_dict(dict(params)))
        metrics = dict(metrics) if metrics else {}
        if metrics:
            self.writer.add_hparams(params, metrics)
        else:
            for key, value in params.items():
                self.writer.add_text(f"hparams/{key}", str(value))

    @distrib.rank_zero_only
    def log_metrics(self, prefix: str, metrics: tp.Mapping[str, tp.Any],
                    step: tp.Optional[int] = None) -> None:
        flat = _add_prefix(_flatten_dict(dict(metrics)), prefix, self.group_separator)
        for key, value in flat.items():
            if isinstance(value, (int, float)) or (
                    torch.is_tensor(value) and value.numel() == 1):
                self.writer.add_scalar(key, value, global_step=step)

    @distrib.rank_zero_only
    def log_audio(self, prefix: str, key: str, audio: tp.Any, sample_rate: int,
                  step: tp.Optional[int] = None, **kwargs) -> None:
        if not self.with_media_logging:
            return
        tag = f"{prefix}{self.group_separator}{key}"
        self.writer.add_audio(tag, torch.as_tensor(audio), global_step=step,
                              sample_rate=sample_rate)

    @distrib.rank_zero_only
    def log_image(self, prefix: str, key: str, image: tp.Any,
                  step: tp.Optional[int] = None, **kwargs) -> None:
        if not self.with_media_logging:
            return
        tag = f"{prefix}{self.group_separator}{key}"
        self.writer.add_image(tag, torch.as_tensor(image), global_step=step, **kwargs)

    @distrib.rank_zero_only
    def log_text(self, prefix: str, key: str, text: str,
                 step: tp.Optional[int] = None, **kwargs) -> None:
        if not self.with_media_logging:
            return
        tag = f"{prefix}{self.group_separator}{key}"
        self.writer.add_text(tag, text, global_step=step)
