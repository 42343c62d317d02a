# Copyright (c) Flashy-AMD authors.
"""Backend interface for experiment logging.

Capability parity with the reference's ``flashy/loggers/base.py:12-104``
(hyperparams / metrics / audio / image / text, ``with_media_logging`` gating,
``save_dir``).  All media methods use the uniform ``(prefix, key, ...)``
argument order — the reference's localfs/tensorboard had it swapped
(SURVEY.md §8.2); here every backend agrees.
"""
from __future__ import annotations

import typing as tp
from abc import ABC, abstractmethod
from pathlib import Path


class ExperimentLogger(ABC):
    group_separator = "/"

    def __init__(self, with_media_logging: bool = True,
                 save_dir: tp.Optional[Path] = None):
        self.with_media_logging = with_media_logging
        self.save_dir = Path(save_dir) if save_dir is not None else None

    @abstractmethod
    def log_hyperparams(self, params: tp.Mapping[str, tp.Any],
                        metrics: tp.Optional[tp.Mapping[str, tp.Any]] = None) -> None:
        ...

    @abstractmethod
    def log_metrics(self, prefix: str, metrics: tp.Mapping[str, tp.Any],
                    step: tp.Optional[int] = None) -> None:
        ...

    def log_audio(self, prefix: str, key: str, audio: tp.Any, sample_rate: int,
                  step: tp.Optional[int] = None, **kwargs) -> None:
        del prefix, key, audio, sample_rate, step, kwargs

    def log_image(self, prefix: str, key: str, image: tp.Any,
                  step: tp.Optional[int] = None, **kwargs) -> None:
        del prefix, key, image, step, kwargs

    def log_text(self, prefix: str, key: str, text: str,
                 step: tp.Optional[int] = None, **kwargs) -> None:
        del prefix, key, text, step, kwargs
