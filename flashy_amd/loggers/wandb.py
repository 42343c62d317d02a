# Copyright (c) Flashy-AMD authors.
"""Weights & Biases experiment logger (soft dependency).

Capability parity with the reference's ``flashy/loggers/wandb.py``: a wandb
run keyed by the XP signature, resume detected via a ``wandb_flag`` touch
file in the XP folder; metrics/audio/image/text.  Unlike the reference,
scalar metric logging is NOT gated on ``with_media_logging``
(SURVEY.md §8.3 bug fixed) and the media methods use (prefix, key, ...).
"""
from __future__ import annotations

import typing as tp
from pathlib import Path

import torch

from .. import distrib
from .base import ExperimentLogger
from .utils import _add_prefix, _flatten_dict, _sanitize_params


class WandbLogger(ExperimentLogger):
    def __init__(self, save_dir: Path, run_id: str, resume: bool,
                 with_media_logging: bool = True, project: tp.Optional[str] = None,
                 name: tp.Optional[str] = None, group: tp.Optional[str] = None,
                 **init_kwargs):
        super().__init__(with_media_logging, Path(save_dir))
        try:
            import wandb
        except ImportError as exc:
            raise RuntimeError(
                "wandb is not installed; `pip install wandb` or skip init_wandb()") from exc
        self.wandb = wandb
        self._run = None
        if distrib.is_rank_zero():
            self._run = wandb.init(dir=str(save_dir), id=run_id, resume=resume,
                                   project=project, name=name or run_id,
                                   group=group, **init_kwargs)

    @classmethod
    def from_xp(cls, with_media_logging: bool = True, **kwargs) -> "WandbLogger":
        from .. import xp as _xp
        xp = _xp.get_xp()
        flag = xp.folder / "wandb_flag"
        resume = flag.exists()
        if distrib.is_rank_zero():
            xp.folder.mkdir(parents=True, exist_ok=True)
            flag.touch()
        return cls(xp.folder, run_id=xp.sig, resume=resume,
                   with_media_logging=with_media_logging, **kwargs)

    @distrib.rank_zero_only
    def log_hyperparams(self, params: tp.Mapping[str, tp.Any],
                        metrics: tp.Optional[tp.Mapping[str, tp.Any]] = None) -> None:
        if self._run is None:
            return
        self._run.config.update(_sanitize_params(_flatten_dict(dict(params))),
                                allow_val_change=True)

    @distrib.rank_zero_only
    def log_metrics(self, prefix: str, metrics: tp.Mapping[str, tp.Any],
                    step: tp.Optional[int] = None) -> None:
        if self._run is None:
            return
        flat = _add_prefix(_flatten_dict(dict(metrics)), prefix, self.group_separator)
        scalars = {k: v for k, v in flat.items()
                   if isinstance(v, (int, float))
                   or (torch.is_tensor(v) and v.numel() == 1)}
        self.wandb.log(scalars, step=step)

    @distrib.rank_zero_only
    def log_audio(self, prefix: str, key: str, audio: tp.Any, sample_rate: int,
                  step: tp.Optional[int] = None, **kwargs) -> None:
        if self._run is None or not self.with_media_logging:
            return
        tag = f"{prefix}{self.group_separator}{key}"
        audio = torch.as_tensor(audio).detach().cpu().reshape(-1).numpy()
        self.wandb.log({tag: self.wandb.Audio(audio, sample_rate=sample_rate)}, step=step)

    @distrib.rank_zero_only
    def log_image(self, prefix: str, key: str, image: tp.Any,
                  step: tp.Optional[int] = None, **kwargs) -> None:
        if self._run is None or not self.with_media_logging:
            return
        tag = f"{prefix}{self.group_separator}{key}"
        self.wandb.log({tag: self.wandb.Image(torch.as_tensor(image).detach().cpu())},
                       step=step)

    @distrib.rank_zero_only
    def log_text(self, prefix: str, key: str, text: str,
                 step: tp.Optional[int] = None, **kwargs) -> None:
        if self._run is None or not self.with_media_logging:
            return
        tag = f"{prefix}{self.group_separator}{key}"
        self.wandb.log({tag: self.wandb.Html(f"<pre>{text}</pre>")}, step=step)
