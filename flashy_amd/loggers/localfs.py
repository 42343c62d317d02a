# Copyright (c) Flashy-AMD authors.
"""Filesystem experiment logger: media under ``<xp>/outputs/``, hyperparams
to ``hyperparams.json``; scalar metrics are a no-op (they already flow into
the persisted history).

Capability parity with the reference's ``flashy/loggers/localfs.py``
(media naming ``{prefix}_{step}/{key}.{wav|png|txt}`` when ``use_subdirs``,
flat ``{prefix}_{step}_{key}.ext`` otherwise; torchaudio/torchvision are
optional lazy deps — missing backends degrade to tensor dumps).
All methods rank-0 gated.
"""
from __future__ import annotations

import json
import typing as tp
from pathlib import Path

import torch

from .. import distrib
from ..utils import write_and_rename
from .base import ExperimentLogger


class LocalFSLogger(ExperimentLogger):
    def __init__(self, save_dir: Path, with_media_logging: bool = True,
                 use_subdirs: bool = False):
        super().__init__(with_media_logging, Path(save_dir))
        self.use_subdirs = use_subdirs

    @classmethod
    def from_xp(cls, with_media_logging: bool = True,
                sub_dir: str = "outputs", **kwargs) -> "LocalFSLogger":
        from .. import xp as _xp
        xp = _xp.get_xp()
        return cls(xp.folder / sub_dir, with_media_logging, **kwargs)

    def _media_path(self, prefix: str, key: str, step: tp.Optional[int], ext: str) -> Path:
        assert self.save_dir is not None
        stamp = f"{prefix}_{step}" if step is not None else prefix
        if self.use_subdirs:
            path = self.save_dir / stamp / f"{key}.{ext}"
        else:
            path = self.save_dir / f"{stamp}_{key}.{ext}"
        path.parent.mkdir(parents=True, exist_ok=True)
        return path

    @distrib.rank_zero_only
    def log_hyperparams(self, params: tp.Mapping[str, tp.Any],
                        metrics: tp.Optional[tp.Mapping[str, tp.Any]] = None) -> None:
        assert self.save_dir is not None
        self.save_dir.mkdir(parents=True, exist_ok=True)
        with write_and_rename(self.save_dir / "hyperparams.json", "w") as fh:
            json.dump(dict(params), fh, indent=1, default=repr)

    def log_metrics(self, prefix: str, metrics: tp.Mapping[str, tp.Any],
                    step: tp.Optional[int] = None) -> None:
        # scalar metrics are persisted through the XP history, not duplicated here
        del prefix, metrics, step

    @distrib.rank_zero_only
    def log_audio(self, prefix: str, key: str, audio: tp.Any, sample_rate: int,
                  step: tp.Optional[int] = None, **kwargs) -> None:
        if not self.with_media_logging:
            return
        audio = torch.as_tensor(audio).detach().cpu()
        try:
            import torchaudio
            path = self._media_path(prefix, key, step, "wav")
            n_ch = max(1, audio.shape[0] if audio.dim() > 1 else 1)
            torchaudio.save(str(path), audio.reshape(n_ch, -1),
                            sample_rate)
        except ImportError:
            path = self._media_path(prefix, key, step, "pt")
            torch.save({"audio": audio, "sample_rate": sample_rate}, path)

    @distrib.rank_zero_only
    def log_image(self, prefix: str, key: str, image: tp.Any,
                  step: tp.Optional[int] = None, **kwargs) -> None:
        if not self.with_media_logging:
            return
        image = torch.as_tensor(image).detach().cpu()
        try:
            import torchvision
            path = self._media_path(prefix, key, step, "png")
            torchvision.utils.save_image(image, str(path))
        except ImportError:
            path = self._media_path(prefix, key, step, "pt")
            torch.save(image, path)

    @distrib.rank_zero_only
    def log_text(self, prefix: str, key: str, text: str,
                 step: tp.Optional[int] = None, **kwargs) -> None:
        if not self.with_media_logging:
            return
        path = self._media_path(prefix, key, step, "txt")
        path.write_text(text)
