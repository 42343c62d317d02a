# Copyright (c) Flashy-AMD authors.
"""Experiment (XP) runtime: run identity, folders, persisted history.

In-house replacement for the Dora surface the reference consumes
(/root/reference/flashy/solver.py:16,33 ``get_xp``,
``xp.link.history``/``update_history``): an :class:`XP` carries the run
signature (config content hash), the per-run artifact folder, the config, and
a :class:`Link` that persists the per-epoch metric history as JSON with atomic
updates so it survives restarts.

The canonical entry point is :func:`entry_point`::

    main = entry_point("my_pkg.train", config_path=Path(__file__).parent / "conf")

    @main.bind
    def run(cfg):
        Solver(cfg).run()

    if __name__ == "__main__":
        main()          # parses sys.argv overrides, enters the XP, runs

Programmatic lookup (notebook workflow) mirrors Dora's
``main.get_xp([])`` / ``get_xp_from_sig``: both return an :class:`XP` that can
be ``enter()``-ed to make it the process-wide current XP.
"""
from __future__ import annotations

import json
import logging
import os
import sys
import typing as tp
from pathlib import Path

from .config import Config, apply_overrides, load_config, signature
from .utils import write_and_rename

logger = logging.getLogger(__name__)

_current_xp: tp.Optional["XP"] = None

DEFAULT_ROOT_ENV = "_FLASHY_AMD_DIR"


def default_root() -> Path:
    env = os.environ.get(DEFAULT_ROOT_ENV)
    if env:
        return Path(env)
    return Path.home() / "flashy_amd_runs"


class Link:
    """Persisted per-epoch metric history (``history.json`` in the XP folder)."""

    def __init__(self, folder: Path):
        self.folder = folder
        self.history: tp.List[tp.Dict[str, tp.Any]] = []

    @property
    def _path(self) -> Path:
        return self.folder / "history.json"

    def load(self) -> tp.List[tp.Dict[str, tp.Any]]:
        if self._path.exists():
            with open(self._path) as fh:
                self.history = json.load(fh)
        return self.history

    def update_history(self, history: tp.List[tp.Dict[str, tp.Any]]) -> None:
        # Keep our copy in sync and write atomically: a kill mid-write leaves
        # the previous history intact (write_and_rename durability point).
        self.history = list(history)
        self.folder.mkdir(parents=True, exist_ok=True)
        with write_and_rename(self._path, "w") as fh:
            json.dump(self.history, fh, indent=1, default=repr)


class XP:
    """One experiment: signature, folder, config, history link."""

    def __init__(self, sig: str, folder: Path, cfg: Config):
        self.sig = sig
        self.folder = folder
        self.cfg = cfg
        self.link = Link(folder)

    def enter(self) -> "XP":
        """Make this the process-wide current XP (returned by get_xp())."""
        global _current_xp
        self.folder.mkdir(parents=True, exist_ok=True)
        self.link.load()
        _current_xp = self
        return self

    def __repr__(self) -> str:
        return f"XP(sig={self.sig}, folder={self.folder})"


def get_xp() -> XP:
    if _current_xp is None:
        raise RuntimeError(
            "No current XP. Run through an entry_point, or build one with "
            "flashy_amd.xp.create_xp(cfg).enter().")
    return _current_xp


def is_xp_active() -> bool:
    return _current_xp is not None


def create_xp(cfg: tp.Union[Config, dict], root: tp.Optional[Path] = None) -> XP:
    """Build an XP from a config: sig = content hash, folder = <root>/xps/<sig>."""
    cfg = Config.wrap(dict(cfg))
    sig = signature(cfg)
    if root is None:
        run = cfg.get("run")
        if isinstance(run, dict) and run.get("dir") and not os.environ.get(DEFAULT_ROOT_ENV):
            root = Path(run["dir"])
        else:
            root = default_root()
    folder = Path(root) / "xps" / sig
    return XP(sig, folder, cfg)


class EntryPoint:
    """Callable main: config loading, overrides, XP bootstrap, user function."""

    def __init__(self, name: str, config_path: tp.Union[str, Path],
                 config_name: str = "config",
                 root: tp.Optional[Path] = None):
        self.name = name
        self.config_path = Path(config_path)
        self.config_name = config_name
        self._root = root
        self._fn: tp.Optional[tp.Callable[[Config], tp.Any]] = None

    # -- decorator ---------------------------------------------------------
    def bind(self, fn: tp.Callable[[Config], tp.Any]) -> tp.Callable[[Config], tp.Any]:
        self._fn = fn
        return fn

    # -- config / xp construction -----------------------------------------
    @property
    def dir(self) -> Path:
        return Path(self._root) if self._root else default_root()

    def load_cfg(self, overrides: tp.Sequence[str] = ()) -> Config:
        cfg = load_config(self.config_path / f"{self.config_name}.yaml")
        apply_overrides(cfg, overrides)
        return cfg

    def get_xp(self, overrides: tp.Sequence[str] = ()) -> XP:
        return create_xp(self.load_cfg(overrides), self._root)

    def get_xp_from_sig(self, sig: str) -> XP:
        """Re-open an existing run by signature (config read back from disk)."""
        folder = self.dir / "xps" / sig
        cfg_path = folder / "config.yaml"
        if not cfg_path.exists():
            raise FileNotFoundError(f"no run with signature {sig} under {self.dir}")
        cfg = load_config(cfg_path)
        return XP(sig, folder, cfg)

    # -- execution ---------------------------------------------------------
    def run(self, overrides: tp.Sequence[str] = (), clear: bool = False) -> tp.Any:
        if self._fn is None:
            raise RuntimeError("entry_point has no bound function; use @main.bind")
        xp = self.get_xp(overrides)
        if clear:
            import shutil
            from . import distrib
            if int(os.environ.get("WORLD_SIZE", "1")) > 1:
                # Multi-worker --clear: rendezvous so no rank can be inside
                # the XP (mkdir/history/restore) while rank 0 wipes it, and
                # no rank enters before the wipe is complete.
                distrib.init()
                distrib.barrier()
                if distrib.is_rank_zero():
                    shutil.rmtree(xp.folder, ignore_errors=True)
                distrib.barrier()
            else:
                shutil.rmtree(xp.folder, ignore_errors=True)
        xp.enter()
        # Persist the resolved config so get_xp_from_sig can rebuild the XP.
        # pid-suffixed temp: concurrent DDP workers of one run write the same
        # content and must not race each other's rename.
        with write_and_rename(xp.folder / "config.yaml", "w", pid=True) as fh:
            import yaml
            yaml.safe_dump(xp.cfg.to_plain(), fh)
        return self._fn(xp.cfg)

    def __call__(self, argv: tp.Optional[tp.Sequence[str]] = None) -> tp.Any:
        if argv is None:
            argv = sys.argv[1:]
        clear = False
        overrides = []
        for arg in argv:
            if arg == "--clear":
                clear = True
            elif arg.startswith("--"):
                raise ValueError(f"unknown flag {arg}")
            else:
                overrides.append(arg)
        return self.run(overrides, clear=clear)


def entry_point(name: str, config_path: tp.Union[str, Path],
                config_name: str = "config",
                root: tp.Optional[Path] = None) -> EntryPoint:
    return EntryPoint(name, config_path, config_name, root)
