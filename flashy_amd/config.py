# Copyright (c) Flashy-AMD authors.
"""YAML config loading with dotted CLI overrides and a stable signature hash.

In-house replacement for the Hydra + Dora config surface the reference
consumes (/root/reference/flashy/solver.py:16, examples/*/train.py — see
SURVEY.md §2.9 and §5.6): a per-project ``conf/config.yaml``, overrides of the
form ``a.b=value`` on the command line, and a content hash of the config
(minus excluded keys) used as the experiment signature / run id.

The ``run`` section is reserved for experiment-runtime settings and never
enters the signature:

    run:
      dir: /tmp/flashy_amd     # artifact root (env _FLASHY_AMD_DIR overrides)
      exclude: ["num_workers"] # extra fnmatch patterns excluded from the sig
"""
from __future__ import annotations

import fnmatch
import hashlib
import json
import typing as tp
from pathlib import Path

import yaml


class Config(dict):
    """A nested dict with attribute access.  Picklable, yaml/json friendly."""

    def __getattr__(self, name: str) -> tp.Any:
        try:
            return self[name]
        except KeyError as exc:
            raise AttributeError(name) from exc

    def __setattr__(self, name: str, value: tp.Any) -> None:
        self[name] = value

    def __delattr__(self, name: str) -> None:
        try:
            del self[name]
        except KeyError as exc:
            raise AttributeError(name) from exc

    @staticmethod
    def wrap(obj: tp.Any) -> tp.Any:
        if isinstance(obj, dict):
            return Config({k: Config.wrap(v) for k, v in obj.items()})
        if isinstance(obj, (list, tuple)):
            return type(obj)(Config.wrap(v) for v in obj)
        return obj

    def to_plain(self) -> dict:
        def _plain(obj: tp.Any) -> tp.Any:
            if isinstance(obj, dict):
                return {k: _plain(v) for k, v in obj.items()}
            if isinstance(obj, (list, tuple)):
                return [_plain(v) for v in obj]
            return obj
        return _plain(self)


def load_config(path: tp.Union[str, Path]) -> Config:
    """Load a YAML file into a :class:`Config`. Missing file -> empty config."""
    path = Path(path)
    if not path.exists():
        return Config()
    with open(path) as fh:
        data = yaml.safe_load(fh) or {}
    if not isinstance(data, dict):
        raise ValueError(f"top level of {path} must be a mapping")
    return Config.wrap(data)


def _parse_value(text: str) -> tp.Any:
    """Parse an override value with YAML scalar rules (1 -> int, true -> bool...)."""
    try:
        return yaml.safe_load(text)
    except yaml.YAMLError:
        return text


def apply_overrides(cfg: Config, overrides: tp.Sequence[str]) -> Config:
    """Apply ``key.sub=value`` strings onto ``cfg`` (in place), creating
    intermediate sections as needed.  Returns ``cfg``."""
    for item in overrides:
        if "=" not in item:
            raise ValueError(f"override {item!r} is not of the form key=value")
        dotted, raw = item.split("=", 1)
        node: tp.Any = cfg
        parts = dotted.strip().split(".")
        for part in parts[:-1]:
            if part not in node or not isinstance(node[part], dict):
                node[part] = Config()
            node = node[part]
        node[parts[-1]] = _parse_value(raw)
    return cfg


def flatten_config(cfg: tp.Mapping[str, tp.Any]) -> tp.Dict[str, tp.Any]:
    """Flatten nested config to ``{"a.b": value}`` with sorted keys."""
    out: tp.Dict[str, tp.Any] = {}

    def _walk(node: tp.Any, prefix: str) -> None:
        if isinstance(node, dict):
            for key in sorted(node, key=str):
                _walk(node[key], f"{prefix}{key}." )
        else:
            out[prefix[:-1]] = node

    _walk(dict(cfg), "")
    return out


def signature(cfg: tp.Mapping[str, tp.Any], extra_exclude: tp.Sequence[str] = ()) -> str:
    """Stable 8-hex-char hash of the config content.

    The ``run`` section is always excluded, plus any fnmatch pattern listed in
    ``cfg["run"]["exclude"]`` or ``extra_exclude`` (matched against dotted
    keys).  Equal configs (after exclusion) -> equal signatures across
    processes and runs.
    """
    flat = flatten_config(cfg)
    patterns = list(extra_exclude)
    run = cfg.get("run") if hasattr(cfg, "get") else None
    if isinstance(run, dict):
        patterns += list(run.get("exclude") or [])
    items = []
    for key, value in flat.items():
        if key == "run" or key.startswith("run."):
            continue
        if any(fnmatch.fnmatch(key, pat) for pat in patterns):
            continue
        items.append((key, value))
    payload = json.dumps(items, sort_keys=True, default=repr).encode()
    return hashlib.sha1(payload).hexdigest()[:8]
