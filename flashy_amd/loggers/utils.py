# Copyright (c) Flashy-AMD authors.
"""Helpers shared by logger backends: prefix joining, dict flattening,
param sanitization (parity: reference flashy/loggers/utils.py)."""
from __future__ import annotations

import typing as tp

import torch


def _add_prefix(metrics: tp.Dict[str, tp.Any], prefix: str,
                separator: str = "/") -> tp.Dict[str, tp.Any]:
    if not prefix:
        return metrics
    return {f"{prefix}{separator}{k}": v for k, v in metrics.items()}


def _flatten_dict(params: tp.Mapping[str, tp.Any],
                  delimiter: str = "/") -> tp.Dict[str, tp.Any]:
    """Flatten nested mappings: ``{"a": {"b": 1}} -> {"a/b": 1}``."""
    out: tp.Dict[str, tp.Any] = {}

    def _walk(node: tp.Mapping[str, tp.Any], prefix: str) -> None:
        for key, value in node.items():
            name = f"{prefix}{delimiter}{key}" if prefix else str(key)
            if isinstance(value, tp.Mapping):
                _walk(value, name)
            else:
                out[name] = value

    _walk(params, "")
    return out


def _sanitize_params(params: tp.Dict[str, tp.Any]) -> tp.Dict[str, tp.Any]:
    """Coerce values to types the logging backends accept."""
    out: tp.Dict[str, tp.Any] = {}
    for key, value in params.items():
        if torch.is_tensor(value) and value.numel() == 1:
            out[key] = value.item()
        elif isinstance(value, (bool, int, float, str)):
            out[key] = value
        elif value is None:
            out[key] = "None"
        else:
            out[key] = str(value)
    return out
