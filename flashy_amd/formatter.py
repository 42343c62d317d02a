# Copyright (c) Flashy-AMD authors.
"""Metric formatting: fnmatch pattern -> format spec, include/exclude filters.

Capability parity with the reference's ``flashy/formatter.py:14-86``:
``formats`` maps fnmatch patterns to format specs (first match wins, default
``.3f``); filtering semantics: keys matching ``exclude_keys`` are dropped
unless they match ``include_keys``; if only ``include_keys`` is given the
formatter acts as a whitelist; keys with an explicit format entry are
implicitly whitelisted when ``include_formatted`` (default True).
``__call__`` returns a dict of formatted *strings* in input order.
"""
from __future__ import annotations

import fnmatch
import typing as tp


class Formatter:
    def __init__(self,
                 formats: tp.Optional[tp.Mapping[str, str]] = None,
                 exclude_keys: tp.Sequence[str] = (),
                 include_keys: tp.Sequence[str] = (),
                 default_format: str = ".3f",
                 include_formatted: bool = True):
        self.formats = dict(formats or {})
        self.exclude_keys = list(exclude_keys)
        self.include_keys = list(include_keys)
        self.default_format = default_format
        self.include_formatted = include_formatted

    def get_format(self, key: str) -> str:
        for pattern, spec in self.formats.items():
            if fnmatch.fnmatch(key, pattern):
                return spec
        return self.default_format

    def _is_included(self, key: str) -> bool:
        included = any(fnmatch.fnmatch(key, pat) for pat in self.include_keys)
        if self.include_formatted and any(
                fnmatch.fnmatch(key, pat) for pat in self.formats):
            included = True
        if included:
            return True
        if self.include_keys and not self.exclude_keys:
            # pure whitelist mode
            return False
        if any(fnmatch.fnmatch(key, pat) for pat in self.exclude_keys):
            return False
        return True

    def get_relevant_metrics(self, metrics: tp.Mapping[str, tp.Any]) -> tp.Dict[str, tp.Any]:
        return {k: v for k, v in metrics.items() if self._is_included(k)}

    def format_value(self, key: str, value: tp.Any) -> str:
        spec = self.get_format(key)
        try:
            return format(value, spec)
        except (TypeError, ValueError):
            return str(value)

    def __call__(self, metrics: tp.Mapping[str, tp.Any]) -> tp.Dict[str, str]:
        relevant = self.get_relevant_metrics(metrics)
        return {k: self.format_value(k, v) for k, v in relevant.items()}
