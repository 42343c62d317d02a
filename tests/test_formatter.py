# Copyright (c) Flashy-AMD authors.
"""Unit tests for the metric formatter (empty stub in the reference)."""
from flashy_amd.formatter import Formatter


def test_default_format():
    f = Formatter()
    assert f({"loss": 1.23456}) == {"loss": "1.235"}


def test_pattern_first_match_wins():
    f = Formatter({"acc*": ".1%", "a*": ".5f"})
    assert f({"acc": 0.5})["acc"] == "50.0%"
    assert f({"alpha": 0.5})["alpha"] == "0.50000"


def test_exclude_and_include():
    f = Formatter(exclude_keys=["debug_*"])
    out = f({"loss": 1.0, "debug_x": 2.0})
    assert set(out) == {"loss"}

    # exclude-all except included
    f = Formatter(exclude_keys=["*"], include_keys=["loss"])
    out = f({"loss": 1.0, "noise": 2.0})
    assert set(out) == {"loss"}


def test_pure_whitelist():
    f = Formatter(include_keys=["a", "b"])
    assert set(f({"a": 1.0, "b": 2.0, "c": 3.0})) == {"a", "b"}


def test_formatted_keys_implicitly_whitelisted():
    f = Formatter({"acc": ".1%"}, exclude_keys=["*"])
    out = f({"acc": 0.25, "other": 1.0})
    assert set(out) == {"acc"}


def test_non_numeric_value():
    f = Formatter()
    assert f({"name": "hello"}) == {"name": "hello"}
