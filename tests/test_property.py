# Copyright (c) Flashy-AMD authors.
"""Property-based tests (hypothesis) for the pure-Python core: averager
math, Formatter filter semantics, StateManager round-trips, config
override parsing.  These encode the DOCUMENTED semantics (reference
parity) as invariants over random inputs."""
import math
import string

from hypothesis import given, settings, strategies as st

from flashy_amd.config import Config, apply_overrides, flatten_config
from flashy_amd.formatter import Formatter
from flashy_amd.state import StateManager
from flashy_amd.utils import averager

keys = st.text(string.ascii_lowercase, min_size=1, max_size=6)
floats = st.floats(min_value=-1e6, max_value=1e6,
                   allow_nan=False, allow_infinity=False)


@settings(max_examples=60, deadline=None)
@given(st.lists(st.tuples(st.dictionaries(keys, floats, max_size=4),
                          st.floats(min_value=0.1, max_value=10.0)),
                min_size=1, max_size=12))
def test_averager_is_exact_weighted_mean(batches):
    avg = averager()  # beta=1 -> exact weighted mean
    out = {}
    for metrics, weight in batches:
        out = avg(metrics, weight)
    for key, got in out.items():
        num = sum(w * m[key] for m, w in batches if key in m)
        den = sum(w for m, w in batches if key in m)
        assert math.isclose(got, num / den, rel_tol=1e-9, abs_tol=1e-9)


@settings(max_examples=60, deadline=None)
@given(st.dictionaries(keys, floats, min_size=1, max_size=8),
       st.lists(keys, max_size=3), st.lists(keys, max_size=3))
def test_formatter_filter_semantics(metrics, include, exclude):
    fmt = Formatter(include_keys=include, exclude_keys=exclude)
    out = fmt(metrics)
    assert set(out) <= set(metrics)
    for k in metrics:
        inc = any(__import__("fnmatch").fnmatch(k, p) for p in include)
        exc = any(__import__("fnmatch").fnmatch(k, p) for p in exclude)
        if inc:
            assert k in out          # include always wins
        elif include and not exclude:
            assert k not in out      # pure whitelist mode
        elif exc:
            assert k not in out
        else:
            assert k in out
    for k, v in out.items():         # values formatted with the default spec
        assert v == format(metrics[k], ".3f")


@settings(max_examples=60, deadline=None)
@given(st.dictionaries(keys, floats, min_size=1, max_size=6))
def test_formatter_explicit_format_whitelists(metrics):
    key = sorted(metrics)[0]
    fmt = Formatter({key: ".1f"}, exclude_keys=["*"])
    out = fmt(metrics)
    assert set(out) == {key}         # include_formatted beats exclude '*'
    assert out[key] == format(metrics[key], ".1f")


@settings(max_examples=60, deadline=None)
@given(st.dictionaries(keys, st.one_of(floats, st.lists(floats, max_size=3),
                                       st.dictionaries(keys, floats, max_size=3)),
                       min_size=1, max_size=6))
def test_state_manager_round_trip(payload):
    class Holder:
        pass

    src, dst = Holder(), Holder()
    ms, md = StateManager(), StateManager()
    for name, value in payload.items():
        setattr(src, name, value)
        setattr(dst, name, type(value)() if isinstance(value, (list, dict))
                else 0.0)
        ms.register(name, _wrap(src, name))
        md.register(name, _wrap(dst, name))
    state = ms.state_dict()
    md.load_state_dict(state)
    for name, value in payload.items():
        assert getattr(dst, name) == value


def _wrap(owner, name):
    from flashy_amd.state import AttributeWrapper
    return AttributeWrapper(owner, name)


@settings(max_examples=60, deadline=None)
@given(st.dictionaries(
    st.lists(keys, min_size=1, max_size=3).map(".".join),
    st.one_of(st.integers(-999, 999), st.booleans(),
              st.floats(min_value=-99.0, max_value=99.0, allow_nan=False),
              st.text(string.ascii_lowercase, max_size=5))))
def test_apply_overrides_round_trip(items):
    cfg = Config()
    # deeper keys can overwrite shallower prefixes; apply in sorted order and
    # keep only keys that survive (no prefix of another key)
    survivors = {k: v for k, v in items.items()
                 if not any(other != k and other.startswith(k + ".")
                            for other in items)}
    overrides = [f"{k}={v}" for k, v in sorted(survivors.items())]
    apply_overrides(cfg, overrides)
    flat = flatten_config(cfg)
    import yaml
    for k, v in survivors.items():
        # the documented contract: values parse with YAML scalar rules
        # (so "no" -> False, "3" -> 3, "" -> None)
        expected = yaml.safe_load(str(v)) if str(v) else None
        assert flat[k] == expected, (k, v, flat[k], expected)
