# Copyright (c) Flashy-AMD authors.
"""State registry implementing the checkpoint state-dict protocol.

Capability parity with the reference's ``flashy/state.py`` (see
/root/reference/flashy/state.py:24-88): a runtime-checkable protocol for
stateful sources, attribute wrappers with late binding and in-place restore
semantics, a write-only wrapper, and the name->source registry used by the
solver to assemble the on-disk checkpoint dict.

Restore precedence (reference flashy/state.py:39-49):
  1. objects exposing ``state_dict``/``load_state_dict`` delegate (in place);
  2. lists restore by in-place slice assignment;
  3. dicts restore by clear+update in place;
  4. anything else is replaced via ``setattr`` on the owner.
"""
from __future__ import annotations

import typing as tp

# drop-in alias kept from the reference API (flashy/state.py)
StateDict = tp.Dict[str, tp.Any]


@tp.runtime_checkable
class StateDictSource(tp.Protocol):
    """Anything with the torch-style state-dict pair of methods."""

    def state_dict(self) -> tp.Any: ...

    def load_state_dict(self, state: tp.Any) -> tp.Any: ...


class AttributeWrapper:
    """StateDictSource over a (owner, attribute-name) pair.

    The attribute is resolved lazily at save/load time, so it may be replaced
    on the owner after registration and the checkpoint still captures the
    current object.
    """

    def __init__(self, owner: tp.Any, name: str):
        self.owner = owner
        self.name = name

    def _get(self) -> tp.Any:
        return getattr(self.owner, self.name)

    def state_dict(self) -> tp.Any:
        value = self._get()
        if isinstance(value, StateDictSource):
            return value.state_dict()
        return value

    def load_state_dict(self, state: tp.Any) -> None:
        value = self._get()
        if isinstance(value, StateDictSource):
            value.load_state_dict(state)
        elif isinstance(value, list):
            value[:] = state
        elif isinstance(value, dict):
            value.clear()
            value.update(state)
        else:
            setattr(self.owner, self.name, state)


class WriteOnlyWrapper(AttributeWrapper):
    """Saved into every checkpoint, ignored on load.

    Used for self-describing metadata (config, signature) that must never be
    restored over the live run's values.
    """

    def load_state_dict(self, state: tp.Any) -> None:  # noqa: ARG002
        return None


class StateManager:
    """Name -> StateDictSource registry.

    ``state_dict()`` returns a flat dict keyed by registered names — the
    on-disk checkpoint payload.  ``load_state_dict`` iterates the *incoming*
    keys: registered sources missing from the checkpoint are left untouched;
    unknown checkpoint keys raise ``KeyError``.
    """

    def __init__(self) -> None:
        self._sources: tp.Dict[str, StateDictSource] = {}

    def register(self, name: str, source: StateDictSource) -> None:
        if name in self._sources:
            raise ValueError(f"{name!r} already registered as a stateful source")
        self._sources[name] = source

    @property
    def names(self) -> tp.List[str]:
        return list(self._sources)

    def state_dict(self) -> tp.Dict[str, tp.Any]:
        return {name: source.state_dict() for name, source in self._sources.items()}

    def load_state_dict(self, state: tp.Mapping[str, tp.Any]) -> None:
        for name, sub_state in state.items():
            if name not in self._sources:
                raise KeyError(f"checkpoint contains unknown stateful source {name!r}")
            self._sources[name].load_state_dict(sub_state)
