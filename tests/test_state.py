# Copyright (c) Flashy-AMD authors.
"""Unit tests for the state registry (the reference shipped only an empty
stub here — SURVEY.md §4)."""
import pytest
import torch
from torch import nn

from flashy_amd.state import AttributeWrapper, StateManager, WriteOnlyWrapper


class Holder:
    pass


def test_module_delegates_in_place():
    h = Holder()
    h.model = nn.Linear(4, 2)
    wrapper = AttributeWrapper(h, "model")
    saved = wrapper.state_dict()
    orig = h.model
    with torch.no_grad():
        h.model.weight.zero_()
    wrapper.load_state_dict(saved)
    assert h.model is orig  # restored in place, not replaced
    assert torch.equal(h.model.weight, saved["weight"])


def test_list_and_dict_restore_in_place():
    h = Holder()
    h.lst = [1, 2]
    h.dct = {"a": 1}
    lst_ref, dct_ref = h.lst, h.dct
    AttributeWrapper(h, "lst").load_state_dict([3, 4, 5])
    AttributeWrapper(h, "dct").load_state_dict({"b": 2})
    assert h.lst is lst_ref and h.lst == [3, 4, 5]
    assert h.dct is dct_ref and h.dct == {"b": 2}


def test_plain_value_replaced():
    h = Holder()
    h.x = 1
    AttributeWrapper(h, "x").load_state_dict(42)
    assert h.x == 42


def test_late_binding():
    h = Holder()
    h.model = nn.Linear(2, 2)
    wrapper = AttributeWrapper(h, "model")
    h.model = nn.Linear(3, 3)  # replaced after registration
    assert wrapper.state_dict()["weight"].shape == (3, 3)


def test_write_only():
    h = Holder()
    h.cfg = {"lr": 0.1}
    w = WriteOnlyWrapper(h, "cfg")
    assert w.state_dict() == {"lr": 0.1}
    w.load_state_dict({"lr": 999})
    assert h.cfg == {"lr": 0.1}  # load ignored


def test_manager_roundtrip_and_errors():
    h = Holder()
    h.model = nn.Linear(4, 2)
    h.meta = [0]
    mgr = StateManager()
    mgr.register("model", AttributeWrapper(h, "model"))
    mgr.register("meta", AttributeWrapper(h, "meta"))
    with pytest.raises(ValueError):
        mgr.register("model", AttributeWrapper(h, "model"))
    import copy
    # state_dict returns live references (serialization snapshots them at
    # save time); deepcopy here to emulate a written checkpoint
    state = copy.deepcopy(mgr.state_dict())
    assert set(state) == {"model", "meta"}

    with torch.no_grad():
        h.model.weight.add_(1)
    h.meta[:] = [7]
    mgr.load_state_dict(state)
    assert torch.equal(h.model.weight, state["model"]["weight"])
    assert h.meta == [0]

    # extra registered sources untouched when missing from checkpoint
    mgr.load_state_dict({"meta": [5]})
    assert h.meta == [5]

    with pytest.raises(KeyError):
        mgr.load_state_dict({"unknown": 1})


def test_optimizer_state_roundtrip():
    h = Holder()
    model = nn.Linear(4, 1)
    h.optim = torch.optim.Adam(model.parameters(), lr=1e-3)
    loss = model(torch.randn(2, 4)).sum()
    loss.backward()
    h.optim.step()
    wrapper = AttributeWrapper(h, "optim")
    saved = wrapper.state_dict()
    h.optim = torch.optim.Adam(model.parameters(), lr=1e-3)
    AttributeWrapper(h, "optim").load_state_dict(saved)
    assert h.optim.state_dict()["state"].keys() == saved["state"].keys()
