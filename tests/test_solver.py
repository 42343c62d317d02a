# Copyright (c) Flashy-AMD authors.
"""Unit tests for BaseSolver (empty stub in the reference — SURVEY.md §4)."""
import pytest
import torch
from torch import nn

from flashy_amd import BaseSolver, Formatter
from flashy_amd import xp as fxp
from flashy_amd.config import Config


class TinySolver(BaseSolver):
    def __init__(self):
        super().__init__()
        torch.manual_seed(0)
        self.model = nn.Linear(4, 1)
        self.optim = torch.optim.Adam(self.model.parameters(), lr=1e-2)
        self.best = {}
        self.register_stateful("model", "optim", "best")

    def get_formatter(self, stage_name):
        return Formatter({"loss": ".4f"})

    def train_one(self):
        x = torch.randn(8, 4)
        loss = (self.model(x) ** 2).mean()
        self.optim.zero_grad()
        loss.backward()
        self.optim.step()
        return {"loss": loss.item()}

    def run(self, epochs=3):
        self.restore()
        for _ in range(self.epoch, epochs + 1):
            self.run_stage("train", self.train_one)
            self.best["loss"] = self.history[-1]["train"]["loss"] \
                if self.history else None
            self.commit()


@pytest.fixture()
def solver(xp_root):
    fxp.create_xp(Config.wrap({"lr": 0.01})).enter()
    return TinySolver()


def test_epoch_indexing_and_run_stage(solver):
    assert solver.epoch == 1
    metrics = solver.run_stage("train", solver.train_one)
    assert "loss" in metrics and "duration" in metrics
    assert metrics["duration"] >= 0
    solver.commit()
    assert solver.epoch == 2
    assert "train" in solver.history[0]


def test_stage_nesting_forbidden(solver):
    def nested():
        solver.run_stage("inner", solver.train_one)
    with pytest.raises(RuntimeError):
        solver.run_stage("outer", nested)


def test_double_log_same_stage_raises(solver):
    solver.log_metrics("train", {"loss": 1.0})
    with pytest.raises(RuntimeError):
        solver.log_metrics("train", {"loss": 2.0})


def test_none_metrics_become_empty(solver):
    out = solver.run_stage("valid", lambda: None)
    assert set(out) == {"duration"}


def test_checkpoint_contains_registered_names(solver):
    solver.run_stage("train", solver.train_one)
    solver.commit()
    state = torch.load(solver.checkpoint_path, weights_only=False)
    assert set(state) == {"history", "xp.cfg", "xp.sig", "model", "optim", "best"}
    assert state["xp.sig"] == solver.xp.sig
    assert state["xp.cfg"]["lr"] == 0.01


def test_commit_restore_roundtrip(xp_root):
    fxp.create_xp(Config.wrap({"lr": 0.01})).enter()
    s1 = TinySolver()
    s1.run(epochs=2)
    weights = s1.model.weight.detach().clone()
    hist = [dict(h) for h in s1.history]
    assert len(hist) == 2

    # new process simulation: fresh XP object + solver, same signature
    fxp._current_xp = None
    fxp.create_xp(Config.wrap({"lr": 0.01})).enter()
    s2 = TinySolver()
    assert not torch.equal(s2.model.weight, weights)  # fresh init != trained
    restored = s2.restore()
    assert restored
    assert torch.equal(s2.model.weight, weights)
    assert s2.epoch == 3
    # continue to epoch 3: history prefix preserved
    s2.run(epochs=3)
    assert len(s2.history) == 3
    assert s2.history[:2] == hist


def test_restore_without_checkpoint(solver):
    assert solver.restore() is False


def test_async_checkpoint(xp_root):
    """async_checkpoint=True defers the write but produces the identical
    artifact; finalize_checkpoint() makes it durable."""
    from flashy_amd import checkpoint as fckpt
    fxp.create_xp(Config.wrap({"lr": 0.01})).enter()
    s = TinySolver()
    s.async_checkpoint = True
    s.run(epochs=2)
    s.finalize_checkpoint()
    assert s.checkpoint_path.exists()
    state = fckpt.load_state(s.checkpoint_path)
    assert len(state["history"]) == 2

    # a fresh sync solver restores the async-written artifact identically
    fxp._current_xp = None
    fxp.create_xp(Config.wrap({"lr": 0.01})).enter()
    s2 = TinySolver()
    assert s2.restore()
    assert s2.epoch == 3
    assert torch.equal(s2.model.weight, s.model.weight)


def test_async_checkpoint_error_propagates(tmp_path):
    """A failed background write must raise at the next wait()/save()."""
    from flashy_amd.checkpoint import AsyncCheckpointer
    ck = AsyncCheckpointer()
    (tmp_path / "blocker").write_text("")   # a FILE where a dir is needed
    ck.save({"x": torch.ones(4)}, tmp_path / "blocker" / "ck.th")
    with pytest.raises(RuntimeError, match="async checkpoint"):
        ck.wait()
    # the checkpointer is reusable after the error
    ck.save({"x": torch.ones(4)}, tmp_path / "ck.th")
    ck.wait()
    assert (tmp_path / "ck.th").exists()


def test_async_checkpoint_snapshot_isolation(xp_root):
    """The async writer must snapshot CPU state at commit time: training
    that continues during the background write may not leak into it."""
    from flashy_amd import checkpoint as fckpt
    fxp.create_xp(Config.wrap({"lr": 0.01})).enter()
    s = TinySolver()
    s.async_checkpoint = True
    s.log_metrics("train", {"loss": 1.0})
    s.commit()
    snap = {k: v.detach().clone() for k, v in s.model.state_dict().items()}
    with torch.no_grad():           # mutate AFTER commit, before the join
        for p in s.model.parameters():
            p.add_(123.0)
    s.finalize_checkpoint()
    state = fckpt.load_state(s.checkpoint_path)
    for k, v in snap.items():
        assert torch.equal(state["model"][k], v), k
