# Copyright (c) Flashy-AMD authors.
"""Tests for the in-house config + XP runtime (the Dora/Hydra replacement)."""
import json

import yaml

from flashy_amd.config import Config, apply_overrides, load_config, signature
from flashy_amd import xp as fxp


def test_overrides_types(tmp_path):
    p = tmp_path / "c.yaml"
    p.write_text("a: 1\nsub:\n  b: 2.0\n  name: x\n")
    cfg = load_config(p)
    apply_overrides(cfg, ["a=3", "sub.b=4.5", "sub.name=hello", "new.flag=true", "n=null"])
    assert cfg.a == 3 and isinstance(cfg.a, int)
    assert cfg.sub.b == 4.5
    assert cfg.sub.name == "hello"
    assert cfg.new.flag is True
    assert cfg.n is None


def test_signature_stable_and_excludes():
    c1 = Config.wrap({"a": 1, "b": {"c": 2}})
    c2 = Config.wrap({"b": {"c": 2}, "a": 1})  # key order must not matter
    assert signature(c1) == signature(c2)
    c3 = Config.wrap({"a": 1, "b": {"c": 3}})
    assert signature(c1) != signature(c3)
    # run.* never enters the signature
    c4 = Config.wrap({"a": 1, "b": {"c": 2}, "run": {"dir": "/x"}})
    assert signature(c4) == signature(c1)
    # run.exclude patterns drop keys from the hash
    c5 = Config.wrap({"a": 1, "b": {"c": 2}, "nw": 8,
                      "run": {"exclude": ["nw"]}})
    assert signature(c5) == signature(c1)


def test_xp_enter_history(xp_root):
    cfg = Config.wrap({"a": 1})
    xp = fxp.create_xp(cfg)
    xp.enter()
    assert fxp.get_xp() is xp
    assert xp.folder.exists()
    xp.link.update_history([{"train": {"loss": 1.0}}])
    # re-open the same XP: history persisted
    xp2 = fxp.create_xp(cfg)
    assert xp2.sig == xp.sig
    xp2.enter()
    assert xp2.link.history == [{"train": {"loss": 1.0}}]
    data = json.loads((xp.folder / "history.json").read_text())
    assert data[0]["train"]["loss"] == 1.0


def test_entry_point_run_and_lookup(xp_root, tmp_path):
    conf = tmp_path / "conf"
    conf.mkdir()
    (conf / "config.yaml").write_text(yaml.safe_dump({"lr": 0.1, "epochs": 2}))
    main = fxp.entry_point("test_pkg", conf)
    seen = {}

    @main.bind
    def _run(cfg):
        seen["cfg"] = cfg
        seen["xp"] = fxp.get_xp()

    main(["lr=0.2"])
    assert seen["cfg"].lr == 0.2
    sig = seen["xp"].sig
    # programmatic lookup by overrides and by signature
    assert main.get_xp(["lr=0.2"]).sig == sig
    xp = main.get_xp_from_sig(sig)
    assert xp.sig == sig and xp.cfg.lr == 0.2
