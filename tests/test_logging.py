# Copyright (c) Flashy-AMD authors.
"""Tests for the logging stack: setup_logging, LogProgressBar, ResultLogger."""
import logging

import pytest

from flashy_amd import Formatter, LogProgressBar, ResultLogger, bold, setup_logging
from flashy_amd import xp as fxp
from flashy_amd.config import Config


def test_bold_wraps_ansi():
    assert bold("hi") == "\033[1mhi\033[0m"


def test_setup_logging_per_rank_file(tmp_path):
    setup_logging(folder=tmp_path)
    logging.getLogger("t").info("hello world")
    for h in logging.getLogger().handlers:
        h.flush()
    log = tmp_path / "solver.log.0"
    assert log.exists()
    assert "hello world" in log.read_text()
    # cleanup handlers so later tests don't write here
    setup_logging(with_file_log=False)


def test_log_progress_bar_updates(caplog):
    logger = logging.getLogger("lp_test")
    lp = LogProgressBar(logger, range(10), updates=5, name="Train")
    with caplog.at_level(logging.INFO, logger="lp_test"):
        for i in lp:
            lp.update(loss=float(i))
    msgs = [r.message for r in caplog.records]
    assert msgs, "no progress lines emitted"
    assert all(m.startswith("Train | ") for m in msgs)
    # the line logged at iteration i reports metrics from update() at i-1
    assert any("loss" in m for m in msgs)


def test_log_progress_no_len(caplog):
    logger = logging.getLogger("lp_test2")
    lp = LogProgressBar(logger, iter(range(7)), updates=3, name="X")
    with caplog.at_level(logging.INFO, logger="lp_test2"):
        total = sum(1 for _ in lp)
    assert total == 7


def test_result_logger_summary_and_history(xp_root, caplog):
    fxp.create_xp(Config.wrap({"a": 1})).enter()
    rl = ResultLogger()
    assert "local" in rl.backends
    with caplog.at_level(logging.INFO):
        rl.log_metrics("train", {"loss": 0.5, "acc": 0.9}, step=3,
                       formatter=Formatter({"acc": ".0%"}))
    joined = " ".join(r.message for r in caplog.records)
    assert "Train Summary" in joined and "Epoch 3" in joined
    assert "90%" in joined


def test_local_fs_media(xp_root):
    import torch
    fxp.create_xp(Config.wrap({"b": 2})).enter()
    rl = ResultLogger()
    rl.log_hyperparams({"lr": 0.1})
    rl.log_text("train", "note", "hello", step=1)
    rl.log_image("train", "img", torch.rand(3, 4, 4), step=1)
    out = fxp.get_xp().folder / "outputs"
    assert (out / "hyperparams.json").exists()
    assert (out / "train_1_note.txt").read_text() == "hello"
    # torchvision missing -> tensor dump fallback
    assert (out / "train_1_img.pt").exists() or (out / "train_1_img.png").exists()


def test_tensorboard_soft_dep(xp_root):
    fxp.create_xp(Config.wrap({"c": 3})).enter()
    rl = ResultLogger()
    try:
        import tensorboard  # noqa: F401
        has_tb = True
    except ImportError:
        has_tb = False
    if not has_tb:
        rl.init_tensorboard()  # lazily constructed: writer only on first use
        with pytest.raises(RuntimeError):
            rl.backends["tensorboard"].log_metrics("train", {"x": 1.0}, 1)
