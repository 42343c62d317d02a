# Copyright (c) Flashy-AMD authors.
import json

import torch

from flashy_amd import profiler


def test_trace_exports_chrome_trace(tmp_path):
    out = tmp_path / "trace.json"
    model = torch.nn.Linear(8, 8)
    with profiler.trace(out):
        model(torch.randn(4, 8)).sum().backward()
    assert out.exists()
    data = json.loads(out.read_text())
    assert "traceEvents" in data and len(data["traceEvents"]) > 0


def test_stage_timer():
    t = profiler.StageTimer(sync=False)
    with t("a"):
        sum(range(1000))
    with t("a"):
        sum(range(1000))
    with t("b"):
        pass
    s = t.summary()
    assert set(s) == {"a", "b"}
    assert s["a"] >= 0 and t.counts["a"] == 2
