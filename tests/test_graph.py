# Copyright (c) Flashy-AMD authors.
"""CapturedStep degrades to eager execution on CPU (tests/CI)."""
import pytest
import torch

from flashy_amd.graph import CapturedStep


@pytest.mark.skipif(torch.cuda.is_available(),
                    reason="CPU-fallback semantics; on a GPU box CapturedStep"
                           " would capture (an empty graph for a CPU step)")
def test_captured_step_cpu_fallback():
    model = torch.nn.Linear(4, 4)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    x = torch.randn(8, 4)

    def step():
        opt.zero_grad()
        loss = (model(x) ** 2).mean()
        loss.backward()
        opt.step()
        return loss

    runner = CapturedStep(step).capture()
    assert runner.graph is None  # no GPU: eager fallback
    l0 = runner().item()
    l1 = runner().item()
    assert l1 < l0
