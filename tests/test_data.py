# Copyright (c) Flashy-AMD authors.
"""DevicePrefetcher: CPU passthrough semantics + GPU data integrity."""
import pytest
import torch

from flashy_amd.data import DevicePrefetcher


def test_cpu_passthrough():
    batches = [(torch.full((4,), float(i)), torch.tensor([i])) for i in range(5)]
    out = list(DevicePrefetcher(iter(batches), "cpu"))
    assert len(out) == 5
    for (x, y), (rx, ry) in zip(out, batches):
        assert x is rx and y is ry  # no copies on CPU


def test_single_tensor_batches_cpu():
    out = list(DevicePrefetcher(iter([torch.ones(3)] * 2), "cpu"))
    assert len(out) == 2


@pytest.mark.gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs a GPU")
def test_gpu_prefetch_integrity():
    torch.manual_seed(0)
    n = 11
    batches = [(torch.randn(64, 32, pin_memory=True),
                torch.randint(10, (64,), pin_memory=True)) for _ in range(n)]
    acc = torch.zeros(64, 32, device="cuda")
    got = []
    for bx, by in DevicePrefetcher(iter(batches), "cuda", depth=3):
        assert bx.is_cuda and by.is_cuda
        acc += bx          # consume on the current stream (slot reuse fence)
        got.append(by.clone())
    torch.cuda.synchronize()
    assert len(got) == n
    ref = torch.stack([b[0] for b in batches]).sum(0).cuda()
    assert torch.allclose(acc, ref, atol=1e-4)
    for by, (_, ry) in zip(got, batches):
        assert torch.equal(by.cpu(), ry)


@pytest.mark.gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs a GPU")
def test_gpu_prefetch_slot_reuse_under_load():
    """Slot overwrite must wait for the consumer's enqueued reads: keep the
    main stream busy so laggard reads would expose a racy refill."""
    torch.manual_seed(1)
    n, depth = 9, 2
    batches = [(torch.full((1 << 20,), float(i), pin_memory=True),) for i in range(n)]
    sums = []
    spin = torch.randn(2048, 2048, device="cuda")
    for (bx,) in DevicePrefetcher(iter(batches), "cuda", depth=depth):  # 1-tuple in -> 1-tuple out
        for _ in range(4):
            spin = spin @ spin.T / 2048  # queue depth on the main stream
        sums.append(bx.sum())
    torch.cuda.synchronize()
    for i, s in enumerate(sums):
        assert s.item() == pytest.approx(float(i) * (1 << 20), rel=1e-6), i


def test_empty_iterator_cpu():
    assert list(DevicePrefetcher(iter([]), "cpu")) == []
