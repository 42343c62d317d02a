# Copyright (c) Flashy-AMD authors.
"""Multi-process tests of the comm layer over gloo on localhost.

Keeps the two reference oracles (reference
tests/test_distrib.py): exact-mean averaging, broadcast, the param-count
deadlock guard raising instead of hanging, and the **virtual-batch gradient
equivalence** proof that the bucketed post-hoc and eager (overlapped) DP sync
paths both produce exactly the gradients of one batch of size world_size.
"""
import os
import socket

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp
from torch import nn

from flashy_amd import distrib

WS = 8


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _worker(rank: int, ws: int, port: int, fn_name: str):
    os.environ.update(RANK=str(rank), LOCAL_RANK=str(rank), WORLD_SIZE=str(ws),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    distrib.init("gloo")
    try:
        globals()[fn_name](rank, ws)
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def _spawn(fn_name: str, ws: int = WS):
    port = _free_port()
    mp.spawn(_worker, args=(ws, port, fn_name), nprocs=ws, join=True)


# --------------------------------------------------------------------------
# check bodies (run inside workers)
# --------------------------------------------------------------------------

def _check_average_tensors(rank: int, ws: int):
    ts = [torch.full((5,), float(rank)), torch.full((3, 3), float(rank * 2))]
    distrib.average_tensors(ts)
    mean = (ws - 1) / 2
    assert torch.allclose(ts[0], torch.full((5,), mean)), ts[0]
    assert torch.allclose(ts[1], torch.full((3, 3), mean * 2)), ts[1]


def _check_many_buckets(rank: int, ws: int):
    # force several buckets: tiny bucket size, mixed dtypes
    ts = [torch.full((64,), float(rank)),
          torch.full((128,), float(rank), dtype=torch.float64),
          torch.full((32,), float(rank))]
    distrib.average_tensors(ts, bucket_bytes=256)
    mean = (ws - 1) / 2
    for t in ts:
        assert torch.allclose(t, torch.full_like(t, mean))


def _check_broadcast(rank: int, ws: int):
    ts = [torch.full((4,), float(rank)), torch.full((2, 2), float(rank + 10))]
    distrib.broadcast_tensors(ts, src=0)
    assert torch.equal(ts[0], torch.zeros(4))
    assert torch.equal(ts[1], torch.full((2, 2), 10.0))


def _check_param_count_guard(rank: int, ws: int):
    n = 2 if rank == 5 else 3
    ts = [torch.zeros(2) for _ in range(n)]
    with pytest.raises(RuntimeError):
        distrib._check_number_of_params(ts)


def _make_model(seed: int = 1234) -> nn.Module:
    g = torch.Generator().manual_seed(seed)
    model = nn.Sequential(nn.Linear(8, 16), nn.ReLU(), nn.Linear(16, 1))
    with torch.no_grad():
        for p in model.parameters():
            p.copy_(torch.randn(p.shape, generator=g))
    return model


def _virtual_batch(ws: int):
    g = torch.Generator().manual_seed(4321)
    x = torch.randn(ws * 4, 8, generator=g)
    y = torch.randn(ws * 4, 1, generator=g)
    return x, y


def _reference_grads(ws: int):
    model = _make_model()
    x, y = _virtual_batch(ws)
    loss = torch.nn.functional.mse_loss(model(x), y)
    loss.backward()
    return [p.grad.clone() for p in model.parameters()]


def _check_sync_model_equivalence(rank: int, ws: int):
    model = _make_model()
    x, y = _virtual_batch(ws)
    xs, ys = x[rank * 4:(rank + 1) * 4], y[rank * 4:(rank + 1) * 4]
    loss = torch.nn.functional.mse_loss(model(xs), ys)
    loss.backward()
    distrib.sync_model(model)
    for got, ref in zip([p.grad for p in model.parameters()], _reference_grads(ws)):
        assert torch.allclose(got, ref, atol=1e-6), (got - ref).abs().max()


def _check_eager_sync_equivalence(rank: int, ws: int):
    model = _make_model()
    x, y = _virtual_batch(ws)
    xs, ys = x[rank * 4:(rank + 1) * 4], y[rank * 4:(rank + 1) * 4]
    with distrib.eager_sync_model(model):
        loss = torch.nn.functional.mse_loss(model(xs), ys)
        loss.backward()
    for got, ref in zip([p.grad for p in model.parameters()], _reference_grads(ws)):
        assert torch.allclose(got, ref, atol=1e-6), (got - ref).abs().max()


def _check_eager_small_buckets(rank: int, ws: int):
    # bucket size smaller than one param -> every grad flushes its own bucket
    model = _make_model()
    x, y = _virtual_batch(ws)
    xs, ys = x[rank * 4:(rank + 1) * 4], y[rank * 4:(rank + 1) * 4]
    with distrib.eager_sync_gradients(model.parameters(), bucket_bytes=8):
        torch.nn.functional.mse_loss(model(xs), ys).backward()
    for got, ref in zip([p.grad for p in model.parameters()], _reference_grads(ws)):
        assert torch.allclose(got, ref, atol=1e-6)


def _check_flat_optimizer_equivalence(rank: int, ws: int):
    """FusedSGD + sync_flat_gradients matches torch SGD on the virtual batch
    (the flat-buffer DP path the bench and solvers use)."""
    from flashy_amd.optim import FusedSGD
    model = _make_model()
    opt = FusedSGD(model.parameters(), lr=0.05, momentum=0.9)
    x, y = _virtual_batch(ws)
    xs, ys = x[rank * 4:(rank + 1) * 4], y[rank * 4:(rank + 1) * 4]
    for _ in range(3):
        loss = torch.nn.functional.mse_loss(model(xs), ys)
        opt.zero_grad()
        loss.backward()
        distrib.sync_flat_gradients(opt)
        opt.step()
    ref = _make_model()
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.05, momentum=0.9)
    for _ in range(3):
        loss = torch.nn.functional.mse_loss(ref(x), y)
        ref_opt.zero_grad()
        loss.backward()
        ref_opt.step()
    for p, q in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p, q, atol=1e-5), (p - q).abs().max()


def _check_overlapped_flat_sync(rank: int, ws: int):
    """OverlappedFlatSync (chunked, backward-overlapped flat all-reduce)
    matches torch SGD on the virtual batch — with chunk_bytes small enough
    that every chunk flushes from inside backward."""
    from flashy_amd.optim import FusedSGD
    model = _make_model()
    opt = FusedSGD(model.parameters(), lr=0.05, momentum=0.9)
    sync = distrib.OverlappedFlatSync(opt, chunk_bytes=64)
    assert sync.n_chunks >= 3  # the point is multiple in-backward flushes
    x, y = _virtual_batch(ws)
    xs, ys = x[rank * 4:(rank + 1) * 4], y[rank * 4:(rank + 1) * 4]
    for _ in range(3):
        opt.zero_grad()
        torch.nn.functional.mse_loss(model(xs), ys).backward()
        sync.finish()
        opt.step()
    ref = _make_model()
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.05, momentum=0.9)
    for _ in range(3):
        ref_opt.zero_grad()
        torch.nn.functional.mse_loss(ref(x), y).backward()
        ref_opt.step()
    for p, q in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p, q, atol=1e-5), (p - q).abs().max()
    sync.remove()


def _check_overlapped_sync_unused_param(rank: int, ws: int):
    """A param that never gets a grad leaves its chunk incomplete; finish()
    must still reduce it (straggler flush) so ranks stay aligned."""
    from flashy_amd.optim import FusedSGD
    model = _make_model()
    model.extra = nn.Parameter(torch.ones(4))  # never used in forward
    opt = FusedSGD(model.parameters(), lr=0.05)
    sync = distrib.OverlappedFlatSync(opt)  # default chunking: 1 big chunk
    x, y = _virtual_batch(ws)
    xs, ys = x[rank * 4:(rank + 1) * 4], y[rank * 4:(rank + 1) * 4]
    opt.zero_grad()
    torch.nn.functional.mse_loss(model(xs), ys).backward()
    sync.finish()
    # grads averaged, unused param's grad stays exactly zero
    assert torch.equal(model.extra.grad, torch.zeros(4))
    grads = [p.grad for n, p in model.named_parameters() if n != "extra"]
    for got, ref in zip(grads, _reference_grads(ws)):
        assert torch.allclose(got, ref, atol=1e-6)
    sync.remove()


def _check_eager_sync_error_drain(rank: int, ws: int):
    """An exception mid-backward inside the eager context must drain the
    in-flight async all-reduces: the next collective on the communicator
    still completes correctly on every rank (no interleave / deadlock)."""
    model = _make_model()
    x, y = _virtual_batch(ws)
    xs, ys = x[rank * 4:(rank + 1) * 4], y[rank * 4:(rank + 1) * 4]
    hidden = torch.relu(model[0](xs))

    def _boom(_grad):
        raise RuntimeError("boom")

    hidden.register_hook(_boom)  # fires AFTER layer-2 grads flushed
    raised = False
    try:
        # bucket_bytes=8: every early grad launches its own async all-reduce
        with distrib.eager_sync_gradients(model.parameters(), bucket_bytes=8):
            torch.nn.functional.mse_loss(model[2](hidden), ys).backward()
    except RuntimeError:
        raised = True
    assert raised
    t = torch.full((4,), float(rank))
    distrib.average_tensors([t])
    assert torch.allclose(t, torch.full((4,), (ws - 1) / 2)), t


def _check_broadcast_object(rank: int, ws: int):
    import collections
    if rank == 0:
        obj = collections.defaultdict(int, {"a": 1, "b": [1, 2]})
    else:
        obj = None
    out = distrib.broadcast_object(obj, src=0)
    assert isinstance(out, collections.defaultdict)
    assert out["a"] == 1 and out["b"] == [1, 2]


def _check_average_metrics(rank: int, ws: int):
    out = distrib.average_metrics({"loss": float(rank)}, count=1)
    assert out["loss"] == pytest.approx((ws - 1) / 2)
    # weighted: rank 0 has weight 3, others 1 -> weighted mean
    out = distrib.average_metrics({"m": 1.0 if rank == 0 else 0.0},
                                  count=3 if rank == 0 else 1)
    assert out["m"] == pytest.approx(3 / (3 + (ws - 1)))


def _check_broadcast_model_and_barrier(rank: int, ws: int):
    model = _make_model(seed=rank)  # every rank different
    distrib.broadcast_model(model)
    ref = _make_model(seed=0)
    for p, q in zip(model.parameters(), ref.parameters()):
        assert torch.equal(p, q)
    distrib.barrier()


def _check_loader_shards(rank: int, ws: int):
    data = list(range(23))
    ld = distrib.loader(data, batch_size=1, shuffle=False)
    seen = [int(b[0]) for b in ld]
    assert seen == list(range(rank, 23, ws))
    # train path: DistributedSampler covers everything across ranks each epoch
    ld = distrib.loader(data, batch_size=1, shuffle=True)
    epoch1 = [int(b[0]) for b in ld]
    epoch2 = [int(b[0]) for b in ld]
    assert epoch1 != epoch2 or ws == 1  # set_epoch -> different order
    gathered = [None] * ws
    dist.all_gather_object(gathered, epoch1)
    flat = sorted(x for sub in gathered for x in sub)
    assert set(flat) == set(range(23))  # full coverage (with padding dup)


# --------------------------------------------------------------------------
# pytest entry points
# --------------------------------------------------------------------------

ALL_CHECKS = [
    "_check_average_tensors",
    "_check_many_buckets",
    "_check_broadcast",
    "_check_param_count_guard",
    "_check_sync_model_equivalence",
    "_check_eager_sync_equivalence",
    "_check_eager_small_buckets",
    "_check_flat_optimizer_equivalence",
    "_check_overlapped_flat_sync",
    "_check_overlapped_sync_unused_param",
    "_check_eager_sync_error_drain",
    "_check_broadcast_object",
    "_check_average_metrics",
    "_check_broadcast_model_and_barrier",
    "_check_loader_shards",
]


def _check_all(rank: int, ws: int):
    # one spawn, all checks in sequence: keeps wall time (and process
    # startups) low while every collective path is exercised at ws=8
    for name in ALL_CHECKS:
        globals()[name](rank, ws)


def test_distributed_world8():
    _spawn("_check_all")


def test_single_process_noop():
    # world_size == 1: every collective is free and exact
    t = torch.ones(3)
    distrib.average_tensors([t])
    assert torch.equal(t, torch.ones(3))
    assert distrib.average_metrics({"a": 2.0}) == {"a": 2.0}
    assert distrib.broadcast_object({"x": 1}) == {"x": 1}
    assert distrib.is_rank_zero()
    model = nn.Linear(2, 2)
    loss = model(torch.randn(3, 2)).sum()
    with distrib.eager_sync_model(model):
        loss.backward()
    distrib.sync_model(model)
    distrib.barrier()
