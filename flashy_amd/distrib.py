# Copyright (c) Flashy-AMD authors.
"""Distributed data-parallel utilities — the DDP alternative, MI355X-native.

Capability parity with the reference's ``flashy/distrib.py`` (rank helpers,
metric averaging, post-hoc and eager gradient sync, model/object broadcast,
loader sharding, barrier — /root/reference/flashy/distrib.py:21-276), built
for one-process-per-GPU over RCCL/xGMI instead of per-tensor collectives:

* Gradient sync is **bucketed**: grads are packed into flat buffers of
  ``bucket_bytes`` per (device, dtype) and reduced with one RCCL all-reduce
  per bucket.  On an 8-GPU xGMI mesh the ring all-reduce is bound by one
  point-to-point link (~153 GB/s), so buckets default large (32 MiB) to keep
  the links saturated while still overlapping with backward.
* The eager path launches bucket all-reduces **on a dedicated side HIP
  stream** as gradients become ready inside ``backward()`` (hooks ->
  event-ordered bucket flushes), so communication overlaps the remainder of
  the backward pass.  The reference's per-param hook scheme
  (flashy/distrib.py:153-191) maps onto this bucketed design.
* Numerics are exact: all-reduce SUM then divide by world size per element —
  verified by the virtual-batch oracle in ``tests/test_distrib.py`` (same
  oracle as reference tests/test_distrib.py:48-68).
* Collectives run through ``torch.distributed`` whose ``nccl`` backend *is*
  RCCL on ROCm; ``gloo`` is the CPU/CI fallback (reference §2.8 table C1-C9).

Everything is a no-op at world_size == 1.
"""
from __future__ import annotations

import functools
import io
import logging
import os
import typing as tp
from contextlib import contextmanager

import torch
import torch.distributed as dist
from torch.utils.data import DataLoader, Subset
from torch.utils.data.distributed import DistributedSampler

logger = logging.getLogger(__name__)

# Default bucket size for gradient sync.  Sized for the 7-link xGMI mesh:
# large enough that a ring all-reduce amortizes per-message latency on a
# single ~153 GB/s link, small enough that several buckets overlap backward.
DEFAULT_BUCKET_BYTES = 32 * 1024 * 1024


# ---------------------------------------------------------------------------
# Process-group bootstrap + rank helpers
# ---------------------------------------------------------------------------

def rank() -> int:
    if dist.is_initialized():
        return dist.get_rank()
    return int(os.environ.get("RANK", 0))


def world_size() -> int:
    if dist.is_initialized():
        return dist.get_world_size()
    return int(os.environ.get("WORLD_SIZE", 1))


def local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", rank()))


def is_distributed() -> bool:
    return world_size() > 1


def is_rank_zero() -> bool:
    return rank() == 0


def rank_zero_only(fn: tp.Callable) -> tp.Callable:
    """Decorator: run only on rank 0, return None elsewhere."""

    @functools.wraps(fn)
    def _wrapped(*args, **kwargs):
        if is_rank_zero():
            return fn(*args, **kwargs)
        return None

    return _wrapped


def init(backend: tp.Optional[str] = None) -> None:
    """Initialize the process group from torchrun-style env variables.

    ``backend=None`` picks ``nccl`` (= RCCL over xGMI on ROCm) when a GPU is
    available, else ``gloo``.  Single-process runs (WORLD_SIZE unset or 1)
    skip initialization entirely, so every collective below is free.
    Also pins this process to ``cuda:LOCAL_RANK``.
    """
    if dist.is_initialized():
        return
    ws = int(os.environ.get("WORLD_SIZE", 1))
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank() % max(1, torch.cuda.device_count()))
    if ws <= 1:
        return
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    # explicit watchdog: a hung collective (rank divergence, a capture gone
    # wrong) must abort the job quickly, never wedge the node
    import datetime
    timeout = datetime.timedelta(
        seconds=int(os.environ.get("FLASHY_AMD_PG_TIMEOUT", "180")))
    dist.init_process_group(backend=backend, init_method="env://",
                            timeout=timeout)


def device() -> torch.device:
    """Device for ad-hoc collective tensors."""
    if torch.cuda.is_available():
        return torch.device("cuda", torch.cuda.current_device())
    return torch.device("cpu")


def barrier() -> None:
    if is_distributed():
        dist.barrier()


# ---------------------------------------------------------------------------
# Small-message collectives
# ---------------------------------------------------------------------------

def all_reduce(tensor: torch.Tensor, op: "dist.ReduceOp" = None) -> torch.Tensor:
    """In-place sum all-reduce (generic helper; reference distrib.py:45-47)."""
    if is_distributed():
        dist.all_reduce(tensor, op=op or dist.ReduceOp.SUM)
    return tensor


def average_metrics(metrics: tp.Mapping[str, float],
                    count: float = 1.0) -> tp.Dict[str, float]:
    """Weighted average of scalar metrics across ranks.

    One float64 all-reduce of length n+1 (values*count .. count), then divide
    by the reduced weight (reference distrib.py:50-62 used float32; we use
    float64 so metric averaging never loses precision).
    """
    if not is_distributed():
        return dict(metrics)
    keys = list(metrics.keys())
    packed = torch.tensor([float(metrics[k]) * count for k in keys] + [count],
                          device=device(), dtype=torch.float64)
    dist.all_reduce(packed, op=dist.ReduceOp.SUM)
    total = packed[-1].item()
    return {k: (packed[i] / packed[-1]).item() if total != 0 else 0.0
            for i, k in enumerate(keys)}


def broadcast_object(obj: tp.Any = None, src: int = 0) -> tp.Any:
    """Broadcast an arbitrary picklable object from ``src`` to all ranks.

    Two broadcasts: int64 size then uint8 payload (reference
    distrib.py:246-269 — with its ``rank != src`` function-vs-int bug fixed:
    the src rank returns its own object without re-deserializing).
    """
    if not is_distributed():
        return obj
    if rank() == src:
        buf = io.BytesIO()
        torch.save(obj, buf)
        payload = torch.frombuffer(bytearray(buf.getvalue()), dtype=torch.uint8).to(device())
        size = torch.tensor([payload.numel()], device=device(), dtype=torch.long)
    else:
        size = torch.zeros(1, device=device(), dtype=torch.long)
    dist.broadcast(size, src=src)
    if rank() == src:
        dist.broadcast(payload, src=src)
        return obj
    payload = torch.empty(int(size.item()), device=device(), dtype=torch.uint8)
    dist.broadcast(payload, src=src)
    buf = io.BytesIO(payload.cpu().numpy().tobytes())
    return torch.load(buf, weights_only=False)


def _check_number_of_params(params: tp.List[torch.Tensor]) -> None:
    """Collective-deadlock guard: cheap all-reduce of the tensor count;
    raise instead of hanging when ranks disagree (reference distrib.py:78-89)."""
    if not is_distributed():
        return
    n = torch.tensor([len(params)], device=device(), dtype=torch.long)
    dist.all_reduce(n)
    if n.item() != len(params) * world_size():
        raise RuntimeError(
            f"Number of tensors to sync differs across ranks: rank {rank()} has "
            f"{len(params)}, sum over ranks is {int(n.item())} "
            f"(expected {len(params) * world_size()}).")


# ---------------------------------------------------------------------------
# Bucketed tensor averaging / broadcast
# ---------------------------------------------------------------------------

def _bucketize(tensors: tp.Sequence[torch.Tensor],
               bucket_bytes: int) -> tp.List[tp.List[torch.Tensor]]:
    """Group tensors into flat-reducible buckets of same (device, dtype),
    each up to ``bucket_bytes`` (a single larger tensor forms its own bucket)."""
    buckets: tp.Dict[tp.Tuple[torch.device, torch.dtype], tp.List[tp.List[torch.Tensor]]] = {}
    sizes: tp.Dict[tp.Tuple[torch.device, torch.dtype], int] = {}
    order: tp.List[tp.List[torch.Tensor]] = []
    for t in tensors:
        key = (t.device, t.dtype)
        nbytes = t.numel() * t.element_size()
        group = buckets.setdefault(key, [])
        if not group or sizes[key] + nbytes > bucket_bytes:
            group.append([])
            order.append(group[-1])
            sizes[key] = 0
        group[-1].append(t)
        sizes[key] += nbytes
    return order


def _flatten(ts: tp.Sequence[torch.Tensor]) -> torch.Tensor:
    return torch._utils._flatten_dense_tensors(tuple(ts))


def _unflatten_into(flat: torch.Tensor, ts: tp.Sequence[torch.Tensor]) -> None:
    outs = torch._utils._unflatten_dense_tensors(flat, tuple(ts))
    with torch.no_grad():  # targets may be leaf params (broadcast_model)
        torch._foreach_copy_(list(ts), list(outs))


def average_tensors(tensors: tp.Sequence[torch.Tensor],
                    bucket_bytes: int = DEFAULT_BUCKET_BYTES) -> None:
    """Average tensors in place across ranks: bucketed async all-reduce SUM,
    wait, divide by world size (reference distrib.py:96-111, per-tensor there;
    bucketed here — one RCCL message per bucket over xGMI)."""
    if not is_distributed():
        return
    tensors = [t for t in tensors if t is not None and t.is_floating_point()]
    if not tensors:
        return
    buckets = _bucketize(tensors, bucket_bytes)
    flats, handles = [], []
    for bucket in buckets:
        flat = _flatten(bucket)
        handles.append(dist.all_reduce(flat, op=dist.ReduceOp.SUM, async_op=True))
        flats.append(flat)
    ws = world_size()
    for bucket, flat, handle in zip(buckets, flats, handles):
        handle.wait()
        flat.div_(ws)
        _unflatten_into(flat, bucket)


def broadcast_tensors(tensors: tp.Sequence[torch.Tensor], src: int = 0,
                      bucket_bytes: int = DEFAULT_BUCKET_BYTES) -> None:
    """Broadcast tensors from ``src`` in place, bucketed
    (reference distrib.py:114-127)."""
    if not is_distributed():
        return
    tensors = [t for t in tensors if t is not None]
    if not tensors:
        return
    buckets = _bucketize(tensors, bucket_bytes)
    flats, handles = [], []
    for bucket in buckets:
        flat = _flatten(bucket)
        handles.append(dist.broadcast(flat, src=src, async_op=True))
        flats.append(flat)
    for bucket, flat, handle in zip(buckets, flats, handles):
        handle.wait()
        _unflatten_into(flat, bucket)


def broadcast_model(model: torch.nn.Module, src: int = 0) -> None:
    """Broadcast parameters and buffers from ``src``
    (reference distrib.py:130-133)."""
    broadcast_tensors(list(model.parameters()) + list(model.buffers()), src=src)


# ---------------------------------------------------------------------------
# Post-hoc gradient / model sync  (sync after loss.backward())
# ---------------------------------------------------------------------------

def sync_gradients(params: tp.Iterable[torch.Tensor],
                   bucket_bytes: int = DEFAULT_BUCKET_BYTES) -> None:
    """All-reduce-average the ``.grad`` of every param that has one
    (reference distrib.py:136-151), with the param-count deadlock guard."""
    if not is_distributed():
        return
    grads = [p.grad for p in params
             if p.grad is not None and (p.grad.is_floating_point() or p.grad.is_complex())]
    _check_number_of_params(grads)
    average_tensors(grads, bucket_bytes)


def sync_model(model: torch.nn.Module, sync_buffers: tp.Union[bool, str] = True,
               average_buffers: bool = True,
               bucket_bytes: int = DEFAULT_BUCKET_BYTES) -> None:
    """Post-backward DP sync: average gradients, then sync buffers
    (BatchNorm running stats): averaged by default, or broadcast from rank 0
    (reference distrib.py:193-210).
    """
    if not is_distributed():
        return
    sync_gradients(model.parameters(), bucket_bytes)
    if sync_buffers:
        buffers = [b for b in model.buffers() if b.is_floating_point()]
        if average_buffers:
            average_tensors(buffers, bucket_bytes)
        else:
            broadcast_tensors(buffers, src=0, bucket_bytes=bucket_bytes)


def sync_flat_gradients(optimizer) -> None:
    """DP gradient sync for a :class:`flashy_amd.optim.FlatOptimizer`: the
    gradients already live in one contiguous flat buffer per (device, dtype)
    group, so the sync is a single RCCL all-reduce per group — the maximal
    bucket for the per-link-bound xGMI ring — followed by the exact
    sum/world-size division."""
    if not is_distributed():
        return
    ws = world_size()
    handles = [dist.all_reduce(g, op=dist.ReduceOp.SUM, async_op=True)
               for g in optimizer.grad_buffers]
    for g, h in zip(optimizer.grad_buffers, handles):
        h.wait()
        g.div_(ws)


class OverlappedFlatSync:
    """Backward-overlapped DP gradient sync for a
    :class:`flashy_amd.optim.FlatOptimizer`.

    The flat gradient buffer is partitioned into contiguous chunks of
    ``chunk_bytes`` following parameter registration order.  Each parameter
    gets a persistent ``register_post_accumulate_grad_hook``; the moment the
    last parameter of a chunk has accumulated its gradient inside
    ``backward()``, that chunk's slice of the flat buffer is all-reduced
    asynchronously (RCCL launches on its own stream, ordered after the
    compute stream at the issue point) — so communication overlaps the
    remainder of backward.  ``finish()`` waits the in-flight reduces and
    divides the whole buffer by the world size (sum-then-divide: numerics
    identical to the post-hoc path and the reference oracle).

    Because chunks are contiguous slices of the already-flat buffer there is
    **zero packing copy** — the all-reduce reads/writes the gradient storage
    in place.

    The whole step (backward with its chunk flushes + ``finish`` +
    ``optimizer.step``) is HIP-graph-capturable: hooks run at capture time
    and the recorded graph replays the overlapped schedule, collectives
    included (validated by scripts/rccl_probe.py: RCCL all-reduce inside
    ``torch.cuda.graph`` capture works on this stack).

    ResNet registration order puts ~75% of the bytes in the deep layers
    whose grads complete EARLY in backward, so most bytes are in flight
    while the wide early-layer backward still runs.

    Usage (per step, capturable)::

        sync = OverlappedFlatSync(optim)      # once; installs hooks
        ...
        optim.zero_grad(set_to_none=False)
        loss.backward()                       # chunks flush as they complete
        sync.finish()                         # wait + /world_size
        optim.step()

    Everything is a no-op at world_size == 1 (hooks are not installed).
    """

    def __init__(self, optimizer, chunk_bytes: tp.Optional[int] = None):
        if chunk_bytes is None:
            chunk_bytes = int(os.environ.get("FLASHY_AMD_CHUNK_MB", "8")) << 20
        self.optimizer = optimizer
        self.chunk_bytes = chunk_bytes
        self._hooks: tp.List[tp.Any] = []
        # segment: [flat_g, start, length, n_params]; countdowns reset per step
        self._segments: tp.List[tp.List[tp.Any]] = []
        self._seg_of: tp.Dict[int, int] = {}
        self._remaining: tp.List[int] = []
        self._handles: tp.List[tp.Any] = []
        if not is_distributed():
            return
        for group in optimizer.groups:
            offset = 0
            seg_start, seg_params, seg_bytes = 0, 0, 0
            for p in group.params:
                n = p.numel()
                seg_params += 1
                seg_bytes += n * p.grad.element_size()
                self._seg_of[id(p)] = len(self._segments)
                offset += n
                if seg_bytes >= self.chunk_bytes:
                    self._segments.append(
                        [group.flat_g, seg_start, offset - seg_start, seg_params])
                    seg_start, seg_params, seg_bytes = offset, 0, 0
            if seg_params:
                self._segments.append(
                    [group.flat_g, seg_start, offset - seg_start, seg_params])
        self._remaining = [s[3] for s in self._segments]
        for group in optimizer.groups:
            for p in group.params:
                self._hooks.append(
                    p.register_post_accumulate_grad_hook(self._on_grad))

    @property
    def n_chunks(self) -> int:
        return len(self._segments)

    def _on_grad(self, param: torch.nn.Parameter) -> None:
        i = self._seg_of[id(param)]
        self._remaining[i] -= 1
        if self._remaining[i] == 0:
            flat_g, start, length, _ = self._segments[i]
            self._handles.append(dist.all_reduce(
                flat_g.narrow(0, start, length), op=dist.ReduceOp.SUM,
                async_op=True))

    def finish(self) -> None:
        """Wait in-flight chunk reduces, divide by world size, reset."""
        if not is_distributed():
            return
        # flush stragglers: a param that never produced a grad leaves its
        # segment incomplete — reduce it anyway (the grad bytes are the
        # zeros zero_grad wrote) so the collective schedule stays aligned
        # across ranks regardless of which params got grads
        for i, rem in enumerate(self._remaining):
            if rem > 0:
                flat_g, start, length, _ = self._segments[i]
                self._handles.append(dist.all_reduce(
                    flat_g.narrow(0, start, length), op=dist.ReduceOp.SUM,
                    async_op=True))
        for h in self._handles:
            h.wait()
        self._handles.clear()
        ws = world_size()
        for group in self.optimizer.groups:
            group.flat_g.div_(ws)
        self._remaining = [s[3] for s in self._segments]

    def remove(self) -> None:
        for h in self._hooks:
            h.remove()
        self._hooks.clear()


# ---------------------------------------------------------------------------
# Eager (overlapped) gradient sync — comm on a side HIP stream during backward
# ---------------------------------------------------------------------------

class _EagerSync:
    """Bucketed overlapped gradient all-reduce driven by grad-ready hooks.

    Each param gets a ``register_post_accumulate_grad_hook``; as grads land
    (reverse-topological order inside ``backward()``), params accumulate into
    the current bucket, and a full bucket is flushed immediately: the flat
    buffer is packed on the compute stream, the all-reduce launches on a
    dedicated comm stream ordered by a HIP event — so communication overlaps
    the remainder of backward.  On exit (after backward) the tail bucket is
    flushed, all buckets are waited, divided by world size and unpacked into
    ``param.grad`` with the compute stream made to wait on the comm stream,
    so ``optimizer.step()`` is correctly ordered.

    Constraints kept from the reference (flashy/distrib.py:156-173): at most
    one backward per context; a param producing two grads raises.
    """

    _comm_streams: tp.Dict[torch.device, torch.cuda.Stream] = {}

    def __init__(self, params: tp.Sequence[torch.nn.Parameter], bucket_bytes: int):
        self.params = [p for p in params if p.requires_grad]
        self.bucket_bytes = bucket_bytes
        self._hooks: tp.List[tp.Any] = []
        self._fired: tp.Set[int] = set()
        self._pending: tp.List[torch.nn.Parameter] = []
        self._pending_bytes = 0
        # (bucket params, flat, handle, comm-done event or None)
        self._inflight: tp.List[tp.Tuple[tp.List[torch.nn.Parameter], torch.Tensor,
                                         tp.Any, tp.Optional[torch.cuda.Event]]] = []
        self._entered = False

    @classmethod
    def _comm_stream(cls, dev: torch.device) -> tp.Optional["torch.cuda.Stream"]:
        if dev.type != "cuda":
            return None
        if dev not in cls._comm_streams:
            cls._comm_streams[dev] = torch.cuda.Stream(dev)
        return cls._comm_streams[dev]

    # -- hook machinery ----------------------------------------------------
    def _on_grad(self, param: torch.nn.Parameter) -> None:
        if id(param) in self._fired:
            raise RuntimeError(
                "same parameter produced a gradient twice inside one "
                "eager_sync_gradients context (at most one backward per context)")
        self._fired.add(id(param))
        if param.grad is None:
            return
        self._pending.append(param)
        self._pending_bytes += param.grad.numel() * param.grad.element_size()
        if self._pending_bytes >= self.bucket_bytes:
            self._flush()

    def _flush(self) -> None:
        if not self._pending:
            return
        # split pending by (device, dtype) — one flat reduce per group
        for bucket in _bucketize([p.grad for p in self._pending], 1 << 62):
            params = [p for p in self._pending
                      if p.grad.device == bucket[0].device and p.grad.dtype == bucket[0].dtype]
            grads = bucket
            stream = self._comm_stream(grads[0].device)
            if stream is not None:
                ready = torch.cuda.Event()
                flat = _flatten(grads)       # packed on the compute stream
                ready.record()
                with torch.cuda.stream(stream):
                    stream.wait_event(ready)
                    handle = dist.all_reduce(flat, op=dist.ReduceOp.SUM, async_op=True)
                    done = torch.cuda.Event()
                    done.record(stream)
                self._inflight.append((params, flat, handle, done))
            else:
                flat = _flatten(grads)
                handle = dist.all_reduce(flat, op=dist.ReduceOp.SUM, async_op=True)
                self._inflight.append((params, flat, handle, None))
        self._pending = []
        self._pending_bytes = 0

    # -- context protocol --------------------------------------------------
    def __enter__(self) -> "_EagerSync":
        if self._entered:
            raise RuntimeError("eager sync context is not reentrant")
        self._entered = True
        if not is_distributed():
            return self
        _check_number_of_params([p for p in self.params])
        for p in self.params:
            self._hooks.append(p.register_post_accumulate_grad_hook(self._on_grad))
        return self

    def __exit__(self, exc_type, exc, tb) -> None:
        for h in self._hooks:
            h.remove()
        self._hooks.clear()
        if not is_distributed():
            return
        if exc_type is not None:
            # Error path: DRAIN in-flight async all-reduces before propagating.
            # Leaving them orphaned lets the next collective on this
            # communicator interleave with them and deadlock ranks that did
            # not throw.  Grads are garbage anyway — just wait.
            for _params, _flat, handle, _done in self._inflight:
                try:
                    handle.wait()
                except Exception:  # noqa: BLE001 — already propagating exc
                    pass
            self._inflight.clear()
            self._pending = []
            self._pending_bytes = 0
            return
        # Deadlock guard on the un-fired set: all ranks must agree on how many
        # params never produced a grad (reference distrib.py:186).
        unfired = [p for p in self.params if id(p) not in self._fired]
        _check_number_of_params(unfired)
        self._flush()
        ws = world_size()
        for params, flat, handle, done in self._inflight:
            handle.wait()
            if done is not None:
                torch.cuda.current_stream().wait_event(done)
            flat.div_(ws)
            _unflatten_into(flat, [p.grad for p in params])
        self._inflight.clear()


def eager_sync_gradients(params: tp.Iterable[torch.nn.Parameter],
                         bucket_bytes: int = DEFAULT_BUCKET_BYTES) -> _EagerSync:
    """Context manager wrapping ``backward()``: overlapped bucketed gradient
    all-reduce (reference distrib.py:153-191)."""
    return _EagerSync(list(params), bucket_bytes)


@contextmanager
def eager_sync_model(model: torch.nn.Module, average_buffers: bool = True,
                     bucket_bytes: int = DEFAULT_BUCKET_BYTES):
    """Eager gradient sync for a model plus buffer averaging on exit
    (reference distrib.py:213-224)."""
    with eager_sync_gradients(model.parameters(), bucket_bytes):
        yield
    if is_distributed():
        buffers = [b for b in model.buffers() if b.is_floating_point()]
        if average_buffers:
            average_tensors(buffers, bucket_bytes)
        else:
            broadcast_tensors(buffers, src=0, bucket_bytes=bucket_bytes)


# ---------------------------------------------------------------------------
# Stock-DDP escape hatch + loader sharding
# ---------------------------------------------------------------------------

def wrap(model: torch.nn.Module, **kwargs) -> torch.nn.Module:
    """Wrap in torch DDP on the current device when distributed
    (reference distrib.py:65-75); identity otherwise."""
    if not is_distributed():
        return model
    from torch.nn.parallel import DistributedDataParallel
    if torch.cuda.is_available():
        dev = torch.cuda.current_device()
        return DistributedDataParallel(model, device_ids=[dev], output_device=dev, **kwargs)
    return DistributedDataParallel(model, **kwargs)


class EpochAwareLoader:
    """Transparent loader proxy that calls ``sampler.set_epoch`` before every
    pass, so distributed shuffling differs across epochs (fixes the
    reference's never-called ``set_epoch``)."""

    def __init__(self, base):
        self._base = base
        self._epoch = 0

    def __iter__(self):
        sampler = getattr(self._base, "sampler", None)
        if isinstance(sampler, DistributedSampler):
            sampler.set_epoch(self._epoch)
        self._epoch += 1
        return iter(self._base)

    def __len__(self) -> int:
        return len(self._base)

    def __getattr__(self, name: str):
        return getattr(self._base, name)


def loader(dataset, *args, shuffle: bool = False, klass: type = DataLoader, **kwargs):
    """Build a data loader with the right sharding for the current world.

    * not distributed -> plain loader;
    * distributed + shuffle (training) -> ``DistributedSampler`` wrapped in
      :class:`EpochAwareLoader` (per-epoch reshuffling handled for you);
    * distributed + no shuffle (eval) -> strided ``Subset`` shard, avoiding
      DistributedSampler's padding/duplication.

    Parity: reference distrib.py:227-243.
    """
    if not is_distributed():
        return klass(dataset, *args, shuffle=shuffle, **kwargs)
    if shuffle:
        sampler = DistributedSampler(dataset, shuffle=True)
        return EpochAwareLoader(klass(dataset, *args, sampler=sampler, **kwargs))
    shard = Subset(dataset, list(range(rank(), len(dataset), world_size())))
    return klass(shard, *args, shuffle=False, **kwargs)
