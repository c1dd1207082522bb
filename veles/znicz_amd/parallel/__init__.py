"""Data parallelism over RCCL/xGMI.

This package replaces the reference's entire master/slave plane
(veles/server.py, veles/client.py, veles/launcher.py master mode, the
per-unit generate/apply diff protocol — reference §3.3): one process per
GPU, ``torch.distributed`` with the NCCL API (= RCCL on ROCm), and

* **loader sharding**: each rank trains on a disjoint 1/world stripe of
  the dataset (same shuffle seed → aligned epochs);
* **bucketed gradient all-reduce overlapped with backward**: as each GD
  unit's wgrad GEMM completes (hipEvent), its fp32 gradient joins a
  bucket; full buckets are all-reduced on a dedicated comm stream while
  the compute stream keeps running earlier layers' backward; the fused
  SGD kernel applies each bucket right after its reduction, still on the
  comm stream;
* **one small all-reduce** for Decision's epoch metrics.

Bucket sizing: xGMI is point-to-point, 7 links x ~153 GB/s per GPU; ring
all-reduce is per-link bound, so buckets default to 25 MB — big enough
to amortize launch/collective setup, small enough that several buckets
pipeline across the backward pass (AlexNet fp32 grads ~244 MB -> ~10
buckets).
"""

import os

import numpy


class DistributedContext(object):
    """Thin wrapper over torch.distributed for the unit-graph engine."""

    def __init__(self, rank, world, local_rank, backend):
        self.rank = rank
        self.world = world
        self.local_rank = local_rank
        self.backend = backend
        self._scalar_buf = None

    @classmethod
    def create(cls, backend=None):
        import torch
        import torch.distributed as dist
        rank = int(os.environ.get("RANK", 0))
        world = int(os.environ.get("WORLD_SIZE", 1))
        local_rank = int(os.environ.get("LOCAL_RANK", rank))
        if backend is None:
            backend = os.environ.get("VELES_DIST_BACKEND") or (
                "nccl" if torch.cuda.is_available() else "gloo")
        if not dist.is_initialized():
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29500")
            if backend == "nccl":
                # wrap at the device count: the correctness test lane
                # runs world=2 with both ranks sharing one MI355X
                torch.cuda.set_device(
                    local_rank % max(1, torch.cuda.device_count()))
            dist.init_process_group(backend=backend, rank=rank,
                                    world_size=world)
        return cls(rank, world, local_rank, backend)

    # -- collectives --------------------------------------------------------
    def all_reduce_scalar(self, value):
        """Sum a python scalar across ranks (blocking, tiny)."""
        import torch
        import torch.distributed as dist
        if self.world == 1:
            return value
        dev = "cuda" if self.backend == "nccl" else "cpu"
        t = torch.tensor([float(value)], dtype=torch.float64, device=dev)
        dist.all_reduce(t)
        out = t.item()
        return int(out) if isinstance(value, (int, numpy.integer)) else out

    def all_reduce_(self, tensor, async_op=False):
        import torch.distributed as dist
        if self.world == 1:
            return None
        if self.backend == "gloo" and tensor.is_cuda:
            # test-lane fallback (2 ranks sharing 1 GPU): gloo reduces
            # host copies — correctness only, never the production path
            host = tensor.detach().cpu()
            dist.all_reduce(host)
            tensor.copy_(host.to(tensor.device))
            return None
        return dist.all_reduce(tensor, async_op=async_op)

    def barrier(self):
        import torch.distributed as dist
        if self.world > 1:
            dist.barrier()

    def finalize(self):
        import torch.distributed as dist
        if dist.is_initialized():
            dist.destroy_process_group()


def coalesced_all_reduce(ctx, tensors):
    """Flatten ``tensors`` into one buffer, all-reduce it, scatter back.

    One collective per bucket instead of one per layer — the xGMI ring
    is per-link bound, so fewer/larger messages win (reference §5.8 had
    one zmq message per unit instead).  Works on any backend (gloo for
    the CPU test lane, nccl/RCCL in production).
    """
    import torch
    if ctx is None or ctx.world == 1 or not tensors:
        return
    flat = torch.cat([t.reshape(-1) for t in tensors])
    ctx.all_reduce_(flat)
    off = 0
    for t in tensors:
        t.reshape(-1).copy_(flat[off:off + t.numel()])
        off += t.numel()


class GradientReducer(object):
    """Bucketed all-reduce overlapped with backward + deferred fused SGD.

    GD units call :meth:`submit` right after their wgrad kernels are
    enqueued on the compute stream.  The reducer:

      1. records a hipEvent on the compute stream,
      2. makes the comm stream wait on it,
      3. flattens the unit's fp32 grads into the current bucket; when the
         bucket reaches ``bucket_bytes`` it issues one RCCL all-reduce on
         the comm stream,
      4. enqueues the fused SGD kernels (master fp32 update + bf16/bf16ᵀ
         refresh) on the comm stream after the reduction,
      5. :meth:`sync` (called once per iteration before the next forward)
         makes the compute stream wait on the comm stream's tail event.

    With world == 1 (single GPU) the collective is skipped but the
    comm-stream SGD path is kept, so weight updates still overlap the
    remaining backward kernels.
    """

    def __init__(self, ctx, device, bucket_bytes=None):
        import torch
        if bucket_bytes is None:
            # tunable against the 7-link xGMI per-link bound without a
            # code change (the driver's 8-GPU scaling runs)
            bucket_bytes = \
                int(os.environ.get("VELES_BUCKET_MB", 25)) << 20
        self.ctx = ctx
        self.device = device
        self.bucket_bytes = bucket_bytes
        self.comm_stream = device.comm_stream
        self._pending_apply = []   # (unit, batch) waiting in current bucket
        self._bucket_fill = 0
        self._tail_event = torch.cuda.Event()
        self._have_tail = False
        # multi-tensor SGD descriptor cache: bucket composition ->
        # (hyper-param signature, desc_blob_dev, b2t_dev)
        self._sgd_cache = {}
        # bucket composition -> persistent flat fp32 buffer whose views
        # BACK the bucket's gradient tensors (zero-copy collectives)
        self._flat = {}

    @staticmethod
    def _grad_tensors(u):
        arrs = [u.gradient_weights]
        if u.include_bias and u.gradient_bias.devmem is not None:
            arrs.append(u.gradient_bias)
        return arrs

    def _flat_bucket(self, key):
        """Persistent flat fp32 buffer backing this bucket's gradients.

        First flush for a composition: allocate one flat buffer, copy
        the current gradients in, then REPARENT each gradient Array's
        ``devmem`` to a view of it — every later step's wgrad kernels
        write straight into the flat buffer and the all-reduce needs no
        cat/scatter copies (the round-1 version paid two full-bucket
        device copies per flush).  Segments are 16-B aligned; the pad
        gaps reduce zeros (harmless).
        """
        import torch
        flat = self._flat.get(key)
        if flat is not None:
            return flat
        segs, total = [], 0
        for u, _ in self._pending_apply:
            for arr in self._grad_tensors(u):
                n = arr.devmem.numel()
                segs.append((arr, total, n))
                total += (n + 3) & ~3
        flat = torch.zeros(total, dtype=torch.float32,
                           device=self.device.torch_device)
        for arr, off, n in segs:
            view = flat[off:off + n].view(arr.devmem.shape)
            view.copy_(arr.devmem)
            arr.devmem = view
        self._flat[key] = flat
        # SGD descriptor blobs hold raw data pointers into the OLD
        # gradient tensors — rebuild them against the flat views
        self._sgd_cache.clear()
        return flat

    @property
    def world(self):
        return self.ctx.world if self.ctx else 1

    def submit(self, gd_unit, batch):
        import torch
        ev = torch.cuda.Event()
        ev.record(torch.cuda.current_stream())
        self.comm_stream.wait_event(ev)
        nbytes = gd_unit.gradient_weights.devmem.numel() * 4
        if gd_unit.include_bias and gd_unit.gradient_bias.devmem is not None:
            nbytes += gd_unit.gradient_bias.devmem.numel() * 4
        self._pending_apply.append((gd_unit, batch))
        self._bucket_fill += nbytes
        if self._bucket_fill >= self.bucket_bytes:
            self._flush()

    def _flush(self):
        if not self._pending_apply:
            return
        import torch
        from veles.znicz_amd import ops
        with torch.cuda.stream(self.comm_stream):
            key = tuple(id(u) for u, _ in self._pending_apply)
            if self.world > 1:
                # ONE collective on the persistent flat buffer that
                # backs this bucket's gradient tensors (no copies)
                self.ctx.all_reduce_(self._flat_bucket(key))
            # ONE fused launch for the whole bucket's SGD updates (the
            # per-layer sgd_k launches were a ~5 µs/launch latency tail)
            sig = tuple(
                (float(u.learning_rate), float(u.weights_decay),
                 float(u.l1_vs_l2), float(u.gradient_moment),
                 float(u.learning_rate_bias), float(u.weights_decay_bias),
                 float(u.l1_vs_l2_bias), float(u.gradient_moment_bias),
                 int(batch))
                for u, batch in self._pending_apply)
            cached = self._sgd_cache.get(key)
            if cached is None or cached[0] != sig:
                blob, tbl = ops.sgd_multi_prepare(self._pending_apply,
                                                  self.world)
                self._sgd_cache[key] = cached = (sig, blob, tbl)
            ops.sgd_multi(cached[1], cached[2])
            for u, batch in self._pending_apply:
                u.weights.touch_device()
                if u.include_bias and u.bias is not None and u.bias:
                    u.bias.touch_device()
                u.post_update()
            self._tail_event.record(self.comm_stream)
            self._have_tail = True
        self._pending_apply = []
        self._bucket_fill = 0

    def sync(self):
        """Compute stream waits for all pending comm work (call once per
        iteration, before the next forward consumes weights)."""
        import torch
        self._flush()
        if self._have_tail:
            torch.cuda.current_stream().wait_event(self._tail_event)
            self._have_tail = False
