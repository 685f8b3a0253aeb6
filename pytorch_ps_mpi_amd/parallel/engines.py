"""Synchronous exchange engines.

Three modes mirroring the reference's two wire protocols plus
the degenerate single-process case:

  LocalEngine       world_size == 1 — no comm, grads -> fp32 agg -> update.
  ReplicatedEngine  the reference's SHIPPED path (ps.py:140-190): every rank
                    receives every rank's encoded grad, decodes, sums and
                    applies the identical update -> params stay bitwise equal
                    by determinism.  Identity codec takes the all-reduce fast
                    path; lossy codecs use fixed-capacity all-gather.
  SyncPSEngine      the reference's README plan (README.md:37-46): gather
                    encoded grads to the PS (rank 0), PS decodes+sums+applies,
                    then broadcasts updated params.

All buffers are flat device spans; comms are issued per bucket in fixed
bucket order on every rank (the reference instead barriered on all encodes —
ps.py:129 — and serialized the exchange).
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from .. import ops


class LocalEngine:
    """world_size == 1.  A lossy codec still runs its REAL encode->decode
    round trip per bucket (what one worker would put on the wire and what
    the PS would apply): a 1-GPU bench point of a codec config carries the
    codec's cost and numerics."""

    name = "local"

    def __init__(self, flat, codec, comm, grad_scale=1.0):
        self.flat = flat
        self.codec = codec
        self.gscale = grad_scale
        self._wire = None

    def step(self, apply_fn, metrics):
        flat, codec = self.flat, self.codec
        if codec.name == "identity":
            for b in flat.buckets:
                with metrics.timer("decode_time"):
                    ops.reduce_accum(flat.agg_view(b), [flat.grad_view(b)],
                                     scale=self.gscale, beta=0.0)
                with metrics.timer("optim_step_time"):
                    apply_fn(b)
            metrics.add("msg_bytes", 0)
            return
        if self._wire is None:
            self._wire = {
                b.idx: torch.zeros(codec.wire_numel(b.numel, flat.dtype),
                                   dtype=codec.wire_dtype(flat.dtype),
                                   device=flat.flat_param.device)
                for b in flat.buckets
            }
        for b in flat.buckets:
            w = self._wire[b.idx]
            with metrics.timer("code_wait"):
                codec.encode(flat.grad_view(b), w)
            with metrics.timer("decode_time"):
                codec.decode_reduce(flat.agg_view(b), [w],
                                    gscale=self.gscale, beta=0.0,
                                    src_dtype=flat.dtype)
            with metrics.timer("optim_step_time"):
                apply_fn(b)
        metrics.add("packaged_bytes",
                    sum(w.numel() * w.dtype.itemsize
                        for w in self._wire.values()))

    def finish(self):
        pass


class ReplicatedEngine:
    """Replicated-update DP with backward-hook comm overlap.

    The reference barriered on all encodes before any traffic (ps.py:129);
    here each bucket's collective launches as soon as its last gradient is
    accumulated, in FIXED bucket order on every rank (RCCL collectives must
    be issued in identical order), overlapping comm with the rest of
    backward.  Buckets whose hooks never fired (unused params / no hooks)
    are launched at step() time.
    """

    name = "replicated"
    wants_hooks = True

    def __init__(self, flat, codec, comm, grad_scale=1.0):
        self.flat = flat
        self.codec = codec
        self.comm = comm
        self.gscale = grad_scale
        self.use_allreduce = getattr(codec, "supports_allreduce", False)
        self._works = {}
        self._next = 0
        self._ready = {}
        if not self.use_allreduce:
            dev = flat.flat_param.device
            self.wire_send = {}
            self.wire_slots = {}
            for b in flat.buckets:
                wn = codec.wire_numel(b.numel, flat.dtype)
                wd = codec.wire_dtype(flat.dtype)
                self.wire_send[b.idx] = torch.zeros(wn, dtype=wd, device=dev)
                self.wire_slots[b.idx] = [
                    torch.zeros(wn, dtype=wd, device=dev)
                    for _ in range(comm.world)
                ]

    # ---- hook-driven overlap -----------------------------------------

    def start_step(self):
        self._works = {}
        self._next = 0
        self._ready = {b.idx: 0 for b in self.flat.buckets}

    def on_param_grad(self, param):
        b = self.flat.param_to_bucket.get(param)
        if b is None or not self._ready:
            return
        self._ready[b.idx] += 1
        if self._ready[b.idx] > len(b.params):
            # a second backward before step(): the bucket's collective may
            # already be in flight with PARTIAL gradients — silent wrong
            # results.  Gradient accumulation requires overlap=False.
            raise RuntimeError(
                "gradient accumulation detected with hook-overlap enabled; "
                "construct the optimizer with overlap=False to accumulate "
                "gradients over multiple backward passes")
        # launch every fully-ready bucket at the head of the fixed schedule
        while self._next < len(self.flat.buckets):
            nb = self.flat.buckets[self._next]
            if self._ready.get(nb.idx, 0) < len(nb.params):
                break
            self._launch(nb)
            self._next += 1

    def _launch(self, b):
        flat = self.flat
        if self.use_allreduce:
            self._works[b.idx] = dist.all_reduce(flat.grad_view(b),
                                                 async_op=True)
        else:
            self.codec.encode(flat.grad_view(b), self.wire_send[b.idx])
            self._works[b.idx] = dist.all_gather(self.wire_slots[b.idx],
                                                 self.wire_send[b.idx],
                                                 async_op=True)

    # ---- step ---------------------------------------------------------

    def step(self, apply_fn, metrics):
        flat = self.flat
        with metrics.timer("isend_time"):
            for b in flat.buckets:  # launch whatever the hooks did not
                if b.idx not in self._works:
                    self._launch(b)
        nbytes = flat.total * flat.dtype.itemsize if self.use_allreduce \
            else sum(w.numel() * w.dtype.itemsize
                     for w in self.wire_send.values())
        for b in flat.buckets:
            with metrics.timer("comm_wait"):
                self._works[b.idx].wait()
            with metrics.timer("decode_time"):
                if self.use_allreduce:
                    # all-reduce already summed across ranks -> cast+scale
                    ops.reduce_accum(flat.agg_view(b), [flat.grad_view(b)],
                                     scale=self.gscale, beta=0.0)
                else:
                    self.codec.decode_reduce(flat.agg_view(b),
                                             self.wire_slots[b.idx],
                                             gscale=self.gscale, beta=0.0,
                                             src_dtype=flat.dtype)
            with metrics.timer("optim_step_time"):
                apply_fn(b)
        self._works = {}
        self._ready = {}
        metrics.add("msg_bytes", nbytes)
        metrics.add("packaged_bytes", nbytes)

    def finish(self):
        pass


class SyncPSEngine:
    """Gather -> PS decode+sum+apply -> broadcast (the reference's README
    plan).  Per-bucket gathers launch FROM THE BACKWARD HOOKS in fixed
    bucket order on every rank (gather is a collective, so the schedule must
    match — same discipline as ReplicatedEngine), overlapping the exchange
    with the rest of backward; buckets whose hooks never fired launch at
    step()."""

    name = "sync_ps"
    wants_hooks = True

    def __init__(self, flat, codec, comm, grad_scale=1.0):
        self.flat = flat
        self.codec = codec
        self.comm = comm
        self.gscale = grad_scale
        dev = flat.flat_param.device
        ident = codec.name == "identity"
        self.ident = ident
        self.wire_send = {}
        self.wire_slots = {}
        self._works = {}
        self._next = 0
        self._ready = {}
        for b in flat.buckets:
            if ident:
                self.wire_send[b.idx] = None  # grad view used directly
            else:
                wn = codec.wire_numel(b.numel, flat.dtype)
                wd = codec.wire_dtype(flat.dtype)
                self.wire_send[b.idx] = torch.zeros(wn, dtype=wd, device=dev)
            if comm.is_ps:
                ref = self.wire_send[b.idx]
                if ref is None:
                    self.wire_slots[b.idx] = [
                        torch.zeros(b.numel, dtype=flat.dtype, device=dev)
                        for _ in range(comm.world)
                    ]
                else:
                    self.wire_slots[b.idx] = [torch.zeros_like(ref)
                                              for _ in range(comm.world)]

    def _send_wire(self, b):
        if self.ident:
            return self.flat.grad_view(b)
        return self.wire_send[b.idx]

    # ---- hook-driven overlap (same schedule rules as ReplicatedEngine) ---

    def start_step(self):
        self._works = {}
        self._next = 0
        self._ready = {b.idx: 0 for b in self.flat.buckets}

    def on_param_grad(self, param):
        b = self.flat.param_to_bucket.get(param)
        if b is None or not self._ready:
            return
        self._ready[b.idx] += 1
        if self._ready[b.idx] > len(b.params):
            raise RuntimeError(
                "gradient accumulation detected with hook-overlap enabled; "
                "construct the optimizer with overlap=False to accumulate "
                "gradients over multiple backward passes")
        while self._next < len(self.flat.buckets):
            nb = self.flat.buckets[self._next]
            if self._ready.get(nb.idx, 0) < len(nb.params):
                break
            self._launch(nb)
            self._next += 1

    def _launch(self, b):
        if not self.ident:
            self.codec.encode(self.flat.grad_view(b), self.wire_send[b.idx])
        gl = self.wire_slots[b.idx] if self.comm.is_ps else None
        self._works[b.idx] = dist.gather(self._send_wire(b), gather_list=gl,
                                         dst=self.comm.ps_rank, async_op=True)

    # ---- step ---------------------------------------------------------

    def step(self, apply_fn, metrics):
        flat, codec, comm = self.flat, self.codec, self.comm
        with metrics.timer("isend_time"):
            for b in flat.buckets:  # launch whatever the hooks did not
                if b.idx not in self._works:
                    self._launch(b)
        bworks = []
        for b in flat.buckets:
            with metrics.timer("comm_wait"):
                self._works[b.idx].wait()
            if comm.is_ps:
                with metrics.timer("decode_time"):
                    codec.decode_reduce(flat.agg_view(b),
                                        self.wire_slots[b.idx],
                                        gscale=self.gscale, beta=0.0,
                                        src_dtype=flat.dtype)
                with metrics.timer("optim_step_time"):
                    apply_fn(b)
            bworks.append(dist.broadcast(flat.param_view(b), src=comm.ps_rank,
                                         async_op=True))
        with metrics.timer("comm_wait"):
            for w in bworks:
                w.wait()
        self._works = {}
        self._ready = {}
        wire_b = flat.total * flat.dtype.itemsize if self.ident else \
            sum(t.numel() * t.dtype.itemsize
                for t in self.wire_send.values())
        metrics.add("msg_bytes", wire_b + flat.total * flat.dtype.itemsize)
        metrics.add("packaged_bytes", wire_b)

    def finish(self):
        pass
