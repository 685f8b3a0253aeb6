"""AsySG-InCon asynchronous parameter server (reference README.md:56-81).

Semantics (arXiv:1506.08272, "inconsistent reads"): workers compute gradients
on whatever parameter version they last received and push them to the PS
without stalling; the PS applies each push as it arrives (or per `quorum`
pushes) and replies with a current parameter snapshot.  Staleness is bounded
by the worker-side window: a worker blocks only when `max_stale` of its
pushes are still unanswered.

MI355X mapping: one process per GPU; each (PS, worker) pair has its OWN
process group, so its RCCL sends/recvs ride a dedicated xGMI p2p channel and
order independently of other peers (the PS drains all 7 peers concurrently —
the reference instead polled MPI ANY_SOURCE on the host).  All payloads are
flat device tensors; no host round trip.

Message protocol per pair (fixed sizes, fixed per-pair order):
  worker -> PS : hdr int64[2] = (worker_step, param_version_used), wire
  PS -> worker : hdr int64[1] = (ps_version,), param snapshot (model dtype)
A worker sends hdr=(-1,-1) + dummy wire to stop; the PS stops serving a peer
after its stop and finish() returns when all peers stopped.

Colocated mode (default): rank 0 trains too and serves peers opportunistically
between its own steps.  Dedicated mode: rank 0 only serves (`serve()`).
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from .. import ops


class _PeerState:
    __slots__ = ("rank", "group", "hdr", "wire", "reqs", "reply_hdr",
                 "reply_buf", "reply_reqs", "stopped")

    def __init__(self, rank, group, hdr, wire, reply_hdr, reply_buf):
        self.rank = rank
        self.group = group
        self.hdr = hdr
        self.wire = wire
        self.reqs = []
        self.reply_hdr = reply_hdr
        self.reply_buf = reply_buf
        self.reply_reqs = []
        self.stopped = False


class AsyncPSEngine:
    name = "async_ps"

    def __init__(self, flat, codec, comm, grad_scale=1.0, window=2,
                 max_stale=8, quorum=1, dedicated=False):
        self.flat = flat
        self.codec = codec
        self.comm = comm
        self.gscale = grad_scale
        self.window = max(1, int(window))
        self.max_stale = int(max_stale)
        self.quorum = max(1, int(quorum))
        self.dedicated = bool(dedicated)
        self._apply_fn = None

        dev = flat.flat_param.device
        self.device = dev
        total = flat.total
        if codec.name == "topk":
            wn = codec.wire_numel(total, flat.dtype)
        else:
            wn = codec.wire_numel(total)
        self.wire_numel = wn
        self.wire_dtype = codec.wire_dtype(flat.dtype)
        self.staleness_hist = {}
        self.ps_version = 0
        self._accum_count = 0

        if comm.world <= 1:
            return
        if comm.is_ps:
            self.peers = {}
            for w in range(comm.world):
                if w == comm.ps_rank:
                    continue
                st = _PeerState(
                    w, comm.pair_group(w),
                    hdr=torch.zeros(2, dtype=torch.int64, device=dev),
                    wire=torch.zeros(wn, dtype=self.wire_dtype, device=dev),
                    reply_hdr=torch.zeros(1, dtype=torch.int64, device=dev),
                    reply_buf=torch.zeros(total, dtype=flat.dtype, device=dev),
                )
                self.peers[w] = st
            for st in self.peers.values():
                self._post_recv(st)
        else:
            g = comm.pair_group(comm.rank)
            self.group = g
            self.slots = []
            for _ in range(self.window):
                self.slots.append({
                    "hdr": torch.zeros(2, dtype=torch.int64, device=dev),
                    "wire": torch.zeros(wn, dtype=self.wire_dtype, device=dev),
                    "phdr": torch.zeros(1, dtype=torch.int64, device=dev),
                    "pbuf": torch.zeros(total, dtype=flat.dtype, device=dev),
                    "reqs": None,
                })
            self.inflight = []  # slot indices, oldest first
            self.worker_step = 0
            self.param_version = 0
            self.last_applied_step = 0

    # ------------------------------------------------------------------ PS

    def _post_recv(self, st):
        st.reqs = [
            dist.irecv(st.hdr, src=st.rank, group=st.group),
            dist.irecv(st.wire, src=st.rank, group=st.group),
        ]

    def _ps_decode_full(self, wire, beta):
        self.codec.decode_reduce(self.flat.agg, [wire], gscale=self.gscale,
                                 beta=beta, src_dtype=self.flat.dtype)

    def poll_serve(self, metrics, block_for=0):
        """Serve any peers whose push has arrived.  PS-side only."""
        served = 0
        for st in self.peers.values():
            if st.stopped or not st.reqs:
                continue
            done = all(r.is_completed() for r in st.reqs)
            if done or block_for > 0:
                if self._serve_one(st, metrics):
                    served += 1
        return served

    def _serve_one(self, st, metrics):
        flat, codec = self.flat, self.codec
        for r in st.reqs:
            r.wait()
        st.reqs = []
        hdr = st.hdr.tolist()
        if hdr[0] < 0:
            st.stopped = True
            return True
        staleness = max(0, self.ps_version - int(hdr[1]))
        self.staleness_hist[staleness] = \
            self.staleness_hist.get(staleness, 0) + 1
        beta = 1.0 if self._accum_count > 0 else 0.0
        if codec.name == "identity":
            for b in flat.buckets:
                codec.decode_reduce(flat.agg_view(b),
                                    [st.wire[b.start:b.end]],
                                    gscale=self.gscale, beta=beta,
                                    src_dtype=flat.dtype)
        else:
            self._ps_decode_full(st.wire, beta)
        self._accum_count += 1
        if self._accum_count >= self.quorum:
            with metrics.timer("optim_step_time"):
                for b in flat.buckets:
                    self._apply_fn(b)
            self._accum_count = 0
            self.ps_version += 1
        for r in st.reply_reqs:
            r.wait()
        st.reply_hdr.fill_(self.ps_version)
        st.reply_buf.copy_(flat.flat_param)
        st.reply_reqs = [
            dist.isend(st.reply_hdr, dst=st.rank, group=st.group),
            dist.isend(st.reply_buf, dst=st.rank, group=st.group),
        ]
        self._post_recv(st)
        return True

    def serve(self, metrics, until_all_stopped=True):
        """Dedicated-PS loop: serve peers until every peer sent a stop."""
        while True:
            alive = [st for st in self.peers.values() if not st.stopped]
            if not alive:
                break
            progressed = 0
            for st in alive:
                if st.reqs and all(r.is_completed() for r in st.reqs):
                    self._serve_one(st, metrics)
                    progressed += 1
            if progressed == 0:
                # block on one peer to make progress without spinning hot
                st = alive[0]
                self._serve_one(st, metrics)

    # -------------------------------------------------------------- worker

    def _worker_apply_params(self, slot, metrics):
        """Copy a completed param reply into live params."""
        with metrics.timer("decode_time"):
            self.flat.flat_param.copy_(slot["pbuf"])
            self.param_version = int(slot["phdr"].item())
            self.last_applied_step = self.worker_step

    def worker_step_exchange(self, metrics):
        """Push the current gradient; harvest any arrived param reply."""
        flat, codec = self.flat, self.codec
        self.worker_step += 1
        slot_idx = self.worker_step % self.window
        slot = self.slots[slot_idx]
        if slot["reqs"] is not None:
            with metrics.timer("comm_wait"):
                for r in slot["reqs"]:
                    r.wait()
            self._worker_apply_params(slot, metrics)
            slot["reqs"] = None
            self.inflight = [i for i in self.inflight if i != slot_idx]
        with metrics.timer("code_wait"):
            if codec.name == "identity":
                slot["wire"].copy_(flat.flat_grad)
            else:
                codec.encode(flat.flat_grad, slot["wire"])
        slot["hdr"][0] = self.worker_step
        slot["hdr"][1] = self.param_version
        with metrics.timer("isend_time"):
            slot["reqs"] = [
                dist.isend(slot["hdr"], dst=self.comm.ps_rank, group=self.group),
                dist.isend(slot["wire"], dst=self.comm.ps_rank, group=self.group),
                dist.irecv(slot["phdr"], src=self.comm.ps_rank, group=self.group),
                dist.irecv(slot["pbuf"], src=self.comm.ps_rank, group=self.group),
            ]
        self.inflight.append(slot_idx)
        # harvest the oldest reply if it is already here (keeps params fresh)
        while self.inflight:
            i = self.inflight[0]
            s = self.slots[i]
            if all(r.is_completed() for r in s["reqs"]):
                with metrics.timer("comm_wait"):
                    for r in s["reqs"]:
                        r.wait()
                self._worker_apply_params(s, metrics)
                s["reqs"] = None
                self.inflight.pop(0)
            else:
                break
        # staleness bound: block if our params are too old
        if self.worker_step - self.last_applied_step > self.max_stale \
                and self.inflight:
            i = self.inflight.pop(0)
            s = self.slots[i]
            with metrics.timer("comm_wait"):
                for r in s["reqs"]:
                    r.wait()
            self._worker_apply_params(s, metrics)
            s["reqs"] = None
        metrics.add("msg_bytes",
                    self.wire_numel * self.wire_dtype.itemsize
                    + flat.total * flat.dtype.itemsize + 24)
        metrics["staleness"] = self.worker_step - self.last_applied_step

    # ---------------------------------------------------------------- step

    def step(self, apply_fn, metrics):
        self._apply_fn = apply_fn
        comm = self.comm
        if comm.world <= 1:
            flat = self.flat
            for b in flat.buckets:
                ops.reduce_accum(flat.agg_view(b), [flat.grad_view(b)],
                                 scale=self.gscale, beta=0.0)
                apply_fn(b)
            self.ps_version += 1
            return
        if comm.is_ps:
            if self.dedicated:
                raise RuntimeError(
                    "dedicated PS rank must call serve(), not step()")
            # colocated: apply own gradient locally, then serve arrivals
            flat = self.flat
            beta = 1.0 if self._accum_count > 0 else 0.0
            for b in flat.buckets:
                ops.reduce_accum(flat.agg_view(b), [flat.grad_view(b)],
                                 scale=self.gscale, beta=beta)
            self._accum_count += 1
            if self._accum_count >= self.quorum:
                with metrics.timer("optim_step_time"):
                    for b in flat.buckets:
                        apply_fn(b)
                self._accum_count = 0
                self.ps_version += 1
            self.poll_serve(metrics)
        else:
            self.worker_step_exchange(metrics)

    def finish(self, metrics=None):
        from ..utils.metrics import StepMetrics
        metrics = metrics if metrics is not None else StepMetrics()
        comm = self.comm
        if comm.world <= 1:
            return
        self._apply_fn = self._apply_fn or (lambda b: None)
        if comm.is_ps:
            self.serve(metrics)
            # wait out the last replies
            for st in self.peers.values():
                for r in st.reply_reqs:
                    r.wait()
        else:
            # drain outstanding replies
            while self.inflight:
                i = self.inflight.pop(0)
                s = self.slots[i]
                for r in s["reqs"]:
                    r.wait()
                self._worker_apply_params(s, metrics)
                s["reqs"] = None
            # send stop
            dev = self.device
            hdr = torch.full((2,), -1, dtype=torch.int64, device=dev)
            dummy = torch.zeros(self.wire_numel, dtype=self.wire_dtype,
                                device=dev)
            dist.isend(hdr, dst=comm.ps_rank, group=self.group).wait()
            dist.isend(dummy, dst=comm.ps_rank, group=self.group).wait()
        if comm.initialized:
            dist.barrier()
