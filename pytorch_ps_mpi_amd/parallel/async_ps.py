"""AsySG-InCon asynchronous parameter server (reference README.md:56-81).

Semantics (arXiv:1506.08272, "inconsistent reads"): workers compute gradients
on whatever parameter version they last received and push them to the PS
without stalling; the PS applies each push as it arrives (or per `quorum`
pushes) and replies with a current parameter snapshot.

MI355X mapping — the device-side gradient ring buffer of SURVEY §2.3:
  * one process per GPU; each (PS, worker) pair gets TWO process groups
    (push channel / reply channel), so each channel's RCCL p2p ops order
    independently and ride a dedicated xGMI link pair;
  * the PS keeps a ring of `ring` pre-posted receive slots PER PEER in HBM —
    workers run ahead of the PS's serve cadence up to the ring depth and
    never stall on the PS (the reference instead polled MPI ANY_SOURCE on
    the host with pickled payloads);
  * all payloads are flat device tensors; no host round trips.

Message protocol (fixed sizes; per-channel order is the matching order):
  push  : hdr int64[2] = (worker_step, param_version_used)  +  wire
  reply : hdr int64[1] = (ps_version,)  +  flat_param snapshot (model dtype)
Workers post exactly one reply-recv pair per push (1:1), so every posted
recv is eventually matched.  To stop, a worker sends `ring` stop markers
(hdr=(-1,-1) + dummy wire) — one for every recv slot the PS keeps posted —
and the PS retires that peer without replying.

Colocated mode (default): rank 0 trains too and serves peers opportunistically
between its own steps.  Dedicated mode: rank 0 only serves (`serve()`).
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from .. import ops


def _done(reqs):
    return all(r.is_completed() for r in reqs)


def _wait(reqs, timeout_s=None, what=""):
    """Wait all reqs; with timeout_s, log + raise if a peer goes silent
    (the reference assumed 'communication is reliable', README.md:7-8 —
    this is the PS-side timeout logging SURVEY §5 calls for)."""
    if timeout_s is None:
        for r in reqs:
            r.wait()
        return
    import datetime
    for r in reqs:
        try:
            r.wait(datetime.timedelta(seconds=timeout_s))
        except Exception as e:
            import logging
            logging.getLogger(__name__).error(
                "async-PS wait timed out after %ss (%s): %r",
                timeout_s, what, e)
            raise


class _Peer:
    """PS-side per-worker state: recv ring + reply ring."""

    __slots__ = ("rank", "push_g", "reply_g", "slots", "head", "replies",
                 "rhead", "stopped")

    def __init__(self, rank, push_g, reply_g, ring, reply_ring, wn, wdt,
                 total, pdt, dev):
        self.rank = rank
        self.push_g = push_g
        self.reply_g = reply_g
        self.slots = [{
            "hdr": torch.zeros(2, dtype=torch.int64, device=dev),
            "wire": torch.zeros(wn, dtype=wdt, device=dev),
            "reqs": None,
        } for _ in range(ring)]
        self.head = 0
        self.replies = [{
            "hdr": torch.zeros(1, dtype=torch.int64, device=dev),
            "buf": torch.zeros(total, dtype=pdt, device=dev),
            "reqs": [],
        } for _ in range(reply_ring)]
        self.rhead = 0
        self.stopped = False

    def post(self, slot):
        slot["reqs"] = [
            dist.irecv(slot["hdr"], src=self.rank, group=self.push_g),
            dist.irecv(slot["wire"], src=self.rank, group=self.push_g),
        ]


class AsyncPSEngine:
    name = "async_ps"

    def __init__(self, flat, codec, comm, grad_scale=1.0, window=4,
                 max_stale=8, quorum=1, dedicated=False, reply_ring=2,
                 serve_timeout_s=None):
        self.flat = flat
        self.codec = codec
        self.comm = comm
        self.gscale = grad_scale
        self.ring = max(1, int(window))     # PS recv slots per peer
        self.window = self.ring             # worker in-flight push bound
        self.max_stale = int(max_stale)
        self.quorum = max(1, int(quorum))
        self.dedicated = bool(dedicated)
        self.serve_timeout_s = serve_timeout_s
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
                st = _Peer(w, comm.push_group(w), comm.reply_group(w),
                           self.ring, reply_ring, wn, self.wire_dtype,
                           total, flat.dtype, dev)
                self.peers[w] = st
            # recv-ring posting is DEFERRED to first step()/serve(): with
            # RCCL the first op on a pair communicator blocks the host in
            # ncclCommInitRank until the peer joins, and at __init__ time
            # the peers are still waiting for the optimizer's initial param
            # broadcast (deadlock).  gloo initializes groups eagerly and
            # does not care.
            self._started = False
        else:
            self.push_g = comm.push_group(comm.rank)
            self.reply_g = comm.reply_group(comm.rank)
            self.pushes = [{
                "hdr": torch.zeros(2, dtype=torch.int64, device=dev),
                "wire": torch.zeros(wn, dtype=self.wire_dtype, device=dev),
                "reqs": None,
            } for _ in range(self.window)]
            self.rslots = [{
                "hdr": torch.zeros(1, dtype=torch.int64, device=dev),
                "buf": torch.zeros(total, dtype=flat.dtype, device=dev),
                "reqs": None,
            } for _ in range(self.window)]
            self.sent = 0
            self.harvested = 0
            self.worker_step = 0
            self.param_version = 0
            self.last_applied_step = 0

    # ------------------------------------------------------------------ PS

    def _start_ps(self):
        """Pre-post every recv slot, per peer, in ring order (first use)."""
        if self._started:
            return
        self._started = True
        for st in self.peers.values():
            for s in st.slots:
                st.post(s)

    def _serve_slot(self, st, metrics, timeout_s=None):
        """Process the head recv slot of peer st (must be completed/waited)."""
        flat, codec = self.flat, self.codec
        slot = st.slots[st.head]
        _wait(slot["reqs"], timeout_s, f"push from worker {st.rank}")
        slot["reqs"] = None
        hdr = slot["hdr"].tolist()
        if hdr[0] < 0:
            # stop marker: drain the remaining posted slots (the worker sends
            # `ring` markers, one per posted slot), never repost
            for k in range(1, self.ring):
                s2 = st.slots[(st.head + k) % self.ring]
                if s2["reqs"] is not None:
                    _wait(s2["reqs"])
                    s2["reqs"] = None
            st.stopped = True
            return
        staleness = max(0, self.ps_version - int(hdr[1]))
        self.staleness_hist[staleness] = \
            self.staleness_hist.get(staleness, 0) + 1
        beta = 1.0 if self._accum_count > 0 else 0.0
        with metrics.timer("decode_time"):
            if codec.name == "identity":
                for b in flat.buckets:
                    codec.decode_reduce(flat.agg_view(b),
                                        [slot["wire"][b.start:b.end]],
                                        gscale=self.gscale, beta=beta,
                                        src_dtype=flat.dtype)
            else:
                codec.decode_reduce(flat.agg, [slot["wire"]],
                                    gscale=self.gscale, beta=beta,
                                    src_dtype=flat.dtype)
        self._accum_count += 1
        if self._accum_count >= self.quorum:
            with metrics.timer("optim_step_time"):
                for b in flat.buckets:
                    self._apply_fn(b)
            self._accum_count = 0
            self.ps_version += 1
        # reply with a parameter snapshot on the reply channel
        rep = st.replies[st.rhead]
        st.rhead = (st.rhead + 1) % len(st.replies)
        _wait(rep["reqs"])  # snapshot buffer must be free
        rep["hdr"].fill_(self.ps_version)
        rep["buf"].copy_(flat.flat_param)
        rep["reqs"] = [
            dist.isend(rep["hdr"], dst=st.rank, group=st.reply_g),
            dist.isend(rep["buf"], dst=st.rank, group=st.reply_g),
        ]
        # repost this recv slot at the tail of the ring
        st.post(slot)
        st.head = (st.head + 1) % self.ring

    def poll_serve(self, metrics, max_per_peer=None):
        """Serve arrived pushes without blocking. PS-side only."""
        self._start_ps()
        served = 0
        budget = max_per_peer if max_per_peer is not None else self.ring
        for st in self.peers.values():
            n = 0
            while (not st.stopped and n < budget
                   and st.slots[st.head]["reqs"] is not None
                   and _done(st.slots[st.head]["reqs"])):
                self._serve_slot(st, metrics)
                served += 1
                n += 1
        return served

    def serve(self, metrics):
        """Dedicated-PS loop: serve until every peer sent its stop."""
        self._start_ps()
        while True:
            alive = [st for st in self.peers.values() if not st.stopped]
            if not alive:
                break
            progressed = self.poll_serve(metrics)
            if progressed == 0:
                # block on one peer's head slot to make progress
                self._serve_slot(alive[0], metrics,
                                 timeout_s=self.serve_timeout_s)

    # -------------------------------------------------------------- worker

    def _apply_reply(self, slot, metrics, skip_copy=False):
        if not skip_copy:
            with metrics.timer("decode_time"):
                self.flat.flat_param.copy_(slot["buf"])
                self.param_version = int(slot["hdr"].item())
                self.last_applied_step = self.worker_step
        slot["reqs"] = None
        self.harvested += 1

    def _harvest_replies(self, metrics, block_one=False):
        """Consume completed replies in order; optionally block for one.
        A reply superseded by an already-arrived newer one is retired without
        copying (only the freshest parameters matter)."""
        blocked = False
        while self.harvested < self.sent:
            slot = self.rslots[self.harvested % self.window]
            if slot["reqs"] is None:
                break
            if _done(slot["reqs"]) or (block_one and not blocked):
                with metrics.timer("comm_wait"):
                    _wait(slot["reqs"])
                nxt = self.rslots[(self.harvested + 1) % self.window] \
                    if self.harvested + 1 < self.sent else None
                newer = (nxt is not None and nxt["reqs"] is not None
                         and _done(nxt["reqs"]))
                self._apply_reply(slot, metrics, skip_copy=newer)
                blocked = True
            else:
                break

    def worker_step_exchange(self, metrics):
        flat, codec = self.flat, self.codec
        self.worker_step += 1
        psh = self.pushes[self.sent % self.window]
        rsl = self.rslots[self.sent % self.window]
        if psh["reqs"] is not None:
            # window full: previous push in this slot must be fully sent and
            # its reply harvested before the buffers are reused
            with metrics.timer("comm_wait"):
                _wait(psh["reqs"])
            if rsl["reqs"] is not None:
                with metrics.timer("comm_wait"):
                    _wait(rsl["reqs"])
                self._apply_reply(rsl, metrics)
            psh["reqs"] = None
        with metrics.timer("code_wait"):
            if codec.name == "identity":
                psh["wire"].copy_(flat.flat_grad)
            else:
                codec.encode(flat.flat_grad, psh["wire"])
        psh["hdr"][0] = self.worker_step
        psh["hdr"][1] = self.param_version
        with metrics.timer("isend_time"):
            psh["reqs"] = [
                dist.isend(psh["hdr"], dst=self.comm.ps_rank,
                           group=self.push_g),
                dist.isend(psh["wire"], dst=self.comm.ps_rank,
                           group=self.push_g),
            ]
            rsl["reqs"] = [
                dist.irecv(rsl["hdr"], src=self.comm.ps_rank,
                           group=self.reply_g),
                dist.irecv(rsl["buf"], src=self.comm.ps_rank,
                           group=self.reply_g),
            ]
        self.sent += 1
        # harvest whatever replies already arrived (keeps params fresh)
        self._harvest_replies(metrics)
        # bounded staleness: block for one reply if params are too old
        if self.worker_step - self.last_applied_step > self.max_stale:
            self._harvest_replies(metrics, block_one=True)
        metrics.add("msg_bytes",
                    self.wire_numel * self.wire_dtype.itemsize
                    + flat.total * flat.dtype.itemsize + 24)
        metrics.add("packaged_bytes",
                    self.wire_numel * self.wire_dtype.itemsize)
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
            flat = self.flat
            beta = 1.0 if self._accum_count > 0 else 0.0
            with metrics.timer("decode_time"):
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
            if self.staleness_hist:
                metrics["pushes_served"] = sum(self.staleness_hist.values())
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
            for st in self.peers.values():
                for rep in st.replies:
                    _wait(rep["reqs"])
        else:
            # every push gets a reply: drain them all
            while self.harvested < self.sent:
                self._harvest_replies(metrics, block_one=True)
            for psh in self.pushes:
                if psh["reqs"] is not None:
                    _wait(psh["reqs"])
                    psh["reqs"] = None
            # one stop marker per PS recv slot so every posted irecv matches
            hdr = torch.full((2,), -1, dtype=torch.int64, device=self.device)
            dummy = torch.zeros(self.wire_numel, dtype=self.wire_dtype,
                                device=self.device)
            for _ in range(self.ring):
                dist.isend(hdr, dst=comm.ps_rank, group=self.push_g).wait()
                dist.isend(dummy, dst=comm.ps_rank, group=self.push_g).wait()
        if comm.initialized:
            dist.barrier()
