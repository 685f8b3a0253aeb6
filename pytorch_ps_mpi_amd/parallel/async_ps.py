"""AsySG-InCon asynchronous parameter server (reference README.md:56-81).

Semantics (arXiv:1506.08272, "inconsistent reads"): workers compute gradients
on whatever parameter version they last received and push them to the PS
without stalling; the PS applies each push as it arrives (or per `quorum`
pushes) and replies with current parameters.

MI355X mapping — the device-side gradient ring buffer of SURVEY §2.3:
  * one process per GPU; each (PS, worker) pair gets TWO process groups
    (push channel / reply channel), so each channel's RCCL p2p ops order
    independently and ride a dedicated xGMI link pair;
  * the PS keeps a ring of `ring` pre-posted receive slots PER PEER in HBM —
    workers run ahead of the PS's serve cadence up to the ring depth and
    never stall on the PS (the reference instead polled MPI ANY_SOURCE on
    the host with pickled payloads);
  * all payloads are flat device tensors; no host round trips.

Round-2 protocol (bucket-pipelined; replaces the round-1 whole-model wire):

  push  : hdr int64[2] = (worker_step, param_version_used)
          + one wire message PER BUCKET, in fixed bucket order.  The worker
          launches each bucket's encode+isend from its backward hooks the
          moment the bucket's last gradient is accumulated (the reference's
          encode-during-backward thread pool, ps.py:85,98-101, done as
          device-side overlap), so by the time backward ends most of the
          push is already on the wire.
  reply : hdr int64[2] = (ps_version, shard_idx) + ONE PARAMETER SHARD
          (a contiguous group of buckets, round-robin per peer).  A push no
          longer costs a full-model snapshot both ways: with S reply shards
          the reply bytes drop S× and a worker's parameters are fully
          refreshed every S pushes — "inconsistent reads" is the algorithm's
          contract, so a patchwork of shard versions is within semantics
          (staleness accounting below stays honest about it).

Workers post exactly one reply-recv pair per push (1:1), so every posted
recv is eventually matched.  To stop, a worker sends `ring` stop markers
(hdr=(-1,-1) + dummy bucket wires) — one for every recv slot the PS keeps
posted — and the PS retires that peer without replying.

Staleness bookkeeping: each push slot records the worker_step at PUSH time;
when that push's reply is applied, `last_applied_step` is set from the
recorded step (not the harvest step), so `max_stale` bounds the true age of
the newest parameter content (advisor round-1 finding).  `param_version` is
the ps_version of the most recently applied reply; with sharded replies the
oldest shard may lag up to S-1 replies behind it.

Failure handling: the dedicated-PS serve loop polls ALL peers (any-source
semantics, like the reference's MPI.ANY_SOURCE plan) and, when
`serve_timeout_s` is set, RETIRES a peer that has gone silent instead of
dying — remaining workers keep training; `peers_dropped` is surfaced in
metrics.  (The reference assumed "communication is reliable", README.md:7-8.)

Colocated mode (default): rank 0 trains too and serves peers opportunistically
between its own steps.  Dedicated mode: rank 0 only serves (`serve()`).

Arrival detection: every push and every reply ends with a TAIL tag message
(int64[1], nonzero).  The receiver detects a complete message by polling the
tail slot's CONTENT (channels deliver in order, so a landed tail implies the
whole message landed), then `wait()`s the posted works (instant at that
point) before touching the payload — the device-ring "sequence tag" idiom.
This is deliberate: torch.distributed's gloo p2p `Work.is_completed()` never
turns true without `wait()` (the data lands, the flag doesn't — verified on
torch 2.10), so request-object polling cannot drive an any-source server;
content tags work identically over gloo and RCCL.
"""

from __future__ import annotations

import logging
import time

import torch
import torch.distributed as dist

from .. import ops
from ..utils.metrics import StepMetrics

log = logging.getLogger(__name__)

WSEG_ALIGN = 256  # wire-segment alignment (elements) — keeps every bucket
#                   segment 16B-vectorizable and int32-viewable for the codecs


_poll_streams = {}


def _tag_value(t):
    """Read an arrival tag.  On GPU the D2H copy goes on a dedicated side
    stream: .item() on the current stream would queue behind every
    outstanding compute kernel — a full stream drain per poll.  Safe
    because tags are written by the comm stream and polled monotonically
    (0 -> nonzero)."""
    if t.is_cuda:
        s = _poll_streams.get(t.device)
        if s is None:
            s = torch.cuda.Stream(device=t.device)
            _poll_streams[t.device] = s
        with torch.cuda.stream(s):
            return int(t.item())
    return int(t.item())


def _tagged(slot):
    """True when the slot's tail tag landed => the whole message landed.

    The check is an EXACT match against the slot's expected sequence value
    (stop markers are negative): a reused slot's previous tag, or a re-arm
    zero_() still in flight on another stream, can never read as "arrived",
    so a non-blocking poll never turns into a blocking wait."""
    if slot["reqs"] is None:
        return False
    v = _tag_value(slot["tail"])
    return v == slot["expect"] or v < 0


def _wait(reqs, timeout_s=None, what=""):
    """Wait all reqs; with timeout_s, log + raise if a peer goes silent."""
    if not reqs:
        return
    if timeout_s is None:
        for r in reqs:
            r.wait()
        return
    import datetime
    for r in reqs:
        try:
            r.wait(datetime.timedelta(seconds=timeout_s))
        except Exception as e:
            log.error("async-PS wait timed out after %ss (%s): %r",
                      timeout_s, what, e)
            raise


class _Peer:
    """PS-side per-worker state: recv ring + reply ring + reply-shard cursor."""

    __slots__ = ("rank", "push_g", "reply_g", "slots", "head", "replies",
                 "rhead", "cursor", "stopped", "dropped", "last_seen",
                 "post_seq")

    def __init__(self, rank, push_g, reply_g, ring, reply_ring, wire_total,
                 wdt, shard_max, pdt, dev):
        self.rank = rank
        self.push_g = push_g
        self.reply_g = reply_g
        self.slots = [{
            "hdr": torch.zeros(2, dtype=torch.int64, device=dev),
            "wire": torch.zeros(wire_total, dtype=wdt, device=dev),
            "tail": torch.zeros(1, dtype=torch.int64, device=dev),
            "reqs": None,
            "expect": 0,
        } for _ in range(ring)]
        self.post_seq = 0  # worker pushes are numbered 1,2,...; a slot's
        #                    expected tail = the sequence of the push that
        #                    will land in it (stop markers are -1)
        self.head = 0
        self.replies = [{
            "hdr": torch.zeros(2, dtype=torch.int64, device=dev),
            "buf": torch.zeros(shard_max, dtype=pdt, device=dev),
            "tail": torch.ones(1, dtype=torch.int64, device=dev),
            "reqs": [],
        } for _ in range(reply_ring)]
        self.rhead = 0
        self.cursor = 0       # next reply shard index for this peer
        self.stopped = False
        self.dropped = False
        self.last_seen = time.monotonic()


class AsyncPSEngine:
    name = "async_ps"

    def __init__(self, flat, codec, comm, grad_scale=1.0, window=4,
                 max_stale=8, quorum=1, dedicated=False, reply_ring=2,
                 serve_timeout_s=None, reply_shards="auto"):
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
        nb = len(flat.buckets)

        # per-bucket wire segment table (aligned so every segment slice is
        # 16B-vectorizable and int32-viewable whatever the codec layout)
        self.wseg = []
        off = 0
        for b in flat.buckets:
            wn = codec.wire_numel(b.numel, flat.dtype)
            self.wseg.append((off, wn))
            off += (wn + WSEG_ALIGN - 1) // WSEG_ALIGN * WSEG_ALIGN
        self.wire_total = off
        self.wire_dtype = codec.wire_dtype(flat.dtype)

        # reply shards: contiguous groups of buckets, balanced by elements
        if reply_shards in (None, "auto"):
            ns = min(nb, max(2, self.window))
        elif reply_shards in ("full", 1):
            ns = 1
        else:
            ns = max(1, min(nb, int(reply_shards)))
        self.shards = self._make_shards(flat.buckets, ns)
        self.n_shards = len(self.shards)
        self.shard_max = max(hi - lo for lo, hi in self.shards)

        self.staleness_hist = {}
        self.ps_version = 0
        self._accum_count = 0
        self.peers_dropped = 0

        if comm.world <= 1:
            return
        self.wants_hooks = not comm.is_ps
        if comm.is_ps:
            self.peers = {}
            for w in range(comm.world):
                if w == comm.ps_rank:
                    continue
                self.peers[w] = _Peer(
                    w, comm.push_group(w), comm.reply_group(w), self.ring,
                    reply_ring, self.wire_total, self.wire_dtype,
                    self.shard_max, flat.dtype, dev)
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
                "wire": torch.zeros(self.wire_total, dtype=self.wire_dtype,
                                    device=dev),
                "tail": torch.zeros(1, dtype=torch.int64, device=dev),
                "reqs": None,
                "step": 0,
            } for _ in range(self.window)]
            self.rslots = [{
                "hdr": torch.zeros(2, dtype=torch.int64, device=dev),
                "buf": torch.zeros(self.shard_max, dtype=flat.dtype,
                                   device=dev),
                "tail": torch.zeros(1, dtype=torch.int64, device=dev),
                "reqs": None,
                "push_step": 0,
                "shard": 0,
                "expect": 0,
            } for _ in range(self.window)]
            self.sent = 0
            self.harvested = 0
            self.worker_step = 0
            self.param_version = 0
            self.last_applied_step = 0
            self.cursor = 0          # worker's mirror of the PS reply cursor
            self._hook_metrics = StepMetrics()
            # hook-launch state (fixed bucket schedule, like ReplicatedEngine)
            self._ready = {}
            self._next = 0
            self._cur = None         # push slot being filled this step

    @staticmethod
    def _make_shards(buckets, ns):
        """Partition buckets into ns contiguous groups, balanced by numel.
        Returns [(flat_lo, flat_hi), ...]."""
        total = buckets[-1].end
        target = total / ns
        shards = []
        lo = 0
        acc = 0
        for i, b in enumerate(buckets):
            acc += b.numel
            last = i == len(buckets) - 1
            if (acc >= target * (len(shards) + 1) - 1e-9 and
                    len(shards) < ns - 1) or last:
                shards.append((lo, b.end))
                lo = b.end
        return shards

    def initial_param_sync(self):
        """Deliver rank 0's initial parameters over the per-peer reply
        channels (p2p) instead of a default-group broadcast.  This keeps the
        whole async engine collective-free, which (a) matches the PS pattern
        — xGMI is point-to-point, rank 0 reaches each peer on its own link —
        and (b) lets world>1 RCCL tests run with two ranks sharing one GPU
        (RCCL rejects same-device COLLECTIVES with 'Duplicate GPU detected'
        but serves same-device p2p fine — tools/nccl_probe.py)."""
        if self.comm.world <= 1:
            return
        fp = self.flat.flat_param
        if self.comm.is_ps:
            for w, st in self.peers.items():
                dist.isend(fp, dst=w, group=st.reply_g).wait()
        else:
            dist.irecv(fp, src=self.comm.ps_rank, group=self.reply_g).wait()

    # ------------------------------------------------------------------ PS

    def _start_ps(self):
        """Pre-post every recv slot, per peer, in ring order (first use)."""
        if self._started:
            return
        self._started = True
        for st in self.peers.values():
            for s in st.slots:
                self._post(st, s)

    def _post(self, st, slot):
        st.post_seq += 1
        slot["expect"] = st.post_seq
        slot["tail"].zero_()  # re-arm the arrival tag BEFORE posting
        reqs = [dist.irecv(slot["hdr"], src=st.rank, group=st.push_g)]
        for off, wn in self.wseg:
            reqs.append(dist.irecv(slot["wire"][off:off + wn],
                                   src=st.rank, group=st.push_g))
        reqs.append(dist.irecv(slot["tail"], src=st.rank, group=st.push_g))
        slot["reqs"] = reqs

    def _drop_peer(self, st, why):
        log.error("async-PS dropping worker %d (%s); continuing with "
                  "remaining peers", st.rank, why)
        st.dropped = True
        st.stopped = True
        self.peers_dropped += 1

    def _serve_slot(self, st, metrics, timeout_s=None):
        """Process the head recv slot of peer st (caller ensured completed,
        or wants a blocking wait with optional timeout)."""
        flat, codec = self.flat, self.codec
        slot = st.slots[st.head]
        _wait(slot["reqs"], timeout_s, f"push from worker {st.rank}")
        slot["reqs"] = None
        st.last_seen = time.monotonic()
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
            for b, (off, wn) in zip(flat.buckets, self.wseg):
                codec.decode_reduce(flat.agg_view(b),
                                    [slot["wire"][off:off + wn]],
                                    gscale=self.gscale, beta=beta,
                                    src_dtype=flat.dtype)
        self._accum_count += 1
        if self._accum_count >= self.quorum:
            with metrics.timer("optim_step_time"):
                for b in flat.buckets:
                    self._apply_fn(b)
            self._accum_count = 0
            self.ps_version += 1
        # reply with the peer's next parameter shard on the reply channel
        rep = st.replies[st.rhead]
        st.rhead = (st.rhead + 1) % len(st.replies)
        _wait(rep["reqs"])  # shard buffer must be free
        sh = st.cursor % self.n_shards
        st.cursor += 1
        lo, hi = self.shards[sh]
        rep["hdr"][0] = self.ps_version
        rep["hdr"][1] = sh
        rep["buf"][:hi - lo].copy_(flat.flat_param[lo:hi])
        rep["tail"].fill_(st.cursor)  # nonzero (cursor was incremented)
        rep["reqs"] = [
            dist.isend(rep["hdr"], dst=st.rank, group=st.reply_g),
            dist.isend(rep["buf"][:hi - lo], dst=st.rank, group=st.reply_g),
            dist.isend(rep["tail"], dst=st.rank, group=st.reply_g),
        ]
        # repost this recv slot at the tail of the ring
        self._post(st, slot)
        st.head = (st.head + 1) % self.ring

    def poll_serve(self, metrics, max_per_peer=None, drop_on_error=False):
        """Serve arrived pushes without blocking. PS-side only.

        With drop_on_error (dedicated serve loop) an exception while serving
        one peer — e.g. gloo "connection closed by peer" from a dead worker —
        retires THAT peer and serving continues for the rest."""
        self._start_ps()
        served = 0
        budget = max_per_peer if max_per_peer is not None else self.ring
        for st in self.peers.values():
            n = 0
            try:
                while (not st.stopped and n < budget
                       and _tagged(st.slots[st.head])):
                    self._serve_slot(st, metrics)
                    served += 1
                    n += 1
            except Exception as e:
                if not drop_on_error:
                    raise
                self._drop_peer(st, f"serve error: {e!r}")
        return served

    def serve(self, metrics):
        """Dedicated-PS event loop: ANY-SOURCE serve until every peer sent
        its stop (or was dropped after `serve_timeout_s` of silence)."""
        self._start_ps()
        for st in self.peers.values():
            st.last_seen = time.monotonic()
        while True:
            alive = [st for st in self.peers.values() if not st.stopped]
            if not alive:
                break
            progressed = self.poll_serve(metrics, drop_on_error=True)
            if progressed == 0:
                now = time.monotonic()
                if self.serve_timeout_s is not None:
                    for st in alive:
                        if now - st.last_seen > self.serve_timeout_s:
                            self._drop_peer(
                                st, f"silent for {self.serve_timeout_s}s")
                time.sleep(0.0002)
        if self.peers_dropped:
            metrics["peers_dropped"] = self.peers_dropped

    # -------------------------------------------------------------- worker

    def start_step(self):
        """Arm the hook-launch schedule for the next backward (worker)."""
        if self.comm.world <= 1 or self.comm.is_ps:
            return
        self._ready = {b.idx: 0 for b in self.flat.buckets}
        self._next = 0
        self._cur = None

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
            self._launch_bucket(nb, self._hook_metrics)
            self._next += 1

    def _acquire_slot(self, metrics):
        """Claim the next push slot (blocks when the window is full) and
        send the push header.  Called at the first bucket launch of a step."""
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
        psh["hdr"][0] = self.worker_step
        psh["hdr"][1] = self.param_version
        psh["step"] = self.worker_step
        with metrics.timer("isend_time"):
            psh["reqs"] = [dist.isend(psh["hdr"], dst=self.comm.ps_rank,
                                      group=self.push_g)]
        self._cur = psh
        return psh

    def _launch_bucket(self, b, metrics):
        psh = self._cur if self._cur is not None \
            else self._acquire_slot(metrics)
        off, wn = self.wseg[b.idx]
        seg = psh["wire"][off:off + wn]
        with metrics.timer("code_wait"):
            self.codec.encode(self.flat.grad_view(b), seg)
        with metrics.timer("isend_time"):
            psh["reqs"].append(
                dist.isend(seg, dst=self.comm.ps_rank, group=self.push_g))

    def _apply_reply(self, slot, metrics):
        with metrics.timer("decode_time"):
            sh = slot["shard"]
            lo, hi = self.shards[sh]
            self.flat.flat_param[lo:hi].copy_(slot["buf"][:hi - lo])
            self.param_version = int(slot["hdr"][0].item())
            # true staleness: age of the parameters just applied = the step
            # at which their push was sent (advisor round-1 fix)
            self.last_applied_step = max(self.last_applied_step,
                                         slot["push_step"])
        slot["reqs"] = None
        self.harvested += 1

    def _harvest_replies(self, metrics, block_one=False):
        """Consume completed replies in order; optionally block for one."""
        blocked = False
        while self.harvested < self.sent:
            slot = self.rslots[self.harvested % self.window]
            if slot["reqs"] is None:
                break
            if _tagged(slot) or (block_one and not blocked):
                with metrics.timer("comm_wait"):
                    _wait(slot["reqs"])
                self._apply_reply(slot, metrics)
                blocked = True
            else:
                break

    def worker_step_exchange(self, metrics):
        flat = self.flat
        # launch whatever the hooks did not (no-overlap mode, frozen params)
        start = self._next if self._ready else 0
        for b in flat.buckets[start:]:
            self._launch_bucket(b, metrics)
        psh = self._cur
        if psh is None:  # no buckets at all (cannot happen in practice)
            return
        # close the push with its arrival tag, post the 1:1 reply recv
        rsl = self.rslots[self.sent % self.window]
        sh = self.cursor % self.n_shards
        self.cursor += 1
        lo, hi = self.shards[sh]
        rsl["shard"] = sh
        rsl["push_step"] = psh["step"]
        rsl["expect"] = self.cursor  # PS reply tail = its cursor (mirrored)
        rsl["tail"].zero_()
        psh["tail"].fill_(self.worker_step)
        with metrics.timer("isend_time"):
            psh["reqs"].append(dist.isend(psh["tail"], dst=self.comm.ps_rank,
                                          group=self.push_g))
            rsl["reqs"] = [
                dist.irecv(rsl["hdr"], src=self.comm.ps_rank,
                           group=self.reply_g),
                dist.irecv(rsl["buf"][:hi - lo], src=self.comm.ps_rank,
                           group=self.reply_g),
                dist.irecv(rsl["tail"], src=self.comm.ps_rank,
                           group=self.reply_g),
            ]
        self.sent += 1
        self._cur = None
        self._ready = {}
        # merge hook-time spans into this step's metrics
        for k, v in self._hook_metrics.items():
            metrics.add(k, v)
        self._hook_metrics = StepMetrics()
        # harvest whatever replies already arrived (keeps params fresh)
        self._harvest_replies(metrics)
        # bounded staleness: block for one reply if params are too old
        if self.worker_step - self.last_applied_step > self.max_stale:
            self._harvest_replies(metrics, block_one=True)
        wire_b = sum(wn for _, wn in self.wseg) * self.wire_dtype.itemsize
        shard_b = (hi - lo) * flat.dtype.itemsize
        metrics.add("msg_bytes", wire_b + shard_b + 32)
        metrics.add("packaged_bytes", wire_b)
        metrics["staleness"] = self.worker_step - self.last_applied_step

    # ---------------------------------------------------------------- step

    def step(self, apply_fn, metrics):
        self._apply_fn = apply_fn
        comm = self.comm
        if comm.world <= 1:
            flat, codec = self.flat, self.codec
            if codec.name == "identity":
                for b in flat.buckets:
                    with metrics.timer("decode_time"):
                        ops.reduce_accum(flat.agg_view(b),
                                         [flat.grad_view(b)],
                                         scale=self.gscale, beta=0.0)
                    with metrics.timer("optim_step_time"):
                        apply_fn(b)
            else:
                # lossy codec at world 1: run the REAL encode->decode wire
                # round trip per bucket (what every worker pays and what the
                # PS applies), so a 1-GPU bench point of a codec config has
                # the codec's cost and its numerics
                if not hasattr(self, "_loop_wire"):
                    self._loop_wire = torch.zeros(
                        self.wire_total, dtype=self.wire_dtype,
                        device=flat.flat_param.device)
                for b, (off, wn) in zip(flat.buckets, self.wseg):
                    seg = self._loop_wire[off:off + wn]
                    with metrics.timer("code_wait"):
                        codec.encode(flat.grad_view(b), seg)
                    with metrics.timer("decode_time"):
                        codec.decode_reduce(flat.agg_view(b), [seg],
                                            gscale=self.gscale, beta=0.0,
                                            src_dtype=flat.dtype)
                    with metrics.timer("optim_step_time"):
                        apply_fn(b)
                metrics.add("packaged_bytes",
                            sum(wn for _, wn in self.wseg)
                            * self.wire_dtype.itemsize)
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
            if self.peers_dropped:
                metrics["peers_dropped"] = self.peers_dropped
        else:
            self.worker_step_exchange(metrics)

    def finish(self, metrics=None, barrier=True):
        metrics = metrics if metrics is not None else StepMetrics()
        comm = self.comm
        if comm.world <= 1:
            return
        self._apply_fn = self._apply_fn or (lambda b: None)
        if comm.is_ps:
            self.serve(metrics)
            for st in self.peers.values():
                if not st.dropped:
                    for rep in st.replies:
                        _wait(rep["reqs"])
            if self.peers_dropped:
                # a dead peer can never reach the barrier — skip it
                return
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
            tail = torch.full((1,), -1, dtype=torch.int64, device=self.device)
            dummy = torch.zeros(self.wire_total, dtype=self.wire_dtype,
                                device=self.device)
            for _ in range(self.ring):
                dist.isend(hdr, dst=comm.ps_rank, group=self.push_g).wait()
                for off, wn in self.wseg:
                    dist.isend(dummy[off:off + wn], dst=comm.ps_rank,
                               group=self.push_g).wait()
                dist.isend(tail, dst=comm.ps_rank, group=self.push_g).wait()
        if barrier and comm.initialized:
            dist.barrier()
