"""PS optimizer API — drop-in `torch.optim.Optimizer` replacements.

API parity with the reference (ps.py:53-261): construct from
`model.named_parameters()`, pluggable gradient codec, `step()` returns
``(loss, metrics_dict)``, SGD/Adam math identical to ps.py:197-214/218-261 —
but the implementation is MI355X-native: flat device buffers, RCCL
collectives, fused HIP update kernels, fp32 master weights under a bf16
model.

Modes:
  "replicated" — the reference's shipped all-gather path (every rank is its
                 own PS; params bitwise identical by determinism).
  "ps"         — the reference's README plan: gather to rank 0, update there,
                 broadcast params.
  "async"      — AsySG-InCon stale-gradient PS (README.md:56-81).

Restrictions vs torch.optim: one param group, uniform parameter dtype
(fp32 or bf16).  lr etc. are re-read from param_groups[0] every apply, so lr
schedulers work unchanged.
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from . import ops
from .codecs import get_codec
from .parallel.async_ps import AsyncPSEngine
from .parallel.comm import Comm
from .parallel.engines import LocalEngine, ReplicatedEngine, SyncPSEngine
from .utils.flat import FlatSpace
from .utils.metrics import StepMetrics


class PS(torch.optim.Optimizer):
    def __init__(self, named_params, defaults, *, code=None, mode="replicated",
                 bucket_mb=50, grad_scale="sum", window=2, max_stale=8,
                 quorum=1, dedicated_ps=False, dtype=None, overlap=True,
                 debug_consistency=0, profile_gpu=False, reply_shards="auto",
                 serve_timeout_s=None):
        named_params = list(named_params)
        if named_params and not isinstance(named_params[0], tuple):
            raise TypeError("pass model.named_parameters(), not parameters()")
        params = [p for _, p in named_params if p.requires_grad]
        super().__init__(params, defaults)
        if len(self.param_groups) != 1:
            raise ValueError("PS supports a single param group")

        self.mode = mode
        self.codec = get_codec(code)
        self.comm = Comm(make_pair_groups=(mode == "async"))
        p0 = params[0]
        self.dtype = dtype or p0.dtype
        bucket_elems = int(bucket_mb * 2 ** 20 / self.dtype.itemsize)
        self.flat = FlatSpace(named_params, bucket_elems=bucket_elems,
                              dtype=self.dtype)

        if grad_scale == "sum":
            gscale = 1.0  # reference semantics: d_p = sum(grads), ps.py:176
        elif grad_scale == "mean":
            gscale = 1.0 / max(1, self.comm.world)
        else:
            gscale = float(grad_scale)
        self.grad_scale = gscale

        if mode == "async":
            self.engine = AsyncPSEngine(self.flat, self.codec, self.comm,
                                        grad_scale=gscale, window=window,
                                        max_stale=max_stale, quorum=quorum,
                                        dedicated=dedicated_ps,
                                        reply_shards=reply_shards,
                                        serve_timeout_s=serve_timeout_s)
        elif self.comm.world <= 1:
            self.engine = LocalEngine(self.flat, self.codec, self.comm, gscale)
        elif mode == "replicated":
            self.engine = ReplicatedEngine(self.flat, self.codec, self.comm,
                                           gscale)
        elif mode in ("ps", "sync_ps"):
            self.engine = SyncPSEngine(self.flat, self.codec, self.comm,
                                       gscale)
        else:
            raise ValueError(f"unknown mode {mode!r}")

        # all ranks start from rank 0's parameters.  The async engine uses
        # per-peer p2p (collective-free control plane); the collective
        # engines use a broadcast.
        if self.comm.initialized:
            if isinstance(self.engine, AsyncPSEngine):
                self.engine.initial_param_sync()
            else:
                dist.broadcast(self.flat.flat_param, src=0)
            self.flat.sync_master_from_param()

        # backward-hook comm overlap (replaces the reference's 200-thread
        # encode pool, ps.py:85,98-101): each bucket's collective launches
        # from autograd as soon as its grads are complete.
        self._hook_handles = []
        if overlap and getattr(self.engine, "wants_hooks", False) \
                and self.comm.world > 1:
            for p in params:
                self._hook_handles.append(
                    p.register_post_accumulate_grad_hook(
                        self.engine.on_param_grad))

        self._step_count = 0
        self._bucket_apply_count = {}
        self.names = [n for n, p in named_params if p.requires_grad]
        # debug: cross-rank param-consistency check every N steps (SURVEY §5
        # race-detection gap; replicated mode must stay bitwise identical)
        self.debug_consistency = int(debug_consistency)
        self.profile_gpu = bool(profile_gpu)
        self._alloc_state()

    # ------------------------------------------------------------------

    def _alloc_state(self):
        raise NotImplementedError

    def _apply_bucket(self, bucket):
        raise NotImplementedError

    def _bump(self, bucket):
        c = self._bucket_apply_count.get(bucket.idx, 0) + 1
        self._bucket_apply_count[bucket.idx] = c
        return c

    def _param_out(self, bucket):
        if self.flat.master is self.flat.flat_param:
            return None
        return self.flat.param_view(bucket)

    # ------------------------------------------------------------------

    def zero_grad(self, set_to_none=False):  # noqa: ARG002 (flat buffers)
        self.flat.zero_grad()
        if self._hook_handles:
            self.engine.start_step()

    def step(self, closure=None, loss=None):
        """Run one exchange+update. Returns (loss, metrics) like ps.py:193."""
        metrics = StepMetrics(gpu=self.profile_gpu)
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        self._step_count += 1
        self.engine.step(self._apply_bucket, metrics)
        metrics.finalize_gpu()
        metrics["step"] = self._step_count
        if (self.debug_consistency and self.comm.initialized
                and self.mode == "replicated"
                and self._step_count % self.debug_consistency == 0):
            cs = [None] * self.comm.world
            dist.all_gather_object(cs, self.flat.param_checksum())
            metrics["consistent"] = len(set(cs)) == 1
            if not metrics["consistent"]:
                raise RuntimeError(
                    f"replicated params diverged at step {self._step_count}: "
                    f"checksums {cs}")
        metrics["wire_codec"] = self.codec.name
        if self.mode == "async" and self.comm.is_ps:
            metrics["staleness_hist"] = dict(
                getattr(self.engine, "staleness_hist", {}))
        return loss, dict(metrics)

    def finish(self, barrier=True):
        """Drain async traffic; call once after the training loop.

        ``barrier=False`` skips the final cross-rank barrier — required when
        a peer may have been dropped (it can never reach the barrier)."""
        if isinstance(self.engine, AsyncPSEngine):
            self.engine.finish(barrier=barrier)
        else:
            self.engine.finish()

    def serve(self):
        """Dedicated-PS event loop (mode='async', dedicated_ps=True)."""
        metrics = StepMetrics()
        self.engine._apply_fn = self._apply_bucket
        self.engine.serve(metrics)
        return dict(metrics)

    # ---------------------------------------------------------------- ckpt

    def state_dict(self):
        flat = self.flat
        sd = {
            "step_count": self._step_count,
            "bucket_apply_count": dict(self._bucket_apply_count),
            "names": list(self.names),
            "offsets": [(n, o, sz) for n, p, o, sz in flat.entries],
            "master": flat.master.detach().cpu().clone(),
            "param_groups": [
                {k: v for k, v in g.items() if k != "params"}
                for g in self.param_groups
            ],
        }
        for k, t in self._state_tensors().items():
            sd[k] = t.detach().cpu().clone()
        return sd

    def load_state_dict(self, sd):
        flat = self.flat
        assert sd["names"] == self.names, "parameter set mismatch"
        self._step_count = sd["step_count"]
        self._bucket_apply_count = {int(k): v for k, v in
                                    sd["bucket_apply_count"].items()}
        flat.master.copy_(sd["master"].to(flat.master.device))
        flat.sync_param_from_master()
        for k, t in self._state_tensors().items():
            t.copy_(sd[k].to(t.device))
        for g, gs in zip(self.param_groups, sd["param_groups"]):
            g.update(gs)

    def _state_tensors(self):
        return {}


class SGD(PS):
    """Reference SGD.optim_step semantics (ps.py:195-214), fused on-device."""

    def __init__(self, named_params, lr=0.01, momentum=0.0, dampening=0.0,
                 weight_decay=0.0, nesterov=False, **ps_kwargs):
        if nesterov and (momentum <= 0 or dampening != 0):
            raise ValueError("nesterov requires momentum and zero dampening")
        defaults = dict(lr=lr, momentum=momentum, dampening=dampening,
                        weight_decay=weight_decay, nesterov=nesterov)
        super().__init__(named_params, defaults, **ps_kwargs)

    def _alloc_state(self):
        g = self.param_groups[0]
        self._mom = None
        if g["momentum"] != 0.0:
            self._mom = torch.zeros_like(self.flat.master)

    def _apply_bucket(self, b):
        g = self.param_groups[0]
        flat = self.flat
        count = self._bump(b)
        buf = self._mom[b.start:b.end] if self._mom is not None else None
        ops.fused_sgd(flat.master_view(b), buf, flat.agg_view(b),
                      self._param_out(b), lr=g["lr"], momentum=g["momentum"],
                      dampening=g["dampening"], wd=g["weight_decay"],
                      nesterov=g["nesterov"], mom_init=(count == 1),
                      gscale=1.0)

    def _state_tensors(self):
        return {"momentum_buffer": self._mom} if self._mom is not None else {}


class Adam(PS):
    """Adam incl. amsgrad (reference Adam.optim_step, ps.py:217-261), fused
    on-device.  Eps placement follows MODERN torch.optim.Adam:
    ``denom = sqrt(v)/sqrt(bias_correction2) + eps``; the reference (like the
    torch of its era) used ``sqrt(v) + eps`` with the correction folded into
    step_size, which scales eps by sqrt(bc2) — a benign O(eps) deviation, but
    not bit-identical to ps.py:257-261.  Matches torch.optim.Adam exactly
    (parity test: tests/test_optim_local.py)."""

    def __init__(self, named_params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0, amsgrad=False, **ps_kwargs):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay, amsgrad=amsgrad)
        super().__init__(named_params, defaults, **ps_kwargs)

    def _alloc_state(self):
        g = self.param_groups[0]
        self._m1 = torch.zeros_like(self.flat.master)
        self._m2 = torch.zeros_like(self.flat.master)
        self._vmax = torch.zeros_like(self.flat.master) if g["amsgrad"] else None

    def _apply_bucket(self, b):
        g = self.param_groups[0]
        flat = self.flat
        count = self._bump(b)
        beta1, beta2 = g["betas"]
        vmax = self._vmax[b.start:b.end] if self._vmax is not None else None
        ops.fused_adam(flat.master_view(b), self._m1[b.start:b.end],
                       self._m2[b.start:b.end], vmax, flat.agg_view(b),
                       self._param_out(b), lr=g["lr"], beta1=beta1,
                       beta2=beta2, eps=g["eps"], wd=g["weight_decay"],
                       step=count, amsgrad=g["amsgrad"], gscale=1.0)

    def _state_tensors(self):
        d = {"exp_avg": self._m1, "exp_avg_sq": self._m2}
        if self._vmax is not None:
            d["max_exp_avg_sq"] = self._vmax
        return d
