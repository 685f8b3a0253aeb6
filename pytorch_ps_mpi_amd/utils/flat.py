"""Flat device-resident parameter/gradient storage with bucketing.

Replaces the reference's per-parameter host-side byte framing
(mpi_comms.py:186-193 / serialization.py): parameters and gradients live as
views into single flat device buffers, so "packing a message" is free and
every collective moves one large contiguous span.

Layout rules:
  * parameters are ordered in REVERSE registration order (≈ autograd backward
    completion order, the order the reference iterates in ps.py:122-123);
  * every parameter start is aligned to ALIGN elements so bucket spans stay
    16-byte vectorizable for the HIP kernels;
  * buckets are contiguous spans of the flat space (default ~25M elements).

Buffers:
  flat_param  — model dtype (bf16 on GPU benches, f32 on CPU); p.data views
  flat_grad   — model dtype; p.grad views (autograd accumulates in place)
  master      — fp32 master copy (aliases flat_param when model is fp32)
  agg         — fp32 aggregated-gradient buffer (the PS sum target)
"""

from __future__ import annotations

import torch

from .. import ops

ALIGN = 256


class Bucket:
    __slots__ = ("idx", "start", "end", "params", "ready")

    def __init__(self, idx, start, end, params):
        self.idx = idx
        self.start = start
        self.end = end
        self.params = params  # list[(name, param, offset, numel)]
        self.ready = 0

    @property
    def numel(self):
        return self.end - self.start


class FlatSpace:
    def __init__(self, named_params, bucket_elems=25_000_000, device=None,
                 dtype=None):
        named_params = list(named_params)
        if not named_params:
            raise ValueError("no parameters")
        seen = set()
        for name, _ in named_params:
            if name in seen:
                raise ValueError(f"duplicate parameter name {name!r}")
            seen.add(name)
        # reverse registration order ≈ backward completion order
        named_params = [(n, p) for n, p in reversed(named_params)
                        if p.requires_grad]
        p0 = named_params[0][1]
        self.device = torch.device(device) if device is not None else p0.device
        self.dtype = dtype if dtype is not None else p0.dtype
        if self.dtype not in (torch.float32, torch.bfloat16):
            raise ValueError(f"unsupported model dtype {self.dtype}")

        self.entries = []  # (name, param, offset, numel)
        off = 0
        for name, p in named_params:
            n = p.numel()
            self.entries.append((name, p, off, n))
            off += n
            off = (off + ALIGN - 1) // ALIGN * ALIGN
        self.total = off

        dev = self.device
        self.flat_param = torch.zeros(self.total, dtype=self.dtype, device=dev)
        self.flat_grad = torch.zeros(self.total, dtype=self.dtype, device=dev)
        self.agg = torch.zeros(self.total, dtype=torch.float32, device=dev)
        if self.dtype == torch.float32:
            self.master = self.flat_param
        else:
            self.master = torch.zeros(self.total, dtype=torch.float32, device=dev)

        # channels_last conv weights keep their NHWC layout: the flat slice
        # stores NHWC-contiguous bytes and the param view is a permute of it.
        self._chlast = set()
        for name, p, o, n in self.entries:
            if (p.dim() == 4 and
                    p.data.is_contiguous(memory_format=torch.channels_last)
                    and not p.data.is_contiguous()):
                self._chlast.add(name)

        with torch.no_grad():
            for name, p, o, n in self.entries:
                src = p.data.to(device=dev, dtype=self.dtype)
                if name in self._chlast:
                    src = src.permute(0, 2, 3, 1).reshape(-1)
                else:
                    src = src.reshape(-1)
                self.flat_param[o:o + n].copy_(src)
                p.data = self._shaped_view(self.flat_param, name, p.shape, o, n)
            if self.master is not self.flat_param:
                self.master.copy_(self.flat_param.float())
        self.attach_grads()

        # bucketize
        self.buckets = []
        cur = []
        cur_start = 0
        for e in self.entries:
            cur.append(e)
            if e[2] + e[3] - cur_start >= bucket_elems:
                end = (e[2] + e[3] + ALIGN - 1) // ALIGN * ALIGN
                self.buckets.append(Bucket(len(self.buckets), cur_start, end, cur))
                cur = []
                cur_start = end
        if cur:
            self.buckets.append(
                Bucket(len(self.buckets), cur_start, self.total, cur))
        self.param_to_bucket = {}
        for b in self.buckets:
            for name, p, o, n in b.params:
                self.param_to_bucket[p] = b

    # ---- views -----------------------------------------------------------

    def grad_view(self, bucket):
        return self.flat_grad[bucket.start:bucket.end]

    def agg_view(self, bucket):
        return self.agg[bucket.start:bucket.end]

    def master_view(self, bucket):
        return self.master[bucket.start:bucket.end]

    def param_view(self, bucket):
        return self.flat_param[bucket.start:bucket.end]

    # ---- grad management -------------------------------------------------

    def _shaped_view(self, buf, name, shape, o, n):
        if name in self._chlast:
            N, C, H, W = shape
            return buf[o:o + n].view(N, H, W, C).permute(0, 3, 1, 2)
        return buf[o:o + n].view(shape)

    def attach_grads(self):
        for name, p, o, n in self.entries:
            p.grad = self._shaped_view(self.flat_grad, name, p.shape, o, n)

    def zero_grad(self):
        self.flat_grad.zero_()
        # re-attach in case user code dropped the views (set_to_none etc.)
        for name, p, o, n in self.entries:
            if p.grad is None or \
                    p.grad.data_ptr() != self.flat_grad[o:o + n].data_ptr():
                p.grad = self._shaped_view(self.flat_grad, name, p.shape, o, n)

    def sync_param_from_master(self):
        """flat_param (model dtype) <- master (fp32)."""
        if self.master is self.flat_param:
            return
        ops.f32_to_bf16(self.master, self.flat_param)

    def sync_master_from_param(self):
        if self.master is self.flat_param:
            return
        ops.bf16_to_f32(self.flat_param, self.master)

    def param_checksum(self):
        """Debug: cross-rank consistency checksum of the raw param bytes."""
        b = self.flat_param.view(torch.uint8) if self.flat_param.is_cuda \
            else self.flat_param.contiguous().view(torch.uint8)
        return int(b.long().sum().item())
