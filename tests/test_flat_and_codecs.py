import torch

from pytorch_ps_mpi_amd import codecs, models
from pytorch_ps_mpi_amd.utils.flat import ALIGN, FlatSpace


def test_flatspace_views_and_buckets():
    m = models.build_model("mlp")
    flat = FlatSpace(m.named_parameters(), bucket_elems=1000)
    # params are views into flat_param
    for name, p, o, n in flat.entries:
        assert p.data.data_ptr() == flat.flat_param[o:o + n].data_ptr()
        assert o % ALIGN == 0
    assert flat.total % ALIGN == 0
    assert len(flat.buckets) >= 2
    assert flat.buckets[0].start == 0
    assert flat.buckets[-1].end == flat.total
    for a, b in zip(flat.buckets, flat.buckets[1:]):
        assert a.end == b.start
    # grads accumulate into flat_grad
    x, y = models.synthetic_batch("mlp", 4, seed=0)
    loss = models.loss_fn("mlp", m, x, y)
    loss.backward()
    assert flat.flat_grad.abs().sum() > 0
    for name, p, o, n in flat.entries:
        assert p.grad.data_ptr() == flat.flat_grad[o:o + n].data_ptr()
    flat.zero_grad()
    assert flat.flat_grad.abs().sum() == 0


def test_flatspace_reattach_after_set_to_none():
    m = models.build_model("mlp")
    flat = FlatSpace(m.named_parameters(), bucket_elems=1 << 20)
    for p in m.parameters():
        p.grad = None
    flat.zero_grad()
    x, y = models.synthetic_batch("mlp", 4, seed=0)
    models.loss_fn("mlp", m, x, y).backward()
    assert flat.flat_grad.abs().sum() > 0


def _codec_roundtrip(codec, n=4096, tol=None):
    torch.manual_seed(0)
    src = torch.randn(n)
    wn = codec.wire_numel(n, torch.float32) if codec.name == "topk" \
        else codec.wire_numel(n)
    wire = torch.zeros(wn, dtype=codec.wire_dtype(torch.float32))
    codec.encode(src, wire)
    dst = torch.zeros(n)
    codec.decode_reduce(dst, [wire], src_dtype=torch.float32)
    return src, dst


def test_identity_codec():
    c = codecs.Identity()
    src, dst = _codec_roundtrip(c)
    assert torch.allclose(src, dst)


def test_quant8_codec():
    c = codecs.QuantInt8()
    src, dst = _codec_roundtrip(c)
    assert (dst - src).abs().max() < 0.05  # half-step of absmax/127 chunks


def test_topk_codec():
    c = codecs.TopK(density=0.1)
    src, dst = _codec_roundtrip(c)
    k = c.k_for(src.numel())
    nz = (dst != 0).sum().item()
    assert nz == k
    mask = dst != 0
    assert torch.allclose(dst[mask], src[mask])


def test_topk_multi_message_sum():
    c = codecs.TopK(density=0.5)
    n = 1024
    torch.manual_seed(1)
    a, b = torch.randn(n), torch.randn(n)
    wn = c.wire_numel(n, torch.float32)
    wa = torch.zeros(wn, dtype=torch.uint8)
    wb = torch.zeros(wn, dtype=torch.uint8)
    c.encode(a, wa)
    c.encode(b, wb)
    dst = torch.zeros(n)
    c.decode_reduce(dst, [wa, wb], gscale=0.5, src_dtype=torch.float32)
    # where both picked the same index the result is the scaled sum
    k = c.k_for(n)
    assert (dst != 0).sum() >= k  # union of supports


def test_get_codec_spec():
    assert codecs.get_codec(None).name == "identity"
    assert codecs.get_codec("topk:0.05").density == 0.05
    assert codecs.get_codec("quant8").name == "quant8"
    c = codecs.TopK()
    assert codecs.get_codec(c) is c


def test_flatspace_channels_last():
    import torch.nn as nn
    torch.manual_seed(0)
    m = nn.Sequential(nn.Conv2d(3, 8, 3, padding=1), nn.ReLU(),
                      nn.Conv2d(8, 4, 1))
    m = m.to(memory_format=torch.channels_last)
    ref = {k: v.detach().clone() for k, v in m.named_parameters()}
    flat = FlatSpace(m.named_parameters(), bucket_elems=1 << 20)
    for name, p in m.named_parameters():
        assert torch.equal(p.detach(), ref[name]), name
        if p.dim() == 4:
            assert p.is_contiguous(memory_format=torch.channels_last)
    x = torch.randn(2, 3, 8, 8).contiguous(memory_format=torch.channels_last)
    y = m(x).sum()
    y.backward()
    assert flat.flat_grad.abs().sum() > 0
    # grads land in the flat buffer through the permuted views
    for name, p, o, n in flat.entries:
        assert p.grad.data_ptr() == flat.flat_grad[o:o + n].data_ptr()
    # grad values match a plain-layout reference model
    torch.manual_seed(0)
    m2 = nn.Sequential(nn.Conv2d(3, 8, 3, padding=1), nn.ReLU(),
                       nn.Conv2d(8, 4, 1))
    m2(x.contiguous()).sum().backward()
    for (n1, p1), (n2, p2) in zip(m.named_parameters(),
                                  m2.named_parameters()):
        assert torch.allclose(p1.grad, p2.grad, atol=1e-5), n1


def test_fused_ce_cpu_fallback():
    import torch.nn.functional as F
    from pytorch_ps_mpi_amd.ops.ce import fused_cross_entropy
    torch.manual_seed(0)
    logits = torch.randn(16, 64, requires_grad=True)
    targets = torch.randint(0, 64, (16,))
    loss = fused_cross_entropy(logits, targets)
    ref = F.cross_entropy(logits.detach(), targets)
    assert torch.allclose(loss, ref)
    loss.backward()
    assert torch.isfinite(logits.grad).all()


def test_fused_ln_cpu_fallback():
    import torch.nn.functional as F
    from pytorch_ps_mpi_amd.ops.ln import FusedLayerNorm
    torch.manual_seed(0)
    ln = FusedLayerNorm(256)
    x = torch.randn(4, 256, requires_grad=True)
    y = ln(x)
    ref = F.layer_norm(x, (256,), ln.weight, ln.bias, ln.eps)
    assert torch.allclose(y, ref)
    y.sum().backward()
    assert torch.isfinite(x.grad).all()


def test_fused_bn_cpu_fallback_module():
    from pytorch_ps_mpi_amd.ops.bn import FusedBatchNorm2d
    import torch.nn.functional as F
    torch.manual_seed(0)
    bn = FusedBatchNorm2d(8, relu=True)
    ref = torch.nn.BatchNorm2d(8)
    with torch.no_grad():
        ref.weight.copy_(bn.weight)
        ref.bias.copy_(bn.bias)
    x = torch.randn(2, 8, 4, 4)
    y = bn(x)
    assert torch.allclose(y, F.relu(ref(x)), atol=1e-6)


def test_fused_ln_no_bias_fallback():
    from pytorch_ps_mpi_amd.ops.ln import FusedLayerNorm
    ln = FusedLayerNorm(256, bias=False)
    x = torch.randn(4, 256)
    assert not ln._fast_ok(x)
    assert torch.isfinite(ln(x)).all()


def test_topk_density_one_and_tiny():
    c = codecs.TopK(density=1.0)
    n = 40
    src = torch.randn(n)
    wire = torch.zeros(c.wire_numel(n, torch.float32), dtype=torch.uint8)
    c.encode(src, wire)
    dst = torch.zeros(n)
    c.decode_reduce(dst, [wire], src_dtype=torch.float32)
    assert torch.allclose(dst, src, atol=1e-6)
    # min_k clamp: k never exceeds numel
    assert codecs.TopK(density=0.001, min_k=100).k_for(7) == 7


def test_quant8_tiny_chunk():
    c = codecs.QuantInt8()
    n = 13  # < one chunk
    src = torch.randn(n)
    wire = torch.zeros(c.wire_numel(n), dtype=torch.uint8)
    c.encode(src, wire)
    dst = torch.zeros(n)
    c.decode_reduce(dst, [wire])
    assert (dst - src).abs().max() < 0.05


def test_flatspace_skips_frozen_params():
    import torch.nn as nn
    m = nn.Sequential(nn.Linear(8, 8), nn.Linear(8, 4))
    m[0].weight.requires_grad_(False)
    flat = FlatSpace(m.named_parameters(), bucket_elems=1 << 20)
    names = [e[0] for e in flat.entries]
    assert "0.weight" not in names and "1.weight" in names
    y = m(torch.randn(2, 8)).sum()
    y.backward()
    assert flat.flat_grad.abs().sum() > 0


def test_host_codec_adaptive_capacity():
    """A content-adaptive plugin (payload ~ #nonzeros) must not overflow the
    fixed wire capacity: HostCodec sizes capacity from a seeded RANDOM probe
    (a zero-gradient probe would have sized it at the 15 KiB floor and a real
    dense gradient would abort mid-training — advisor round-1 finding)."""
    import numpy as np

    class SparseAdaptive:
        """Stores (idx, val) pairs of nonzero entries — size tracks content."""

        def encode(self, arr):
            nz = np.flatnonzero(arr)
            return (nz.astype(np.int32), arr[nz].astype(np.float32))

        def decode(self, obj):
            idx, val = obj
            n = 20000
            out = np.zeros(n, dtype=np.float32)
            out[idx] = val
            return out

    c = codecs.HostCodec(SparseAdaptive(), headroom=1.5)
    n = 20000
    cap = c.wire_numel(n)
    # capacity reflects a dense payload (8B/elem + pickle framing), not the
    # 15 KiB zero-probe floor
    assert cap > 8 * n
    src = torch.randn(n)  # fully dense — worst case for this plugin
    wire = torch.zeros(cap, dtype=torch.uint8)
    c.encode(src, wire)
    dst = torch.zeros(n)
    c.decode_reduce(dst, [wire], src_dtype=torch.float32)
    assert torch.allclose(src, dst, atol=1e-6)


def test_host_codec_explicit_capacity():
    class Dense:
        def encode(self, arr):
            return arr

        def decode(self, obj):
            return obj

    c = codecs.HostCodec(Dense(), capacity=1 << 20)
    assert c.wire_numel(123) == 1 << 20


def test_topk_threshold_codec_variable_k():
    """Device-side variable-length wire: k_used is data-dependent and rides
    in the wire header; decode touches exactly the used span."""
    c = codecs.TopKThreshold(alpha=0.25, max_density=0.5)
    n = 4096
    torch.manual_seed(3)
    # two magnitude populations: 32 spikes far above the rest
    src = torch.randn(n) * 0.01
    spikes = torch.randperm(n)[:32]
    src[spikes] = torch.sign(torch.randn(32)) * (1.0 + torch.rand(32))
    wn = c.wire_numel(n, torch.float32)
    wire = torch.zeros(wn, dtype=torch.uint8)
    c.encode(src, wire)
    k, hdr, idx, val = c._views(wire, n, torch.float32)
    k_used = int(hdr[0])
    assert 32 <= k_used < k, f"k_used {k_used} should be data-dependent"
    dst = torch.zeros(n)
    c.decode_reduce(dst, [wire], src_dtype=torch.float32)
    nz = dst.nonzero().flatten()
    assert len(nz) == k_used
    assert torch.allclose(dst[nz], src[nz])
    # every spike is included (they are within alpha of the peak)
    assert torch.allclose(dst[spikes], src[spikes])


def test_topk_threshold_kmax_cap():
    c = codecs.TopKThreshold(alpha=1e-6, max_density=0.01)  # selects all -> cap
    n = 10000
    src = torch.randn(n)
    wire = torch.zeros(c.wire_numel(n, torch.float32), dtype=torch.uint8)
    c.encode(src, wire)
    k, hdr, _, _ = c._views(wire, n, torch.float32)
    assert int(hdr[0]) == k == c.kmax_for(n)


def test_get_codec_topkt_spec():
    c = codecs.get_codec("topkt:0.1:0.2")
    assert isinstance(c, codecs.TopKThreshold)
    assert abs(c.alpha - 0.1) < 1e-9 and abs(c.max_density - 0.2) < 1e-9
