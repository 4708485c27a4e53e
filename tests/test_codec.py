"""Fixed-rate ZFP codec: reference roundtrip properties (CPU) and
GPU-kernel bit-exactness against the numpy spec."""

import numpy as np
import pytest
import torch

from defer_amd.ops import codec, zfp_ref


def test_roundtrip_error_decreases_with_rate():
    rng = np.random.default_rng(0)
    a = rng.standard_normal((8, 12, 16)).astype(np.float32) * 2
    errs = []
    for rate in (4, 8, 12, 16):
        w = zfp_ref.encode(a, rate)
        assert w.nbytes == zfp_ref.wire_bytes(a.shape, rate)
        b = zfp_ref.decode(w, a.shape, rate)
        errs.append(np.abs(a - b).max())
    assert errs[0] > errs[1] > errs[2] > errs[3]
    assert errs[2] / np.abs(a).max() < 0.02   # rate 12: <2% max rel err


def test_zero_block_exact():
    z = np.zeros((4, 4, 8), np.float32)
    w = zfp_ref.encode(z, 8)
    assert np.all(w == 0)
    assert np.all(zfp_ref.decode(w, z.shape, 8) == 0)


def test_smooth_data_compresses_well():
    x = np.linspace(0, 1, 4096, dtype=np.float32).reshape(16, 16, 16)
    y = zfp_ref.decode(zfp_ref.encode(x, 4), x.shape, 4)
    # 8x compression vs f32 on smooth data: tight reconstruction
    assert np.abs(x - y).max() < 2e-3


def test_edge_clamped_partial_blocks():
    rng = np.random.default_rng(1)
    a = rng.standard_normal((7, 7, 24)).astype(np.float32)
    w = zfp_ref.encode(a, 12)
    b = zfp_ref.decode(w, a.shape, 12)
    assert np.abs(a - b).max() / np.abs(a).max() < 0.02


def test_codec_op_cpu_tensor():
    x = torch.randn(4, 8, 16)
    w = codec.zfp_encode(x, 8)
    assert w.dtype == torch.uint8
    assert w.numel() == codec.zfp_wire_bytes(x.shape, 8)
    y = codec.zfp_decode(w, x.shape, 8)
    assert (x - y).abs().max() / x.abs().max() < 0.2


@pytest.mark.gpu
def test_gpu_encode_bitexact_vs_reference():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    rng = np.random.default_rng(2)
    for shape, rate in [((8, 16, 32), 8), ((2, 6, 10, 24), 12),
                        ((64, 1000), 8), ((5, 5, 7), 6),
                        # panel-staged path (b2 % 16 == 0, C >= 64):
                        # dense-row LDS staging must stay bit-exact,
                        # incl. d0/d1 edge clamping and d2 tails
                        ((4, 16, 64), 8), ((2, 8, 8, 256), 8),
                        ((3, 7, 5, 64), 10), ((2, 6, 6, 253), 12)]:
        a = (rng.standard_normal(shape) * 3).astype(np.float32)
        want = zfp_ref.encode(a, rate)
        xg = torch.from_numpy(a).cuda()
        got = codec.zfp_encode(xg, rate).cpu().numpy()
        assert got.shape == want.shape
        assert np.array_equal(got, want), (
            f"wire mismatch {shape} rate={rate}: "
            f"first diff {np.nonzero(got != want)[0][:5]}")
        # decode path: GPU decode == reference decode, bitwise (f32)
        dec_ref = zfp_ref.decode(want, shape, rate)
        dec_gpu = codec.zfp_decode(torch.from_numpy(want).cuda(), shape,
                                   rate).cpu().numpy()
        assert np.array_equal(dec_gpu, dec_ref)


@pytest.mark.gpu
def test_gpu_roundtrip_bf16():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    x = (torch.randn(2, 28, 28, 512) * 2).bfloat16().cuda()
    w = codec.zfp_encode(x, 8)
    y = codec.zfp_decode(w, x.shape, 8, dtype=torch.bfloat16)
    rel = (x.float() - y.float()).abs().max() / x.float().abs().max()
    assert rel < 0.15
    assert w.numel() * 1.0 / (x.numel() * 2) < 0.51  # >=2x vs bf16


def test_spec_properties_random_shapes():
    """Property sweep of the numpy spec: for random field shapes and
    scales, the fixed-rate stream has the exact declared size, decode
    never produces non-finite values, and the error bound tightens with
    rate (hypothesis-driven shapes)."""
    from hypothesis import given, settings, strategies as st

    @settings(max_examples=40, deadline=None)
    @given(st.integers(1, 9), st.integers(1, 9), st.integers(1, 17),
           st.floats(1e-3, 1e3), st.integers(0, 2**31 - 1))
    def check(d0, d1, d2, scale, seed):
        rng = np.random.default_rng(seed)
        a = (rng.standard_normal((d0, d1, d2)) * scale).astype(np.float32)
        prev = None
        for rate in (6, 12):
            w = zfp_ref.encode(a, rate)
            assert w.nbytes == zfp_ref.wire_bytes(a.shape, rate)
            b = zfp_ref.decode(w, a.shape, rate)
            assert np.isfinite(b).all()
            err = float(np.abs(a - b).max())
            if prev is not None:
                assert err <= prev + 1e-6
            prev = err
        # rate 12 keeps max rel error small on smooth-scaled data
        assert prev <= max(1e-6, 0.05 * float(np.abs(a).max()))

    check()


def test_fp8_wire_codec_roundtrip():
    """comm.Codec "fp8": fixed-size uint8 wire, decode(encode(x)) within
    e4m3 quantization error for activation-range data."""
    import torch

    from defer_amd.config import PipelineConfig
    from defer_amd.parallel.comm import Codec

    shape = (2, 7, 7, 32)
    cfg = PipelineConfig(compression="fp8")
    c = Codec(cfg, shape, torch.float32, "cpu")
    assert c.wire_numel == 2 * 7 * 7 * 32 + 4   # +4: per-tensor scale
    assert c.wire_dtype == torch.uint8 and not c.variable
    torch.manual_seed(3)
    x = torch.randn(*shape).clamp(-8, 8)
    out = c.alloc_wire()
    wire = c.encode(x, out=out)
    assert wire.dtype == torch.uint8
    y = c.decode(wire)
    assert y.shape == x.shape
    mask = x.abs() > 0.05          # skip the subnormal tail
    rel = ((y - x).abs() / x.abs().clamp_min(1e-6))[mask]
    assert rel.max() <= 0.0667, rel.max()   # e4m3: 3 mantissa bits

    # amax scaling: values past e4m3's 448 max must survive (the
    # unscaled cast overflows them to NaN)
    big = torch.randn(*shape) * 500
    yb = c.decode(c.encode(big, out=c.alloc_wire()))
    assert torch.isfinite(yb).all()
    mask = big.abs() > 1.0
    rel = ((yb - big).abs() / big.abs())[mask]
    assert rel.max() <= 0.0667, rel.max()

    # all-zero input round-trips to zeros
    z = c.decode(c.encode(torch.zeros(*shape), out=c.alloc_wire()))
    assert torch.equal(z, torch.zeros(*shape))


def test_codec_sizing_on_all_real_boundary_shapes():
    """Every codec mode must size a wire for every boundary shape the
    four models actually produce under auto/defer8 cuts (the zfp block
    math and the lz4 worst-case bound run on real NHWC shapes, incl.
    odd spatial sizes) — a sizing error here would mismatch RCCL
    send/recv sizes and hang the round-end multi-GPU runs."""
    import torch

    from defer_amd.config import PipelineConfig
    from defer_amd.models import DEFER_8STAGE_CUTS, MODELS
    from defer_amd.parallel.comm import Codec
    from defer_amd.parallel.partitioner import (auto_partition,
                                                partition_model)

    B = 3
    for name, ctor in MODELS.items():
        model = ctor()
        cut_sets = [auto_partition(model, 4)[0]]
        if name == "resnet50":
            cut_sets.append(DEFER_8STAGE_CUTS)
        for cuts in cut_sets:
            stages = partition_model(model, cuts)
            # chain shapes via batch-1 trace
            shape = (1, 224, 224, 3)
            outs = []
            for s in stages:
                x = torch.zeros(*shape)
                with torch.no_grad():
                    y = s.graph.forward(x)
                outs.append((B,) + tuple(y.shape[1:]))
                shape = tuple(y.shape)
            for mode in ("none", "fp8", "zfp", "zfp+lz4"):
                cfg = PipelineConfig(compression=mode)
                for i in range(len(stages) - 1):
                    c = Codec(cfg, outs[i], torch.bfloat16, "cpu")
                    numel = 1
                    for d in outs[i]:
                        numel *= d
                    assert c.wire_numel > 0
                    if mode == "none":
                        assert c.wire_numel == numel
                        assert c.wire_dtype == torch.bfloat16
                    elif mode == "fp8":
                        assert c.wire_numel == numel + 4
                    elif mode == "zfp":
                        # fixed-rate: rate bits per value, padded to
                        # whole 4x4x4 blocks
                        assert c.wire_numel >= numel * cfg.zfp_rate_bits // 8
                    else:
                        assert c.variable
                        assert c.wire_numel > numel * cfg.zfp_rate_bits // 8


def test_fp8_wire_codec_property_sweep():
    """fp8 codec across random shapes and dynamic ranges: finite
    output, e4m3 relative error bound above the scaled-subnormal floor,
    exact zeros."""
    import torch
    from hypothesis import given, settings, strategies as st

    from defer_amd.config import PipelineConfig
    from defer_amd.parallel.comm import Codec

    @settings(max_examples=40, deadline=None)
    @given(st.integers(1, 4), st.integers(1, 9), st.integers(1, 9),
           st.integers(1, 19),
           st.floats(min_value=-3.0, max_value=3.0),
           st.integers(0, 2**31 - 1))
    def check(b, h, w, c, log_scale, seed):
        shape = (b, h, w, c)
        torch.manual_seed(seed)
        x = torch.randn(*shape) * (10.0 ** log_scale)
        codec = Codec(PipelineConfig(compression="fp8"), shape,
                      torch.float32, "cpu")
        y = codec.decode(codec.encode(x, out=codec.alloc_wire()))
        assert y.shape == x.shape
        assert torch.isfinite(y).all()
        amax = x.abs().max()
        mask = x.abs() > amax * 2 ** -6   # above the scaled denormal floor
        if mask.any():
            rel = ((y - x).abs() / x.abs())[mask]
            assert rel.max() <= 0.0667, float(rel.max())
        assert torch.equal(y == 0, x == 0) or (x == 0).sum() == 0

    check()


@pytest.mark.gpu
def test_fp8_wire_hip_matches_torch_cast():
    """The fused HIP fp8 wire (csrc/codec.hip fp8_encode/decode) must be
    bit-identical to the torch-fallback wire in parallel/comm.py: same
    amax bytes, same e4m3fn payload (RNE + saturate-to-448 + subnormals),
    and the decode must reproduce torch's dequantization in bf16."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    import defer_amd.ops as _ops

    m = _ops._load_hip()
    assert m is not None
    torch.manual_seed(3)
    cases = [
        torch.randn(4096) * 10,
        torch.randn(2, 56, 56, 64) * 0.01,          # subnormal-heavy
        torch.linspace(-600, 600, 8192),            # saturation range
        torch.zeros(256),
        torch.full((64,), 4.375e-3),                # near min-normal/8
    ]
    for x in cases:
        xg = x.to("cuda", torch.bfloat16).contiguous()
        n = xg.numel()
        out = torch.empty(n + 4, dtype=torch.uint8, device="cuda")
        m.fp8_encode(xg, out)
        # torch-fallback wire (double-division scale, see comm.Codec)
        amax = xg.detach().abs().amax().float().clamp_min(1e-12)
        q = (xg.float() * (448.0 / amax.double()).float()) \
            .to(torch.float8_e4m3fn)
        want = torch.empty_like(out)
        want[:4] = amax.reshape(1).view(torch.uint8)
        want[4:] = q.view(torch.uint8).reshape(-1)
        mism = (out != want).sum().item()
        assert mism == 0, (
            f"{mism}/{n} byte mismatches; first "
            f"{(out != want).nonzero()[:5].flatten().tolist()}")
        dec = m.fp8_decode(out, list(xg.shape))
        vals = want[4:].view(torch.float8_e4m3fn).to(torch.float32)
        want_dec = (vals * (amax.double() / 448.0).float()) \
            .to(torch.bfloat16).view(xg.shape)
        assert torch.equal(dec, want_dec)


@pytest.mark.gpu
def test_var_ring_cuda_sender_bookkeeping(monkeypatch):
    """VarP2PRing's CUDA sender (deferred payload, device-side size
    message, pinned readback one item later) without a peer: stub
    dist.isend and verify the issued sequence s_0,s_1,p_0,s_2,p_1,...
    with correct device-computed lengths and decodable payloads."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    import torch.distributed as dist

    from defer_amd.config import PipelineConfig
    from defer_amd.parallel.comm import Codec, VarP2PRing

    sent = []

    class _W:
        def wait(self):
            pass

    def fake_isend(t, dst=None, **kw):
        torch.cuda.synchronize()          # value must be final by now
        sent.append(t.detach().clone())
        return _W()

    monkeypatch.setattr(dist, "isend", fake_isend)

    shape = (2, 14, 14, 256)
    cfg = PipelineConfig(compression="zfp+lz4", zfp_rate_bits=8,
                         ring_depth=2)
    c = Codec(cfg, shape, torch.bfloat16, "cuda")
    ring = VarP2PRing(c, cfg.ring_depth)
    assert ring.defer and ring.cuda
    torch.manual_seed(9)
    xs = [(torch.randn(*shape) * 2).to("cuda", torch.bfloat16)
          for _ in range(4)]
    for k, x in enumerate(xs):
        ring.send_encoded(k, x, dst=1)
    ring.flush()
    # sequence: s0, s1, p0, s2, p1, s3, p2, p3
    kinds = ["s", "s", "p", "s", "p", "s", "p", "p"]
    items = [0, 1, 0, 2, 1, 3, 2, 3]
    assert len(sent) == len(kinds)
    from defer_amd.ops import codec as zc

    sizes = {}
    for t, kind, it in zip(sent, kinds, items):
        if kind == "s":
            assert t.dtype == torch.long and t.numel() == 1
            sizes[it] = int(t.item())
        else:
            n = sizes[it]
            assert t.numel() == n, (it, t.numel(), n)
            want = zc.zfp_decode(zc.zfp_encode(xs[it], c.rate),
                                 shape, c.rate, dtype=torch.bfloat16)
            got = c.decode(t)
            assert torch.equal(got, want), f"item {it} corrupted"
