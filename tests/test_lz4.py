"""LZ4-style byte codec: CPU spec roundtrip + GPU-vs-reference bit
exactness (the reference's lz4.frame stage, dispatcher.py:81-84, rebuilt
as gfx950 wave-per-block kernels in csrc/lz4.hip)."""

import numpy as np
import pytest
import torch

from defer_amd.ops import codec, lz4_ref


def _cases(rng):
    return {
        "zeros": np.zeros(40000, np.uint8),
        "pattern": np.tile(np.arange(37, dtype=np.uint8), 1000),
        "random": rng.integers(0, 256, 20000).astype(np.uint8),
        "low-entropy": rng.integers(0, 4, 65536).astype(np.uint8),
        "text": np.frombuffer(b"the quick brown fox " * 2048,
                              np.uint8).copy(),
        "tiny": rng.integers(0, 256, 5).astype(np.uint8),
        "partial-block": rng.integers(0, 256, 4096 * 2 + 123).astype(
            np.uint8),
    }


def test_cpu_roundtrip_all_cases():
    rng = np.random.default_rng(7)
    for name, x in _cases(rng).items():
        c = lz4_ref.compress(x)
        y = lz4_ref.decompress(c)
        assert np.array_equal(x, y), name


def test_compressible_data_shrinks():
    x = np.tile(np.arange(16, dtype=np.uint8), 4096)
    c = lz4_ref.compress(x)
    assert c.size < x.size * 0.1


def test_incompressible_bounded_expansion():
    rng = np.random.default_rng(3)
    x = rng.integers(0, 256, 65536).astype(np.uint8)
    c = lz4_ref.compress(x)
    # header + <=2% expansion for random bytes
    assert c.size < x.size * 1.02 + 4 * (2 + 17)


def test_codec_op_cpu_tensor():
    rng = np.random.default_rng(11)
    x = torch.from_numpy(rng.integers(0, 8, 30000).astype(np.uint8))
    c = codec.lz4_compress(x)
    y = codec.lz4_decompress(c, x.numel())
    assert torch.equal(x, y)


@pytest.mark.gpu
def test_gpu_compress_bitexact_vs_reference():
    rng = np.random.default_rng(5)
    for name, x in _cases(rng).items():
        xg = torch.from_numpy(x).cuda()
        cg = codec.lz4_compress(xg).cpu().numpy()
        cr = lz4_ref.compress(x)
        assert cg.size == cr.size, (name, cg.size, cr.size)
        assert np.array_equal(cg, cr), name


@pytest.mark.gpu
def test_gpu_decompress_inverts_cpu_compress():
    rng = np.random.default_rng(9)
    for name, x in _cases(rng).items():
        c = torch.from_numpy(lz4_ref.compress(x).copy()).cuda()
        y = codec.lz4_decompress(c, x.size).cpu().numpy()
        assert np.array_equal(x, y), name


@pytest.mark.gpu
def test_gpu_roundtrip_on_zfp_payload():
    """The production composition: lz4(zfp(activation)) and back."""
    torch.manual_seed(0)
    x = torch.randn(8, 28, 28, 512, device="cuda", dtype=torch.bfloat16)
    rate = 8
    w = codec.zfp_encode(x, rate)
    c = codec.lz4_compress(w)
    w2 = codec.lz4_decompress(c, w.numel())
    assert torch.equal(w, w2)
    y = codec.zfp_decode(w2, x.shape, rate, dtype=torch.bfloat16)
    rel = ((y.float() - x.float()).abs().max()
           / x.float().abs().max()).item()
    assert rel < 0.15, rel


def test_spec_roundtrip_property_sweep():
    """Hypothesis sweep: compress∘decompress is the identity for
    arbitrary byte strings (mixed runs/random segments), and the wire
    never exceeds the declared worst-case bound."""
    from hypothesis import given, settings, strategies as st

    @settings(max_examples=60, deadline=None)
    @given(st.lists(
        st.tuples(st.sampled_from(["run", "rand", "cycle"]),
                  st.integers(1, 3000), st.integers(0, 255)),
        min_size=1, max_size=6), st.integers(0, 2**31 - 1))
    def check(segments, seed):
        rng = np.random.default_rng(seed)
        parts = []
        for kind, n, b in segments:
            if kind == "run":
                parts.append(np.full(n, b, np.uint8))
            elif kind == "cycle":
                parts.append(np.resize(
                    np.arange(b % 7 + 2, dtype=np.uint8), n))
            else:
                parts.append(rng.integers(0, 256, n).astype(np.uint8))
        data = np.concatenate(parts)
        comp = lz4_ref.compress(data)
        # worst-case wire bound (comm.py's VarP2PRing buffer sizing)
        nb = (data.nbytes + lz4_ref.BLK - 1) // lz4_ref.BLK
        assert comp.nbytes <= 4 * (2 + nb + 1) + nb * 4352
        out = lz4_ref.decompress(comp)
        assert np.array_equal(out, data)

    check()


@pytest.mark.gpu
def test_lz4_compress_into_matches_lz4_compress():
    """The async hop encode (lz4_compress_into: preallocated out +
    device-side wire length, comm.VarP2PRing's CUDA path) must produce
    byte-identical streams to the allocating lz4_compress binding."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    import defer_amd.ops as _ops

    m = _ops._load_hip()
    assert m is not None
    torch.manual_seed(11)
    for n in (4096, 40000, 1 << 20, (1 << 20) + 37):
        # compressible: repeated byte runs + random tail
        a = torch.randint(0, 4, (n,), dtype=torch.uint8, device="cuda")
        a[: n // 2] = 7
        ref = m.lz4_compress(a)
        scratch = torch.empty(int(m.lz4_scratch_bytes(n)),
                              dtype=torch.uint8, device="cuda")
        out = torch.empty(ref.numel() + n, dtype=torch.uint8,
                          device="cuda")
        ln = torch.zeros(1, dtype=torch.long, device="cuda")
        m.lz4_compress_into(a, scratch, out, ln)
        k = int(ln.item())
        assert k == ref.numel(), (k, ref.numel())
        assert torch.equal(out[:k], ref)
        # and the stream round-trips
        back = m.lz4_decompress(out[:k].contiguous(), n)
        assert torch.equal(back, a)


@pytest.mark.gpu
def test_var_hop_cuda_encode_roundtrip():
    """The full CUDA zfp+lz4 hop encode sequence (zfp_encode into a
    reused buffer -> lz4_compress_into a ring slot -> device wire
    length), decoded through comm.Codec.decode — the exact kernel
    sequence VarP2PRing.send_encoded issues per item."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    import defer_amd.ops as _ops
    from defer_amd.config import PipelineConfig
    from defer_amd.ops import codec as zc
    from defer_amd.parallel.comm import Codec

    m = _ops._load_hip()
    torch.manual_seed(5)
    shape = (2, 28, 28, 512)
    cfg = PipelineConfig(compression="zfp+lz4", zfp_rate_bits=8)
    c = Codec(cfg, shape, torch.bfloat16, "cuda")
    x = (torch.randn(*shape) * 2).to("cuda", torch.bfloat16)
    zfp_buf = torch.empty(c.zfp_bytes, dtype=torch.uint8, device="cuda")
    zc.zfp_encode(x, c.rate, out=zfp_buf)
    slot = c.alloc_wire()
    scratch = torch.empty(int(m.lz4_scratch_bytes(c.zfp_bytes)),
                          dtype=torch.uint8, device="cuda")
    ln = torch.zeros(1, dtype=torch.long, device="cuda")
    m.lz4_compress_into(zfp_buf, scratch, slot, ln)
    n = int(ln.item())
    assert 0 < n <= c.wire_numel
    y = c.decode(slot[:n])
    assert y.shape == shape and y.dtype == torch.bfloat16
    # lz4 is lossless over zfp: equals the zfp-only roundtrip exactly
    want = zc.zfp_decode(zfp_buf, shape, c.rate, dtype=torch.bfloat16)
    assert torch.equal(y, want)
