"""Activation codec ops: fixed-rate ZFP (GPU kernels; numpy reference on
CPU). Wire format is fixed-size per tensor shape+rate, so pipeline recv
buffers are preallocated rings (comm.py)."""

import numpy as np
import torch

from defer_amd.ops import zfp_ref


def zfp_wire_bytes(shape, rate: int) -> int:
    return zfp_ref.wire_bytes(tuple(shape), rate)


def zfp_encode(x: torch.Tensor, rate: int, out=None) -> torch.Tensor:
    if x.is_cuda:
        from defer_amd import ops as _ops

        m = _ops._load_hip()
        if m is None:
            raise RuntimeError("HIP extension missing for zfp_encode")
        return m.zfp_encode(x.contiguous(), rate, out)
    w = zfp_ref.encode(x.float().numpy(), rate)
    t = torch.from_numpy(np.ascontiguousarray(w).copy())
    if out is not None:
        out.copy_(t)
        return out
    return t


def lz4_compress(x: torch.Tensor) -> torch.Tensor:
    """x: uint8 tensor -> uint8 stream (variable length). GPU tensors use
    the gfx950 HIP kernels (csrc/lz4.hip); CPU uses the bit-exact
    reference (ops/lz4_ref.py)."""
    if x.is_cuda:
        from defer_amd import ops as _ops

        m = _ops._load_hip()
        if m is None:
            raise RuntimeError("HIP extension missing for lz4_compress")
        return m.lz4_compress(x.contiguous().view(-1))
    from defer_amd.ops import lz4_ref

    return torch.from_numpy(
        lz4_ref.compress(x.contiguous().view(-1).numpy()).copy())


def lz4_decompress(comp: torch.Tensor, raw_len: int) -> torch.Tensor:
    if comp.is_cuda:
        from defer_amd import ops as _ops

        m = _ops._load_hip()
        if m is None:
            raise RuntimeError("HIP extension missing for lz4_decompress")
        return m.lz4_decompress(comp.contiguous().view(-1), raw_len)
    from defer_amd.ops import lz4_ref

    out = lz4_ref.decompress(comp.contiguous().view(-1).numpy())
    assert out.size == raw_len, (out.size, raw_len)
    return torch.from_numpy(out.copy())


def zfp_decode(wire: torch.Tensor, shape, rate: int,
               dtype=torch.float32) -> torch.Tensor:
    if wire.is_cuda:
        from defer_amd import ops as _ops

        m = _ops._load_hip()
        if m is None:
            raise RuntimeError("HIP extension missing for zfp_decode")
        return m.zfp_decode(wire, list(shape), rate,
                            dtype == torch.bfloat16)
    a = zfp_ref.decode(wire.numpy(), tuple(shape), rate)
    return torch.from_numpy(a).to(dtype)
