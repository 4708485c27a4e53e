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
    t = torch.from_numpy(np.ascontiguousarray(w))
    if out is not None:
        out.copy_(t)
        return out
    return t


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
