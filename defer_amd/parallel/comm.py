"""Inter-stage relay: p2p send/recv with optional activation codec.

The reference ships boundary activations node-to-node over framed TCP with
lz4(zfp(x)) compression (node.py:107-108, node_state.py:43-101). Here a hop
is an RCCL point-to-point send/recv over a dedicated xGMI link
(torch.distributed, backend "nccl" = RCCL on ROCm), optionally preceded by
the GPU ZFP(+LZ4) codec. Fixed-rate ZFP means fixed message sizes, so recv
buffers are preallocated rings — no size handshake per item (the
reference's 8-byte length prefix, node_state.py:44-54, is not needed).
"""

from typing import List, Optional

import torch
import torch.distributed as dist

from defer_amd.config import PipelineConfig


class Codec:
    """Encode/decode boundary activations. "none" = raw tensor."""

    def __init__(self, cfg: PipelineConfig, shape, dtype, device):
        self.mode = cfg.compression
        self.shape = tuple(shape)
        self.dtype = dtype
        self.device = device
        self.rate = cfg.zfp_rate_bits
        if self.mode == "none":
            self.wire_numel = int(torch.tensor(self.shape).prod())
            self.wire_dtype = dtype
        elif self.mode in ("zfp", "zfp+lz4"):
            from defer_amd.ops import codec as zc
            self.wire_numel = zc.zfp_wire_bytes(self.shape, self.rate)
            self.wire_dtype = torch.uint8
        else:
            raise ValueError(f"unknown compression {self.mode!r}")

    def alloc_wire(self) -> torch.Tensor:
        return torch.empty(self.wire_numel, dtype=self.wire_dtype,
                           device=self.device)

    def encode(self, x: torch.Tensor, out: Optional[torch.Tensor] = None):
        if self.mode == "none":
            return x.reshape(-1) if out is None else out.copy_(x.reshape(-1))
        from defer_amd.ops import codec as zc
        return zc.zfp_encode(x, self.rate, out=out)

    def decode(self, wire: torch.Tensor) -> torch.Tensor:
        if self.mode == "none":
            return wire.view(self.shape)
        from defer_amd.ops import codec as zc
        return zc.zfp_decode(wire, self.shape, self.rate,
                             dtype=self.dtype)


def dtype_bytes(dt) -> int:
    import torch as _t

    return _t.empty(0, dtype=dt).element_size()


class P2PRing:
    """Ring of preallocated recv (or send) buffers with outstanding-work
    tracking — the device-resident analogue of the reference's bounded
    Queue(1000) activation buffer (node.py:114)."""

    def __init__(self, codec: Codec, depth: int):
        self.codec = codec
        self.depth = depth
        self.bufs: List[torch.Tensor] = [codec.alloc_wire()
                                         for _ in range(depth)]
        self.works: List[Optional[dist.Work]] = [None] * depth

    def slot(self, k: int):
        return self.bufs[k % self.depth]

    def wait(self, k: int):
        w = self.works[k % self.depth]
        if w is not None:
            w.wait()
            self.works[k % self.depth] = None

    def set_work(self, k: int, w):
        self.works[k % self.depth] = w
