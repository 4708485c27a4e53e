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
        self.variable = False
        if self.mode == "none":
            self.wire_numel = int(torch.tensor(self.shape).prod())
            self.wire_dtype = dtype
        elif self.mode == "zfp":
            from defer_amd.ops import codec as zc
            self.wire_numel = zc.zfp_wire_bytes(self.shape, self.rate)
            self.wire_dtype = torch.uint8
        elif self.mode == "zfp+lz4":
            # lz4(zfp(x)) — the reference's full wire codec
            # (dispatcher.py:81-84). LZ4 output is data-dependent, so the
            # wire is variable-size: buffers are worst-case, each hop
            # ships an 8-byte size then the exact payload (the RCCL
            # analogue of the reference's length-prefixed framing,
            # node_state.py:44-54).
            from defer_amd.ops import codec as zc
            from defer_amd.ops import lz4_ref
            self.zfp_bytes = zc.zfp_wire_bytes(self.shape, self.rate)
            nb = (self.zfp_bytes + lz4_ref.BLK - 1) // lz4_ref.BLK
            self.wire_numel = 4 * (2 + nb + 1) + nb * 4352
            self.wire_dtype = torch.uint8
            self.variable = True
        else:
            raise ValueError(f"unknown compression {self.mode!r}")

    def alloc_wire(self) -> torch.Tensor:
        return torch.empty(self.wire_numel, dtype=self.wire_dtype,
                           device=self.device)

    def encode(self, x: torch.Tensor, out: Optional[torch.Tensor] = None):
        if self.mode == "none":
            return x.reshape(-1) if out is None else out.copy_(x.reshape(-1))
        from defer_amd.ops import codec as zc
        if self.variable:
            return zc.lz4_compress(zc.zfp_encode(x, self.rate))
        return zc.zfp_encode(x, self.rate, out=out)

    def decode(self, wire: torch.Tensor) -> torch.Tensor:
        if self.mode == "none":
            return wire.view(self.shape)
        from defer_amd.ops import codec as zc
        if self.variable:
            wire = zc.lz4_decompress(wire, self.zfp_bytes)
        return zc.zfp_decode(wire, self.shape, self.rate,
                             dtype=self.dtype)


def dtype_bytes(dt) -> int:
    import torch as _t

    return _t.empty(0, dtype=dt).element_size()


class P2PRing:
    """Ring of preallocated recv (or send) buffers with outstanding-work
    tracking — the device-resident analogue of the reference's bounded
    Queue(1000) activation buffer (node.py:114). Fixed wire size: recvs
    are pre-posted `depth` deep, no per-item size handshake."""

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

    # -- uniform hop API (pipeline.py drives rings only through these)
    def prime(self, steps: int, src: int):
        for k in range(min(self.depth, steps)):
            self.set_work(k, dist.irecv(self.slot(k), src=src))

    def wait_recv(self, k: int, steps: int, src: int) -> torch.Tensor:
        self.wait(k)
        return self.slot(k)

    def repost(self, k: int, steps: int, src: int):
        """Called after the compute consuming item k was enqueued."""
        if k + self.depth < steps:
            self.set_work(k + self.depth,
                          dist.irecv(self.slot(k + self.depth), src=src))

    def send_encoded(self, k: int, y: torch.Tensor, dst: int) -> int:
        self.wait(k)   # slot free (send k-depth completed)
        wire = self.codec.encode(y, out=self.slot(k))
        self.set_work(k, dist.isend(wire, dst=dst))
        return wire.numel() * dtype_bytes(wire.dtype)


class VarP2PRing(P2PRing):
    """Variable-size hop (zfp+lz4): each item is an 8-byte size message
    followed by the exact-size payload — the RCCL analogue of the
    reference's length-prefixed TCP framing (node_state.py:44-54).

    RCCL p2p matches send/recv strictly in issue order per (src,dst)
    pair (no tags), so the receiver must alternate size/payload recvs in
    the same order the sender issues them; size recvs are posted one
    item ahead (right after the previous payload recv), never `depth`
    ahead."""

    def __init__(self, codec: Codec, depth: int):
        super().__init__(codec, depth)
        self.sizes = [torch.zeros(1, dtype=torch.long, device=codec.device)
                      for _ in range(depth)]
        self.size_works: List[Optional[dist.Work]] = [None] * depth
        self._last_wire = 0

    def _wait_size(self, k: int):
        w = self.size_works[k % self.depth]
        if w is not None:
            w.wait()
            self.size_works[k % self.depth] = None

    # -- receiver side
    def prime(self, steps: int, src: int):
        if steps > 0:
            self.size_works[0] = dist.irecv(self.sizes[0], src=src)

    def wait_recv(self, k: int, steps: int, src: int) -> torch.Tensor:
        i = k % self.depth
        self._wait_size(k)
        n = int(self.sizes[i].item())
        view = self.bufs[i][:n]
        dist.recv(view, src=src)          # exact-size, in-order payload
        if k + 1 < steps:                 # lookahead: next item's size
            j = (k + 1) % self.depth
            self.size_works[j] = dist.irecv(self.sizes[j], src=src)
        self._last_wire = n
        return view

    def repost(self, k: int, steps: int, src: int):
        pass                              # size lookahead in wait_recv

    # -- sender side
    def send_encoded(self, k: int, y: torch.Tensor, dst: int) -> int:
        i = k % self.depth
        self.wait(k)                      # payload slot free
        self._wait_size(k)
        wire = self.codec.encode(y)
        n = wire.numel()
        self.sizes[i].fill_(n)
        self.size_works[i] = dist.isend(self.sizes[i], dst=dst)
        buf = self.bufs[i][:n]
        buf.copy_(wire)
        self.set_work(k, dist.isend(buf, dst=dst))
        return n


def make_ring(codec: Codec, depth: int) -> P2PRing:
    return VarP2PRing(codec, depth) if codec.variable \
        else P2PRing(codec, depth)
