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


def choose_hop_modes(stage_us, hop_bytes, link_gbps, dual_boosts=None):
    """Per-hop wire mode for compression="auto": a hop whose raw-bf16
    relay time would exceed the slowest stage's compute becomes the
    pipeline bottleneck — ship THAT hop as fp8 (2x fewer bytes, codec
    ~free at 1.0/2.6 TB/s) and keep every other hop lossless. Pure
    function of (per-stage us, per-hop bytes, link rate): every rank
    computes the same answer, so send/recv codecs always agree.

    stage_us: per-stage compute (us/item); hop_bytes: bf16 bytes/item
    per hop; dual_boosts: per-hop effective-bandwidth multiplier
    (comm.dual_bw_boost). Returns ["none"|"fp8"] per hop."""
    bottleneck = max(stage_us) if stage_us else 0.0
    modes = []
    for i, b in enumerate(hop_bytes):
        boost = dual_boosts[i] if dual_boosts else 1.0
        hop_us = b / (link_gbps * 1e3 * boost)
        modes.append("fp8" if hop_us > bottleneck else "none")
    return modes


class Codec:
    """Encode/decode boundary activations. "none" = raw tensor."""

    def __init__(self, cfg: PipelineConfig, shape, dtype, device,
                 mode: str = None):
        self.mode = mode if mode is not None else cfg.compression
        self.shape = tuple(shape)
        self.dtype = dtype
        self.device = device
        self.rate = cfg.zfp_rate_bits
        self.variable = False
        if self.mode == "none":
            self.wire_numel = int(torch.tensor(self.shape).prod())
            self.wire_dtype = dtype
        elif self.mode == "fp8":
            # cast-only lossy wire (fp8 e4m3, 1 B/value + 4-byte
            # per-tensor scale): halves hop bytes at streaming cost,
            # where the ZFP kernel's ~100-130 GB/s is below xGMI line
            # rate. Values are scaled by amax/448 before the cast
            # (e4m3fn overflows to NaN, and deep-layer activations
            # exceed 448). Shipped as uint8 (p2p backends do not carry
            # float8 dtypes); the scale rides in bytes [0, 4).
            self.wire_numel = int(torch.tensor(self.shape).prod()) + 4
            self.wire_dtype = torch.uint8
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
        if self.mode == "fp8":
            if out is None:
                out = self.alloc_wire()
            if x.is_cuda and x.dtype == torch.bfloat16:
                # fused HIP path (init + atomic amax + cast); wire
                # layout identical to the torch fallback below
                from defer_amd import ops as _ops

                m = _ops._load_hip()
                if m is None:
                    raise RuntimeError("HIP extension missing for fp8")
                m.fp8_encode(x.contiguous(), out)
                return out
            amax = x.detach().abs().amax().float().clamp_min(1e-12)
            # scale via double division then fp32: torch's scalar/tensor
            # 448.0/amax lowers to reciprocal-multiply (1 ulp off true
            # RN); this form is bit-identical to the HIP kernel's
            scale = (448.0 / amax.double()).float()
            q = (x.float() * scale).to(torch.float8_e4m3fn)
            out[:4].copy_(amax.reshape(1).view(torch.uint8))
            out[4:].copy_(q.view(torch.uint8).reshape(-1))
            return out
        from defer_amd.ops import codec as zc
        if self.variable:
            return zc.lz4_compress(zc.zfp_encode(x, self.rate))
        return zc.zfp_encode(x, self.rate, out=out)

    def decode(self, wire: torch.Tensor) -> torch.Tensor:
        if self.mode == "none":
            return wire.view(self.shape)
        if self.mode == "fp8":
            if wire.is_cuda and self.dtype == torch.bfloat16:
                from defer_amd import ops as _ops

                m = _ops._load_hip()
                if m is None:
                    raise RuntimeError("HIP extension missing for fp8")
                return m.fp8_decode(wire.contiguous(),
                                    list(self.shape))
            amax = wire[:4].view(torch.float32)
            vals = wire[4:].view(torch.float8_e4m3fn).to(torch.float32)
            scale = (amax.double() / 448.0).float()
            return (vals * scale).to(self.dtype) \
                .view(self.shape)
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

    def flush(self) -> int:
        """Issue any deferred tail send (variable-size rings); no-op
        for fixed-size rings whose sends are issued eagerly."""
        return 0


class VarP2PRing(P2PRing):
    """Variable-size hop (zfp+lz4): each item is an 8-byte size message
    followed by the exact-size payload — the RCCL analogue of the
    reference's length-prefixed TCP framing (node_state.py:44-54).

    RCCL and gloo match p2p ops strictly in issue order per (src,dst)
    pair (no tags), so BOTH sides follow one interleave for K items:

        s_0, s_1, p_0, s_2, p_1, ..., s_{K-1}, p_{K-2}, p_{K-1}

    (s_k = size of item k, p_k = its exact-size payload). The payload
    send of item k is deferred until item k+1's encode has been issued
    (or flush()): on RCCL the size message is sent straight from device
    memory — lz4_compress_into writes the device int64, no host
    round-trip — and the host reads a pinned copy one item later, so
    the sender's host loop never blocks on the codec kernels of the
    item it is currently issuing (the round-1 per-item .to(kCPU) sync
    serialized the 8-stage compressed config).

    The deferred interleave requires that no rank host-blocks between
    issuing s_k and p_k. True on RCCL (Work.wait is stream-level) but
    NOT on gloo: rank0's host-blocking result wait for item k would
    deadlock against its own deferred p_k. gloo (CPU) therefore uses
    the eager interleave s_0, p_0, s_1, p_1, ... (sizes are known
    synchronously there anyway); both sides of a hop share a backend,
    so the orders always agree. DEFER_AMD_VARRING_DEFER=1 forces the
    deferred order on gloo for protocol tests (only safe without a
    result-return cycle)."""

    def __init__(self, codec: Codec, depth: int):
        # the one-item payload deferral needs >= 2 slots (at depth 1 the
        # undelivered payload's slot would be re-encoded)
        super().__init__(codec, max(depth, 2))
        self.depth = max(depth, 2)
        self.sizes = [torch.zeros(1, dtype=torch.long, device=codec.device)
                      for _ in range(self.depth)]
        self.size_works: List[Optional[dist.Work]] = [None] * self.depth
        self._last_wire = 0
        self.cuda = torch.device(codec.device).type == "cuda"
        import os

        self.defer = self.cuda or \
            os.environ.get("DEFER_AMD_VARRING_DEFER") == "1"
        # sender-side deferral state
        self._pending: Optional[int] = None   # item awaiting payload send
        self._dst: Optional[int] = None
        self._host_n = [0] * self.depth       # CPU path: size known now
        if self.cuda:
            from defer_amd import ops as _ops

            self._hip = _ops._load_hip()
            if self._hip is None:
                raise RuntimeError("HIP extension missing for zfp+lz4 hop")
            zb = codec.zfp_bytes
            self._zfp_buf = torch.empty(zb, dtype=torch.uint8,
                                        device=codec.device)
            self._scratch = torch.empty(
                int(self._hip.lz4_scratch_bytes(zb)), dtype=torch.uint8,
                device=codec.device)
            self._pinned = [torch.zeros(1, dtype=torch.long,
                                        pin_memory=True)
                            for _ in range(self.depth)]
            self._events = [torch.cuda.Event() for _ in range(self.depth)]

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
        n = int(self.sizes[i].item())     # host sync: size arrival
        if self.defer and k + 1 < steps:  # s_{k+1} precedes p_k
            j = (k + 1) % self.depth
            self.size_works[j] = dist.irecv(self.sizes[j], src=src)
        view = self.bufs[i][:n]
        w = dist.irecv(view, src=src)     # exact-size, in-order payload
        w.wait()    # stream-level on RCCL (host moves on); host on gloo
        if not self.defer and k + 1 < steps:  # eager: s_{k+1} after p_k
            j = (k + 1) % self.depth
            self.size_works[j] = dist.irecv(self.sizes[j], src=src)
        self._last_wire = n
        return view

    def repost(self, k: int, steps: int, src: int):
        pass                              # size lookahead in wait_recv

    # -- sender side
    def send_encoded(self, k: int, y: torch.Tensor, dst: int) -> int:
        i = k % self.depth
        self._dst = dst
        self.wait(k)                      # payload slot free (k-depth)
        self._wait_size(k)                # size slot free (k-depth)
        if self.cuda:
            from defer_amd.ops import codec as zc

            zc.zfp_encode(y, self.codec.rate, out=self._zfp_buf)
            self._hip.lz4_compress_into(self._zfp_buf, self._scratch,
                                        self.bufs[i], self.sizes[i])
            self.size_works[i] = dist.isend(self.sizes[i], dst=dst)
            self._pinned[i].copy_(self.sizes[i], non_blocking=True)
            self._events[i].record()
        else:
            wire = self.codec.encode(y)
            n = wire.numel()
            self.bufs[i][:n].copy_(wire)
            self.sizes[i].fill_(n)
            self._host_n[i] = n
            self.size_works[i] = dist.isend(self.sizes[i], dst=dst)
        if not self.defer:                # eager: p_k right after s_k
            self._pending = k
            return self.flush()
        sent = self.flush()               # deferred p_{k-1}
        self._pending = k
        return sent

    def flush(self) -> int:
        """Issue the deferred payload send (item issued last). Called
        between items (p_{k-1} after s_k) and by the pipeline's drain
        for the final item."""
        if self._pending is None:
            return 0
        j = self._pending
        i = j % self.depth
        if self.cuda:
            self._events[i].synchronize()   # encode j + D2H done
            n = int(self._pinned[i].item())
        else:
            n = self._host_n[i]
        self.set_work(j, dist.isend(self.bufs[i][:n], dst=self._dst))
        self._pending = None
        return n


def make_ring(codec: Codec, depth: int) -> P2PRing:
    return VarP2PRing(codec, depth) if codec.variable \
        else P2PRing(codec, depth)


# --------------------------------------------------------------------------
# dual-rail relay: one hop over two xGMI rails
# --------------------------------------------------------------------------

def split_point(numel: int) -> int:
    """Element index splitting a wire buffer into the direct half and the
    forwarded half (64-element aligned when the buffer is big enough for
    alignment to matter)."""
    half = numel // 2
    if numel >= 256:
        half -= half % 64
    return max(half, 1)


def dual_active(cfg: PipelineConfig, world: int) -> bool:
    """Single source of truth for whether dual-rail rings are built:
    needs a third rank to route through and a fixed-size wire (zfp+lz4
    is variable-size). The partitioner's bandwidth credit
    (dual_bw_boost) derives from this so the two can't drift apart."""
    return bool(cfg.dual_rail) and world > 2 \
        and cfg.compression != "zfp+lz4"


def dual_bw_boost(cfg: PipelineConfig, world: int) -> float:
    """Effective hop-bandwidth multiplier the auto-partitioner may
    assume. 2.0 only when EVERY hop runs dual-rail (world >= 4; at
    world == 3 the last hop stays single-rail — see hop_via — so the
    cut chooser stays conservative there)."""
    return 2.0 if dual_active(cfg, world) and world >= 4 else 1.0


def hop_via(hop: int, world: int):
    """Forwarder rank for pipeline hop `hop` (ranks hop -> hop+1), or
    None if the hop stays single-rail.

    In a chain on the fully connected xGMI topology, route the second
    half via hop+2 (links hop->hop+2 and hop+2->hop+1 are idle: the
    first is a distance-2 pair, the second is the reverse direction of
    the next hop). The last hop wraps to rank 0 — its links
    (W-2 -> 0, 0 -> W-1) are also idle for world > 3; at world == 3 the
    wrap would share the directed link 0->2 with hop 0's rail, so the
    last hop stays single-rail there. This assignment uses every
    directed link at most once across all hops (incl. the W-1 -> 0
    result return), so each rail runs at full link rate."""
    if hop + 2 < world:
        return hop + 2
    return 0 if world > 3 else None


class DualRailRing(P2PRing):
    """Fixed-size hop whose wire is split across two rails: bytes
    [0, split) go direct src->dst, bytes [split, end) go src->via->dst
    through a HopForwarder on the via rank. Store-and-forward is at item
    granularity: it adds one pipeline slot of latency but halves the
    steady-state traffic per link, doubling effective hop bandwidth —
    the lever for boundaries over one xGMI link's 153 GB/s (ResNet50
    layer1, profiles/README.md "Predicted pipeline scaling")."""

    def __init__(self, codec: Codec, depth: int, via: int):
        super().__init__(codec, depth)
        self.via = via
        self.split = split_point(codec.wire_numel)
        self.works2: List[Optional[dist.Work]] = [None] * depth

    def _halves(self, k: int):
        b = self.slot(k)
        return b[:self.split], b[self.split:]

    def _wait2(self, k: int):
        w = self.works2[k % self.depth]
        if w is not None:
            w.wait()
            self.works2[k % self.depth] = None

    # -- receiver side
    def prime(self, steps: int, src: int):
        for k in range(min(self.depth, steps)):
            a, b = self._halves(k)
            self.set_work(k, dist.irecv(a, src=src))
            self.works2[k % self.depth] = dist.irecv(b, src=self.via)

    def wait_recv(self, k: int, steps: int, src: int) -> torch.Tensor:
        self.wait(k)
        self._wait2(k)
        return self.slot(k)

    def repost(self, k: int, steps: int, src: int):
        kk = k + self.depth
        if kk < steps:
            a, b = self._halves(kk)
            self.set_work(kk, dist.irecv(a, src=src))
            self.works2[kk % self.depth] = dist.irecv(b, src=self.via)

    # -- sender side
    def send_encoded(self, k: int, y: torch.Tensor, dst: int) -> int:
        self.wait(k)
        self._wait2(k)
        wire = self.codec.encode(y, out=self.slot(k))
        a, b = wire[:self.split], wire[self.split:]
        self.set_work(k, dist.isend(a, dst=dst))
        self.works2[k % self.depth] = dist.isend(b, dst=self.via)
        return wire.numel() * dtype_bytes(wire.dtype)


class HopForwarder:
    """The via rank's relay for one dual-rail hop: receive the second
    half of each item from `src`, re-send it to `dst`, in item order,
    through a `depth`-deep buffer ring.

    Two drive modes, matching the two backends' wait semantics:
    - CUDA/RCCL: Work.wait() is a stream-level wait that never blocks
      the host, so pump() is called inline from the run loop and simply
      enqueues recv -> send chains, ordered by the stream dependency.
    - gloo: Work.wait() blocks the host (and is_completed() never turns
      true for p2p works without it), so the relay runs on its own
      daemon thread doing plain blocking waits. The thread only touches
      the (src -> via) and (via -> dst) pairs, which no other flow uses,
      so it never contends with the owning rank's own hops."""

    def __init__(self, half_numel: int, wire_dtype, device, depth: int,
                 src: int, dst: int):
        self.bufs = [torch.empty(half_numel, dtype=wire_dtype,
                                 device=device)
                     for _ in range(depth)]
        self.depth = depth
        self.src = src
        self.dst = dst
        self.nbytes = half_numel * dtype_bytes(wire_dtype)
        self.inline = (torch.device(device).type == "cuda")
        self.recv_works: List[Optional[dist.Work]] = [None] * depth
        self.send_works: List[Optional[dist.Work]] = [None] * depth
        self.steps = 0
        self.posted = 0
        self.relayed = 0
        self._thread = None
        self._error = None

    def begin(self, steps: int):
        if self.relayed != self.steps:
            # a previous run() aborted mid-stream (feed/collect raised,
            # or drain surfaced a forwarder error): in-flight works on
            # the rails are unrecoverable from here — the pipeline must
            # be rebuilt, not silently restarted on inconsistent rings
            raise RuntimeError(
                f"dual-rail forwarder {self.src}->{self.dst} holds "
                f"{self.steps - self.relayed} undelivered items from an "
                f"aborted run; rebuild the pipeline before running again")
        self.steps = steps
        self.posted = 0
        self.relayed = 0
        if self.inline:
            self.pump()
        else:
            import threading

            self._thread = threading.Thread(target=self._relay_all,
                                            daemon=True)
            self._thread.start()

    # ---- gloo: blocking relay loop on the side thread
    def _relay_all(self):
        try:
            for k in range(min(self.depth, self.steps)):
                self.recv_works[k % self.depth] = dist.irecv(
                    self.bufs[k % self.depth], src=self.src)
                self.posted += 1
            for k in range(self.steps):
                i = k % self.depth
                self.recv_works[i].wait()
                self.recv_works[i] = None
                self.send_works[i] = dist.isend(self.bufs[i],
                                                dst=self.dst)
                self.relayed += 1
                if k + self.depth < self.steps:
                    # slot reuse: item k must be delivered before its
                    # buffer takes item k+depth
                    self.send_works[i].wait()
                    self.send_works[i] = None
                    self.recv_works[i] = dist.irecv(self.bufs[i],
                                                    src=self.src)
                    self.posted += 1
            for i, sw in enumerate(self.send_works):
                if sw is not None:
                    sw.wait()
                    self.send_works[i] = None
        except Exception as e:   # surfaced by drain()
            self._error = e

    # ---- CUDA: stream-ordered inline pump
    def pump(self):
        if not self.inline:
            return
        # enqueue relays for every item; wait() only orders streams
        while self.relayed < self.posted:
            i = self.relayed % self.depth
            self.recv_works[i].wait()
            self.recv_works[i] = None
            self.send_works[i] = dist.isend(self.bufs[i], dst=self.dst)
            self.relayed += 1
        while (self.posted < self.steps
               and self.posted - self.relayed < self.depth):
            i = self.posted % self.depth
            sw = self.send_works[i]
            if sw is not None:
                sw.wait()
                self.send_works[i] = None
            self.recv_works[i] = dist.irecv(self.bufs[i], src=self.src)
            self.posted += 1

    def drain(self):
        if self.inline:
            while self.relayed < self.steps:
                self.pump()
            for i, sw in enumerate(self.send_works):
                if sw is not None:
                    sw.wait()
                    self.send_works[i] = None
        else:
            self._thread.join()
            self._thread = None
            if self._error is not None:
                e, self._error = self._error, None
                raise RuntimeError(
                    f"dual-rail forwarder (hop {self.src}->{self.dst}) "
                    f"failed") from e
