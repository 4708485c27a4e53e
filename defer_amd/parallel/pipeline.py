"""Pipeline runtime: stage executors, the DEFER orchestrator, and the
distributed (one process per GPU, RCCL/xGMI) pipeline.

Reference behavior being rebuilt:
  - `DEFER.run_defer(model, partition_layers, input_stream, output_stream)`
    partitions the model, distributes stages to compute nodes, feeds inputs
    and serves results (dispatcher.py:107-115). Here compute nodes are GPUs
    (threads+streams in-process, or ranks of a torch.distributed job).
  - Each compute node runs rx -> compute -> tx with bounded queues
    (node.py:80-108,114). Here: device-resident rings + RCCL p2p on the
    NCCL side stream, overlapping the compute stream.
  - Readiness uses real events/barriers, not sleeps (fixes reference bug
    B4: dispatcher.py:112, node.py:32-33,95-96).
"""

import queue
import threading
import time
from dataclasses import dataclass
from typing import Callable, List, Optional

import torch
import torch.distributed as dist

from defer_amd.config import PipelineConfig
from defer_amd.graph import GraphModel
from defer_amd.parallel.comm import (Codec, DualRailRing, HopForwarder,
                                     P2PRing, dtype_bytes, dual_active,
                                     dual_bw_boost, hop_via, make_ring,
                                     split_point)
from defer_amd.parallel.partitioner import (as_graph_model, auto_partition,
                                            partition_model)


# --------------------------------------------------------------------------
# stage execution
# --------------------------------------------------------------------------

@dataclass
class StageStats:
    items: int = 0
    images: int = 0
    compute_s: float = 0.0    # host wall (threaded DEFER orchestrator)
    compute_ms: float = 0.0   # device time, hipEvent pairs (DistPipeline)
    bytes_in: int = 0
    bytes_out: int = 0
    bytes_fwd: int = 0        # dual-rail: bytes relayed for another hop


class StageExecutor:
    """Runs one stage model on one device (the reference Node's
    model.predict hot loop, node.py:103-108, minus the sockets).

    Weights are pre-cast: conv/dense weights to the compute dtype, folded
    BN scale/bias kept fp32 for the HIP kernel epilogues. Optionally
    captures the stage forward into a hipGraph (torch.cuda.CUDAGraph) to
    amortize launch overhead for small batches; captures are serialized
    process-wide (hipGraph capture is not safe concurrently with other
    threads' work on the shared legacy stream).
    """

    _capture_lock = threading.Lock()

    def __init__(self, stage: GraphModel, device, dtype: torch.dtype,
                 use_graph: bool = False, fuse: bool = True):
        if fuse:
            from defer_amd.parallel.fusion import fuse_residual_adds

            stage = GraphModel(fuse_residual_adds(stage.graph),
                               name=stage.model_name)
        self.model = stage
        self.device = torch.device(device)
        self.dtype = dtype
        self.use_graph = use_graph and self.device.type == "cuda"
        stage.to(self.device)
        if self.device.type == "cuda":
            for p in stage.parameters():
                if p.dim() >= 2:  # conv RSCK / dense weights
                    p.data = p.data.to(dtype).contiguous()
                else:             # BN scale/bias, conv bias: fp32 epilogue
                    p.data = p.data.float()
        self._graph = None
        self._static_in = None
        self._static_out = None

    def _capture(self, x: torch.Tensor):
        self._static_in = x.clone()
        torch.cuda.synchronize()
        # warm once on a side stream (allocator warm-up) before capture
        s = torch.cuda.Stream()
        with torch.cuda.stream(s):
            self.model(self._static_in)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._static_out = self.model(self._static_in)
        self._graph = g

    def run(self, x: torch.Tensor) -> torch.Tensor:
        if not self.use_graph:
            with torch.no_grad():
                return self.model(x)
        if self._graph is None:
            try:
                with StageExecutor._capture_lock:
                    self._capture(x)
            except Exception:
                # capture not supported for this stage: eager fallback
                self.use_graph = False
                with torch.no_grad():
                    return self.model(x)
        self._static_in.copy_(x)
        self._graph.replay()
        return self._static_out

    def output_shape(self, input_shape) -> tuple:
        was = self.use_graph
        self.use_graph = False
        try:
            with torch.no_grad():
                x = torch.zeros(*input_shape, device=self.device,
                                dtype=self.dtype
                                if self.device.type == "cuda"
                                else torch.float32)
                return tuple(self.run(x).shape)
        finally:
            self.use_graph = was


# --------------------------------------------------------------------------
# in-process DEFER orchestrator (API parity; threads + queues + streams)
# --------------------------------------------------------------------------

class DEFER:
    """API-shape parity with the reference orchestrator
    (dispatcher.py:21,107): compute nodes are torch devices ("cuda:0",
    "cuda:1", ..., or "cpu" for the plumbing config); stages are mapped
    one-to-one onto them and chained by device-to-device copies (xGMI
    peer copies on GPU). `None` on the input stream shuts the pipeline
    down cleanly and run_defer returns after draining (the reference runs
    forever, dispatcher.py:115)."""

    def __init__(self, computeNodes: List, config: Optional[PipelineConfig] = None):
        self.computeNodes = list(computeNodes)
        self.cfg = config or PipelineConfig()
        self.stats: List[StageStats] = []

    def run_defer(self, model, partition_layers: Optional[List[str]],
                  input_stream: "queue.Queue", output_stream: "queue.Queue"):
        cfg = self.cfg
        n = len(self.computeNodes)
        if partition_layers is None and cfg.weights_dir and n > 1:
            # partition exactly as the checkpoint was partitioned (same
            # contract as DistPipeline): auto cuts that differ from the
            # saved cuts would otherwise fail load with a raw
            # state_dict key mismatch
            from defer_amd import checkpoint

            mani = checkpoint.load_manifest(cfg.weights_dir)
            if mani["num_stages"] != n:
                raise ValueError(
                    f"checkpoint has {mani['num_stages']} stages but "
                    f"{n} compute nodes were given")
            partition_layers = list(mani["cut_points"])
        if partition_layers is None:
            from defer_amd.parallel.calibrate import find_calibration

            gm0 = as_graph_model(model)
            partition_layers, stages = auto_partition(
                gm0, n,
                input_shape=tuple(cfg.input_shape)
                if cfg.input_shape else (1, 224, 224, 3),
                measured_us=find_calibration(gm0.model_name,
                                             cfg.calibration_file))
        else:
            stages = partition_model(model, partition_layers)
        if len(stages) != n:
            raise ValueError(
                f"{len(stages)} stages for {n} compute nodes")
        if cfg.partition_dump_dir:
            from defer_amd.utils.visualize import dump_partition

            dump_partition(stages, cfg.partition_dump_dir,
                           cut_points=partition_layers)
        if cfg.weights_dir:
            from defer_amd import checkpoint

            for i, s in enumerate(stages):
                checkpoint.load_stage(s, cfg.weights_dir, i)
        dtype = cfg.torch_dtype()
        execs = [StageExecutor(s, dev, dtype, cfg.use_hip_graphs)
                 for s, dev in zip(stages, self.computeNodes)]
        self.stats = [StageStats() for _ in range(n)]
        self.errors: List = [None] * n
        qs = [input_stream] + [queue.Queue(cfg.ring_depth)
                               for _ in range(n)]
        ready = [threading.Event() for _ in range(n)]

        def worker(i: int):
            ex = execs[i]
            dev = ex.device
            if dev.type == "cuda":
                torch.cuda.set_device(dev)
            ready[i].set()  # real readiness event, not sleep (fixes B4)
            st = self.stats[i]
            while True:
                x = qs[i].get()
                if x is None:
                    qs[i + 1].put(None)
                    return
                try:
                    if dev.type == "cuda":
                        x = x.to(dev, dtype, non_blocking=True)
                    t0 = time.perf_counter()
                    with torch.no_grad():
                        y = ex.run(x)
                    st.compute_s += time.perf_counter() - t0
                except Exception as e:   # fail-fast per stage: record the
                    # error, poison downstream, then drain upstream so
                    # earlier stages never block on a full queue — a dead
                    # stage must not hang the chain (the reference just
                    # hangs, SURVEY.md §5 failure detection)
                    self.errors[i] = e
                    qs[i + 1].put(None)
                    while qs[i].get() is not None:
                        pass
                    return
                st.items += 1
                st.images += x.shape[0] if x.dim() > 1 else 1
                qs[i + 1].put(y)

        threads = [threading.Thread(target=worker, args=(i,), daemon=True)
                   for i in range(n)]
        for t in threads:
            t.start()
        for e in ready:
            e.wait()  # all stages up before feeding (reference ACK 0x06,
        #             node.py:42, dispatcher.py:64-65)

        # result server: drain final queue into output_stream
        while True:
            y = qs[n].get()
            if y is None:
                break
            output_stream.put(y.float().cpu() if y.is_cuda else y)
        for t in threads:
            t.join()
        failed = [(i, e) for i, e in enumerate(self.errors)
                  if e is not None]
        if failed:
            i, e = failed[0]
            raise RuntimeError(
                f"stage {i} on {self.computeNodes[i]} failed: {e!r}") from e


# --------------------------------------------------------------------------
# distributed pipeline (one process per GPU over RCCL) — the bench path
# --------------------------------------------------------------------------

class DistPipeline:
    """Rank r runs stage r; boundary activations hop r -> r+1 with RCCL
    send/recv over xGMI (reference: node_i -> node_{i+1}:5000 relay,
    node.py:108/89); the last rank returns results to rank 0
    (dispatcher.py:55,103) when cfg.return_results.

    Requires torch.distributed initialized. world_size == num stages.
    """

    def __init__(self, model, cfg: PipelineConfig, batch_shape,
                 device: Optional[torch.device] = None):
        assert dist.is_initialized()
        self.cfg = cfg
        self.rank = dist.get_rank()
        self.world = dist.get_world_size()
        self.device = device or (
            torch.device("cuda", self.rank % torch.cuda.device_count())
            if cfg.device == "cuda" else torch.device("cpu"))
        self.dtype = (cfg.torch_dtype() if self.device.type == "cuda"
                      else torch.float32)
        self.batch_shape = tuple(batch_shape)

        gm = as_graph_model(model)
        if cfg.weights_dir and cfg.partition_layers is None:
            # partition exactly as the checkpoint was partitioned
            from defer_amd import checkpoint

            mani = checkpoint.load_manifest(cfg.weights_dir)
            if mani["num_stages"] != self.world:
                raise ValueError(
                    f"checkpoint has {mani['num_stages']} stages, "
                    f"world is {self.world}")
            if self.world > 1:
                cfg.partition_layers = list(mani["cut_points"])
        if self.world == 1:
            cuts, stages = [], [gm]
        elif cfg.partition_layers is not None:
            cuts = list(cfg.partition_layers)
            stages = partition_model(gm, cuts)
        else:
            from defer_amd.parallel.calibrate import find_calibration
            from defer_amd.parallel.partitioner import XGMI_LINK_GBPS

            cuts, stages = auto_partition(
                gm, self.world,
                input_shape=(1,) + self.batch_shape[1:],
                bytes_per_elem=self._wire_bytes_per_elem(),
                link_gbps=XGMI_LINK_GBPS
                * dual_bw_boost(cfg, self.world),
                measured_us=find_calibration(gm.model_name,
                                             cfg.calibration_file))
        if len(stages) != self.world:
            raise ValueError(f"{len(stages)} stages != world {self.world}")
        self.cuts = cuts
        if cfg.partition_dump_dir and self.rank == 0:
            from defer_amd.utils.visualize import dump_partition

            dump_partition(stages, cfg.partition_dump_dir,
                           cut_points=cuts)
        # shape-trace on CPU BEFORE weights move to the GPU
        self.in_shape, self.out_shape = self._boundary_shapes(stages)
        # compression="auto": per-hop wire choice (fp8 on would-be
        # bottleneck hops, lossless bf16 elsewhere) — deterministic from
        # calibration + cuts, identical on every rank
        self._hop_modes = None
        if cfg.compression == "auto" and self.world > 1:
            from defer_amd.parallel.calibrate import find_calibration
            from defer_amd.parallel.comm import choose_hop_modes
            from defer_amd.parallel.partitioner import (XGMI_LINK_GBPS,
                                                        node_times)

            cal = find_calibration(gm.model_name, cfg.calibration_file)
            _, _, tu = node_times(gm.graph,
                                  (1,) + self.batch_shape[1:])
            if cal:
                tu = {n: cal.get(n, v) for n, v in tu.items()}
            stage_us = [sum(tu[n.name] for n in s.graph.nodes)
                        for s in stages]
            bpe = 2.0 if self.dtype == torch.bfloat16 else 4.0
            hop_bytes = [float(torch.tensor(shp[1:]).prod()) * bpe
                         for shp in self._chain_shapes[:-1]]
            d_act = dual_active(cfg, self.world)
            boosts = [2.0 if d_act
                      and hop_via(i, self.world) is not None else 1.0
                      for i in range(self.world - 1)]
            self._hop_modes = choose_hop_modes(
                stage_us, hop_bytes, XGMI_LINK_GBPS, boosts)
            if self.rank == 0 and cfg.log_stage_stats:
                import sys

                print(f"defer_amd: auto wire modes {self._hop_modes}",
                      file=sys.stderr)
        if cfg.weights_dir:
            # per-rank stage weights only (reference: per-node weight
            # shipping, dispatcher.py:57) — before fusion/pre-cast
            from defer_amd import checkpoint

            checkpoint.load_stage(stages[self.rank], cfg.weights_dir,
                                  self.rank)
        self.stage = StageExecutor(stages[self.rank], self.device,
                                   self.dtype, cfg.use_hip_graphs)

        self.recv_ring = None
        self.send_ring = None
        self.result_ring = None
        self.fwd = None
        dual = dual_active(cfg, self.world)
        if cfg.dual_rail and not dual and self.rank == 0:
            import sys

            print("defer_amd: dual_rail ignored "
                  + ("(variable-size zfp+lz4 wire)"
                     if cfg.compression == "zfp+lz4"
                     else "(needs world > 2)"), file=sys.stderr)
        if self.world > 1:
            # a wire too small to split (codec.wire_numel < 16) stays
            # single-rail: a zero/one-element second half has
            # backend-dependent isend/irecv behavior
            def _ring(codec, hop):
                via = hop_via(hop, self.world) if dual else None
                if via is not None and codec.wire_numel >= 16:
                    return DualRailRing(codec, cfg.ring_depth, via)
                return make_ring(codec, cfg.ring_depth)

            def _mode(hop):
                return (self._hop_modes[hop] if self._hop_modes
                        else None)

            if self.rank > 0:
                self.in_codec = Codec(cfg, self.in_shape, self.dtype,
                                      self.device,
                                      mode=_mode(self.rank - 1))
                self.recv_ring = _ring(self.in_codec, self.rank - 1)
            if self.rank < self.world - 1:
                self.out_codec = Codec(cfg, self.out_shape, self.dtype,
                                       self.device,
                                       mode=_mode(self.rank))
                self.send_ring = _ring(self.out_codec, self.rank)
            if dual:
                # this rank's forwarding duty (at most one hop routes
                # through any given rank — see comm.hop_via)
                fwd_hop = next((i for i in range(self.world - 1)
                                if hop_via(i, self.world) == self.rank),
                               None)
                if fwd_hop is not None:
                    c = Codec(cfg, self._chain_shapes[fwd_hop],
                              self.dtype, self.device,
                              mode=_mode(fwd_hop))
                if fwd_hop is not None and c.wire_numel >= 16:
                    # same small-wire guard as _ring: a hop that stayed
                    # single-rail has no second half to relay
                    self.fwd = HopForwarder(
                        c.wire_numel - split_point(c.wire_numel),
                        c.wire_dtype, self.device, cfg.ring_depth,
                        src=fwd_hop, dst=fwd_hop + 1)
            if cfg.return_results:
                # logits hop last -> 0, never compressed (tiny)
                res_cfg = PipelineConfig(compression="none")
                if self.rank == 0:
                    final_shape = self._final_shape(stages)
                    self.res_codec = Codec(res_cfg, final_shape, self.dtype,
                                           self.device)
                    self.result_ring = P2PRing(self.res_codec,
                                               cfg.ring_depth)
                elif self.rank == self.world - 1:
                    self.res_codec = Codec(res_cfg, self.out_shape,
                                           self.dtype, self.device)
                    self.result_ring = P2PRing(self.res_codec,
                                               cfg.ring_depth)
        if self.world > 1:
            self._warmup_p2p(dual)
        self.stats = StageStats()
        self.timer = None
        if cfg.log_stage_stats:
            from defer_amd.utils.trace import EventTimer

            self.timer = EventTimer(self.device)

    def reset_stats(self):
        """Zero counters/timers (e.g. between warmup and the timed run)."""
        self.stats = StageStats()
        if self.timer is not None:
            from defer_amd.utils.trace import EventTimer

            self.timer = EventTimer(self.device)

    def _warmup_p2p(self, dual: bool):
        """Establish every p2p pair this pipeline uses, in one canonical
        global order, before the overlapped run loop issues anything.

        ProcessGroupNCCL creates a dedicated communicator per p2p pair
        lazily, and the init is host-blocking and collective over the
        pair — if ranks first-touch their pairs in inconsistent orders
        (rank 0's first op is the result-ring irecv from the last rank
        while rank 1 blocks on rank 0's data hop), init can cycle and
        hang. A blocking 1-element exchange per directed pair, walked by
        all ranks in the same sorted order, is cycle-free by induction
        and makes every later isend/irecv hit an existing communicator.
        (The reference's analogue is its per-channel connect handshake,
        dispatcher.py:48,60; on gloo this is just a cheap hello.)"""
        W = self.world
        pairs = [(i, i + 1) for i in range(W - 1)]
        if self.cfg.return_results:
            pairs.append((W - 1, 0))
        if dual:
            for i in range(W - 1):
                via = hop_via(i, W)
                if via is not None:
                    pairs.append((i, via))
                    pairs.append((via, i + 1))
        t = torch.zeros(1, device=self.device)
        for src, dst in sorted(set(pairs)):
            if self.rank == src:
                dist.send(t, dst=dst)
            elif self.rank == dst:
                dist.recv(t, src=src)
        dist.barrier()

    def _wire_bytes_per_elem(self):
        if self.cfg.compression.startswith("zfp"):
            return self.cfg.zfp_rate_bits / 8.0
        if self.cfg.compression == "fp8":
            return 1.0
        return 2.0 if self.cfg.dtype == "bf16" else 4.0

    def _boundary_shapes(self, stages):
        """Chain output shapes across stages via a batch-1 CPU shape trace
        (all ranks compute all; cheap, avoids a collective). Batch dim is
        then restored — every layer here is batch-pointwise."""
        B = self.batch_shape[0]
        shape = (1,) + tuple(self.batch_shape[1:])
        self._chain_shapes = []
        in_shape = out_shape = None
        for i, s in enumerate(stages):
            x = torch.zeros(*shape)
            with torch.no_grad():
                y = s.graph.forward(x)
            if i == self.rank:
                in_shape = (B,) + tuple(shape[1:])
                out_shape = (B,) + tuple(y.shape[1:])
            shape = tuple(y.shape)
            self._chain_shapes.append((B,) + tuple(y.shape[1:]))
        return in_shape, out_shape

    def _final_shape(self, stages):
        return self._chain_shapes[-1]

    # ------------------------------------------------------------------ run
    def run(self, steps: int,
            feed: Optional[Callable[[int], torch.Tensor]] = None,
            collect: Optional[Callable[[int, torch.Tensor], None]] = None):
        """Process `steps` items through the pipeline. rank0 calls
        feed(k) for each item; the final output lands at rank 0 (if
        return_results) or at the last rank, passed to collect(k, y)."""
        cfg = self.cfg
        r, W = self.rank, self.world
        D = cfg.ring_depth

        tm = self.timer

        if W == 1:
            for k in range(steps):
                if tm:
                    tm.start()
                y = self.stage.run(feed(k))
                if tm:
                    tm.stop()
                if collect:
                    collect(k, y)
                self.stats.items += 1
                self.stats.images += self.batch_shape[0]
            if tm:
                self.stats.compute_ms = tm.total_ms()
            return

        nxt, prv = r + 1, r - 1
        last = W - 1

        # pre-post recvs (reverse-order ready before senders start)
        if r > 0:
            self.recv_ring.prime(steps, prv)
        if r == 0 and cfg.return_results:
            self.result_ring.prime(steps, last)
        if self.fwd is not None:
            self.fwd.begin(steps)

        for k in range(steps):
            if self.fwd is not None:
                self.fwd.pump()
            # ---- obtain input
            if r == 0:
                x = feed(k)
            else:
                wire = self.recv_ring.wait_recv(k, steps, prv)
                self.stats.bytes_in += (wire.numel()
                                        * dtype_bytes(wire.dtype))
                x = self.in_codec.decode(wire)
            # ---- compute
            if tm:
                tm.start()
            y = self.stage.run(x)
            if tm:
                tm.stop()
            # ---- repost recv. Safe to reuse the slot: the irecv is posted
            # AFTER the compute consuming it was enqueued, and
            # ProcessGroupNCCL orders the recv after the current stream's
            # already-enqueued work (CPU/gloo compute is synchronous).
            if r > 0:
                self.recv_ring.repost(k, steps, prv)
            # ---- forward result
            if r < last:
                self.stats.bytes_out += self.send_ring.send_encoded(
                    k, y, nxt)
            elif cfg.return_results:
                self.result_ring.send_encoded(k, y, 0)
            elif collect:
                collect(k, y)
            # ---- rank0 result collection
            if r == 0 and cfg.return_results:
                buf = self.result_ring.wait_recv(k, steps, last)
                if collect:
                    collect(k, buf.view(self.res_codec.shape))
                self.result_ring.repost(k, steps, last)
            self.stats.items += 1
            self.stats.images += self.batch_shape[0]

        if tm:
            self.stats.compute_ms = tm.total_ms()
        # finish forwarding duty first: downstream ranks still wait on
        # relayed halves of the tail items
        if self.fwd is not None:
            self.fwd.drain()
            self.stats.bytes_fwd += self.fwd.relayed * self.fwd.nbytes
        # drain outstanding sends, CLEARING the slots: a consumed Work
        # must never be waited again (gloo's Work.wait() is not
        # idempotent — re-waiting one deadlocks, which hung the second
        # run() call of a warmup+timed sequence at world_size > 1)
        for ring in (self.send_ring, self.result_ring):
            if ring is not None:
                self.stats.bytes_out += ring.flush()  # deferred tail send
                for i, w in enumerate(ring.works):
                    if w is not None:
                        w.wait()
                        ring.works[i] = None
                for i, w in enumerate(getattr(ring, "works2", [])):
                    if w is not None:
                        w.wait()
                        ring.works2[i] = None
                for i, w in enumerate(getattr(ring, "size_works", [])):
                    if w is not None:
                        w.wait()
                        ring.size_works[i] = None
