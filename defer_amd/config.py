"""Pipeline configuration.

The reference hardcodes every constant (ports 5000/5001/5002, chunk size
512000, queue depths 1000/10 — dispatcher.py:18,24, node.py:111,114,
test/test.py:39-40). Here the equivalent knobs are one dataclass, so every
config named in BASELINE.json is expressible.
"""

from dataclasses import dataclass, field
from typing import List, Optional


@dataclass
class PipelineConfig:
    # --- partitioning -----------------------------------------------------
    # Names of cut layers (the reference's `partition_layers`,
    # dispatcher.py:107; e.g. ["add_2", "add_4", ...] for ResNet50,
    # test/test.py:18). Empty list -> single stage. If None and
    # num_stages > 1, cuts are chosen by the cost-model auto-partitioner.
    partition_layers: Optional[List[str]] = None
    num_stages: int = 1
    # Single-item input shape for the auto-partitioner's cost trace
    # (batch dim 1). Defaults to the reference workload's 224x224x3 NHWC
    # image (test/test.py:20) when None.
    input_shape: Optional[tuple] = None

    # --- execution --------------------------------------------------------
    device: str = "cuda"          # "cuda" (MI355X) or "cpu" (tests/plumbing)
    dtype: str = "bf16"           # compute dtype on GPU; "fp32" on CPU
    batch_size: int = 64          # images per pipeline item (micro-batch)
    # Capture each stage's forward in a hipGraph. Off by default:
    # measured no gain at bench batch sizes (launch overhead is already
    # negligible, profiles/README.md), and concurrent captures from the
    # threaded orchestrator's workers race on the shared legacy stream
    # (observed intermittent first-item corruption). DistPipeline (one
    # process per GPU) can enable it safely via bench --graphs.
    use_hip_graphs: bool = False

    # --- inter-stage relay (the data plane) -------------------------------
    # Codec for boundary activations. The reference compresses with
    # lz4(zfp(x)) (dispatcher.py:81-84). "none" ships raw bf16 over xGMI;
    # "fp8" ships a cast-only e4m3 wire (1 B/value, ~free); "zfp" ships
    # fixed-rate ZFP blocks; "zfp+lz4" adds the LZ4 stage.
    # "auto" picks the wire PER HOP: fp8 on hops whose raw-bf16 relay
    # would exceed the slowest stage's compute (the pipeline bottleneck),
    # lossless bf16 elsewhere (comm.choose_hop_modes — deterministic
    # from the calibration + cuts, so all ranks agree).
    compression: str = "none"  # "none"|"fp8"|"zfp"|"zfp+lz4"|"auto"
    zfp_rate_bits: int = 8        # fixed-rate bits per value (ZFP)

    # Depth of the per-stage device-resident activation ring buffers (the
    # reference's Queue(1000) backpressure buffer, node.py:114 — here sized
    # in micro-batches; 288 GB HBM per GPU means depth is cheap).
    ring_depth: int = 4

    # Dual-rail relay: split each fixed-size hop across two xGMI rails —
    # half direct src->dst, half via an idle third GPU (src->via->dst,
    # item-granularity store-and-forward). In a pipeline chain every
    # non-adjacent directed link is idle, so this doubles effective hop
    # bandwidth for the early fat boundaries that exceed one 153 GB/s
    # link (ResNet50 layer1: 1.6 MB/img = 10.5 us/hop, the 8-stage
    # bottleneck — profiles/README.md "Predicted pipeline scaling").
    # Requires world > 2 and a fixed-size wire (not "zfp+lz4").
    # bench.py defaults it ON at world >= 4 (with a warmup stall
    # watchdog that re-execs single-rail, since the first RCCL
    # execution is the round-end multi-GPU run itself); the raw
    # PipelineConfig default stays False.
    dual_rail: bool = False

    # Directory of per-stage checkpoints written by
    # checkpoint.save_stages (part{i}.pt + manifest.json). Each rank
    # loads ONLY its own stage's weights — the reference's per-node
    # weight shipping (dispatcher.py:57, node.py:53-75) without moving
    # the full model to every GPU. When set and partition_layers is
    # None, the manifest's cut list is used. None -> random-init (the
    # BASELINE.json benchmark path).
    weights_dir: Optional[str] = None

    # Per-layer measured-cost file (JSON written by tools/calibrate.py /
    # defer_amd.parallel.calibrate) used by the auto-partitioner instead
    # of the static cost model. None -> auto: use the in-tree
    # defer_amd/calib/{model}.json profile when one exists (SURVEY §7
    # "per-layer-cost profiler to pick cuts", measured not hand-tuned).
    calibration_file: Optional[str] = None

    # --- distributed ------------------------------------------------------
    backend: str = "nccl"         # "nccl" (RCCL over xGMI) or "gloo" (CPU)
    # Whether the last stage sends results back to rank 0 (the reference's
    # node(N-1) -> dispatcher return hop, dispatcher.py:55,103).
    return_results: bool = True

    # --- observability ----------------------------------------------------
    log_stage_stats: bool = False  # per-stage imgs/s, bytes relayed, ratios
    # When set, write per-stage DOT + text partition dumps here (the
    # reference writes model_{ip}.png per node, node.py:39).
    partition_dump_dir: Optional[str] = None

    extra: dict = field(default_factory=dict)

    def __post_init__(self):
        if self.compression not in ("none", "fp8", "zfp", "zfp+lz4",
                                    "auto"):
            raise ValueError(
                f"unknown compression {self.compression!r} "
                "(none|fp8|zfp|zfp+lz4|auto)")
        if self.dtype not in ("bf16", "fp16", "fp32"):
            raise ValueError(f"unknown dtype {self.dtype!r} "
                             "(bf16|fp16|fp32)")
        if self.ring_depth < 1:
            raise ValueError("ring_depth must be >= 1")
        if not 1 <= self.zfp_rate_bits <= 30:
            raise ValueError("zfp_rate_bits must be in [1, 30]")

    def torch_dtype(self):
        import torch

        return {"bf16": torch.bfloat16, "fp16": torch.float16,
                "fp32": torch.float32}[self.dtype]
