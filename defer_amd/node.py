"""Stage-worker entry point — the reference's self-running compute-node
script (node.py:126-127: `Node().run()` serving one stage of the chain
forever) as a torchrun module: rank r serves pipeline stage r on GPU r,
rank 0 doubles as the dispatcher (data feeder + result server,
dispatcher.py:85-105).

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 -m defer_amd.node \
        --model resnet50 --cuts auto --batch 64

Feeds synthetic batches (there is no dataset on the node; the reference
streams one preprocessed image repeatedly, test/test.py:20-23) and prints
a rolling images/sec line from rank 0 every --report items — the
reference's count-per-window throughput protocol (test/test.py:25-37).
Runs until --items are served (0 = forever, like the reference's
`a.join()`), then shuts down cleanly.
"""

import argparse
import os
import sys
import time

import torch
import torch.distributed as dist


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="resnet50",
                    choices=["resnet50", "resnet101", "resnet152",
                             "vgg19", "vgg19_gap", "densenet121"])
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--cuts", default="auto")
    ap.add_argument("--items", type=int, default=0,
                    help="items to serve (0 = forever)")
    ap.add_argument("--report", type=int, default=64,
                    help="rank 0 prints throughput every N items")
    ap.add_argument("--compression", default="none",
                    choices=["none", "fp8", "zfp", "zfp+lz4", "auto"])
    ap.add_argument("--device", default="cuda", choices=["cuda", "cpu"])
    ap.add_argument("--dual-rail", action="store_true")
    ap.add_argument("--calibration", default=None,
                    help="per-layer measured-cost file for auto cuts "
                         "(tools/calibrate.py; default: the in-tree "
                         "defer_amd/calib profile for the model)")
    ap.add_argument("--weights-dir", default=None,
                    help="per-stage checkpoint dir (checkpoint."
                         "save_stages); cuts come from its manifest")
    args = ap.parse_args()

    from defer_amd.config import PipelineConfig
    from defer_amd.models import DEFER_8STAGE_CUTS, MODELS
    from defer_amd.parallel.pipeline import DistPipeline

    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    backend = "nccl" if args.device == "cuda" else "gloo"
    if args.device == "cuda":
        torch.cuda.set_device(local_rank)
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29772")
        if world == 1:
            dist.init_process_group(backend, rank=0, world_size=1)
        else:
            dist.init_process_group(backend)

    if args.cuts == "auto":
        cuts = None
    elif args.cuts == "defer8":
        cuts = DEFER_8STAGE_CUTS
    else:
        cuts = [c for c in args.cuts.split(",") if c]

    cfg = PipelineConfig(
        partition_layers=cuts, num_stages=world, device=args.device,
        dtype="bf16" if args.device == "cuda" else "fp32",
        batch_size=args.batch, use_hip_graphs=False,
        compression=args.compression, dual_rail=args.dual_rail,
        weights_dir=args.weights_dir,
        calibration_file=args.calibration,
        backend=backend, return_results=True)
    dev = (torch.device("cuda", local_rank) if args.device == "cuda"
           else torch.device("cpu"))
    B = args.batch
    model = MODELS[args.model]()
    pipe = DistPipeline(model, cfg, (B, 224, 224, 3), device=dev)
    dtype = torch.bfloat16 if args.device == "cuda" else torch.float32
    x = torch.randn(B, 224, 224, 3, device=dev, dtype=dtype)

    dist.barrier()   # collective comm init before any p2p op

    state = {"n": 0, "t0": time.perf_counter()}

    def collect(k, y):
        state["n"] += 1
        if state["n"] % args.report == 0:
            dt = time.perf_counter() - state["t0"]
            print(f"[node rank0] {state['n']} items, "
                  f"{state['n'] * B / dt:.0f} images/sec", flush=True)

    chunk = args.report
    served = 0
    try:
        while args.items == 0 or served < args.items:
            n = chunk if args.items == 0 else min(chunk,
                                                  args.items - served)
            pipe.run(n, feed=lambda k: x, collect=collect)
            served += n
    except KeyboardInterrupt:
        pass
    dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
