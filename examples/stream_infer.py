#!/usr/bin/env python3
"""Minimal end-to-end example — the reference's usage pattern
(test/test.py:10-49: build the model, hand DEFER the compute nodes and
cut points, stream inputs through queues, count results) on defer_amd.

CPU works out of the box; pass --device cuda on an MI355X box. For the
multi-GPU RCCL pipeline use `python -m defer_amd.node` under torchrun
(see README).
"""
import argparse
import os
import queue
import sys
import threading
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))

from defer_amd import DEFER, PipelineConfig
from defer_amd.models import DEFER_8STAGE_CUTS, MODELS


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--device", default="cpu")
    ap.add_argument("--model", default="resnet50", choices=sorted(MODELS))
    ap.add_argument("--nodes", type=int, default=2,
                    help="pipeline stages (threads on --device)")
    ap.add_argument("--items", type=int, default=8)
    ap.add_argument("--batch", type=int, default=2)
    ap.add_argument("--cuts", default="auto",
                    help='"auto", "defer8", or comma-separated names')
    args = ap.parse_args()

    model = MODELS[args.model]()
    cuts = (None if args.cuts == "auto"
            else DEFER_8STAGE_CUTS if args.cuts == "defer8"
            else [c for c in args.cuts.split(",") if c])

    on_gpu = args.device.startswith("cuda")
    engine = DEFER([args.device] * args.nodes,
                   config=PipelineConfig(device="cuda" if on_gpu
                                         else "cpu",
                                         dtype="bf16" if on_gpu
                                         else "fp32"))
    if cuts is not None and len(cuts) + 1 != args.nodes:
        raise SystemExit(f"{len(cuts)} cuts make {len(cuts) + 1} stages "
                         f"but --nodes is {args.nodes}")

    inputs, outputs = queue.Queue(10), queue.Queue(10)
    err = []

    def serve():
        try:
            engine.run_defer(model, cuts, inputs, outputs)
        except Exception as e:              # surface engine errors to the
            err.append(e)                   # feeder instead of hanging it
            outputs.put(None)

    t = threading.Thread(target=serve)
    t.start()

    dtype = torch.bfloat16 if on_gpu else torch.float32
    x = torch.randn(args.batch, 224, 224, 3, dtype=dtype,
                    device=args.device if on_gpu else "cpu")
    t0 = time.perf_counter()
    for _ in range(args.items):
        inputs.put(x)
    inputs.put(None)                      # clean shutdown
    got = []
    for _ in range(args.items):
        y = outputs.get(timeout=600)
        if y is None:
            raise SystemExit(f"pipeline failed: {err[0]!r}")
        got.append(y)
    t.join(timeout=600)
    dt = time.perf_counter() - t0
    print(f"{args.items} items x batch {args.batch} through "
          f"{args.nodes} {args.device} stages: "
          f"{args.items * args.batch / dt:.1f} images/sec; "
          f"output {tuple(got[-1].shape)}")
    for i, st in enumerate(engine.stats):
        print(f"  stage {i}: {st.items} items, "
              f"{st.compute_s * 1e3 / max(st.items, 1):.1f} ms/item")


if __name__ == "__main__":
    main()
