#!/usr/bin/env python3
"""One-shot per-layer cost calibration on the GPU.

    python tools/calibrate.py resnet50 [--batch 64] [--out PATH]

Writes defer_amd/calib/{model}.json (us per image per graph node),
consumed by auto_partition via PipelineConfig.calibration_file or the
in-tree default lookup. Commit the artifact: cuts are then chosen from
measured stage times on MI355X, not hand-tuned constants."""

import argparse

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from defer_amd.models import MODELS
from defer_amd.parallel.calibrate import calibrate_and_save


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("model", choices=sorted(MODELS))
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--out", default=None)
    ap.add_argument("--device", default="cuda")
    args = ap.parse_args()
    torch.manual_seed(0)
    gm = MODELS[args.model]()
    path = calibrate_and_save(gm, args.out,
                              (args.batch, 224, 224, 3), args.device)
    print(f"wrote {path}")


if __name__ == "__main__":
    main()
