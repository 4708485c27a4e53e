#!/usr/bin/env python3
"""Per-shape conv kernel microbenchmark (run on MI355X via gpurun).

Times every distinct ResNet50/VGG19 conv shape at the bench batch size and
prints us/call, TFLOP/s and achieved GB/s next to a roofline estimate.
"""

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

import defer_amd.ops as ops

RESNET_SHAPES = [
    # (name, H, W, Cin, Cout, R, stride, pad, res)
    ("stem7x7", 224, 224, 3, 64, 7, 2, 3, False),
    ("L1.c1x1_64", 56, 56, 64, 64, 1, 1, 0, False),
    ("L1.c3x3_64", 56, 56, 64, 64, 3, 1, 1, False),
    ("L1.c1x1_256+res", 56, 56, 64, 256, 1, 1, 0, True),
    ("L1.proj256", 56, 56, 64, 256, 1, 1, 0, False),
    ("L2.c1x1_128from256", 56, 56, 256, 128, 1, 1, 0, False),
    ("L2.c3x3_128s2", 56, 56, 128, 128, 3, 2, 1, False),
    ("L2.c3x3_128", 28, 28, 128, 128, 3, 1, 1, False),
    ("L2.c1x1_512+res", 28, 28, 128, 512, 1, 1, 0, True),
    ("L2.proj512s2", 56, 56, 256, 512, 1, 2, 0, False),
    ("L3.c1x1_256from512", 28, 28, 512, 256, 1, 1, 0, False),
    ("L3.c3x3_256s2", 28, 28, 256, 256, 3, 2, 1, False),
    ("L3.c3x3_256", 14, 14, 256, 256, 3, 1, 1, False),
    ("L3.c1x1_1024+res", 14, 14, 256, 1024, 1, 1, 0, True),
    ("L4.c1x1_512from1024", 14, 14, 1024, 512, 1, 1, 0, False),
    ("L4.c3x3_512s2", 14, 14, 512, 512, 3, 2, 1, False),
    ("L4.c3x3_512", 7, 7, 512, 512, 3, 1, 1, False),
    ("L4.c1x1_2048+res", 7, 7, 512, 2048, 1, 1, 0, True),
]

VGG_SHAPES = [
    ("b1.c1", 224, 224, 3, 64, 3, 1, 1, False),
    ("b1.c2", 224, 224, 64, 64, 3, 1, 1, False),
    ("b2.c1", 112, 112, 64, 128, 3, 1, 1, False),
    ("b2.c2", 112, 112, 128, 128, 3, 1, 1, False),
    ("b3.c", 56, 56, 256, 256, 3, 1, 1, False),
    ("b4.c", 28, 28, 512, 512, 3, 1, 1, False),
    ("b5.c", 14, 14, 512, 512, 3, 1, 1, False),
]


def bench_shape(name, H, W, Cin, Cout, R, stride, pad, res, B, iters=30):
    dev = "cuda:0"
    x = torch.randn(B, H, W, Cin, device=dev, dtype=torch.bfloat16)
    w = torch.randn(Cout, R, R, Cin, device=dev, dtype=torch.bfloat16) * 0.05
    sc = torch.rand(Cout, device=dev) + 0.5
    bi = torch.randn(Cout, device=dev) * 0.1
    OH = (H + 2 * pad - R) // stride + 1
    rt = (torch.randn(B, OH, OH, Cout, device=dev, dtype=torch.bfloat16)
          if res else None)
    for _ in range(5):
        y = ops.conv2d_bn_act(x, w, sc, bi, stride=stride, padding=pad,
                              act="relu", residual=rt)
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(True)
    t1 = torch.cuda.Event(True)
    t0.record()
    for _ in range(iters):
        y = ops.conv2d_bn_act(x, w, sc, bi, stride=stride, padding=pad,
                              act="relu", residual=rt)
    t1.record()
    torch.cuda.synchronize()
    us = t0.elapsed_time(t1) * 1e3 / iters
    M = B * OH * OH
    K = R * R * Cin
    fl = 2.0 * M * Cout * K
    tf = fl / us / 1e6
    # min traffic: x once, out once, weights once (tiny)
    gb = (x.numel() * 2 + y.numel() * 2 + w.numel() * 2
          + (rt.numel() * 2 if res else 0)) / us / 1e3
    return dict(name=name, us=round(us, 1), tflops=round(tf, 1),
                min_gbps=round(gb, 0), M=M, K=K, N=Cout)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--iters", type=int, default=30)
    args = ap.parse_args()
    shapes = RESNET_SHAPES if args.model == "resnet50" else VGG_SHAPES
    total = 0.0
    rows = []
    for s in shapes:
        r = bench_shape(*s, args.batch, args.iters)
        rows.append(r)
        print(f"{r['name']:22s} M={r['M']:7d} N={r['N']:4d} K={r['K']:5d} "
              f"{r['us']:8.1f} us  {r['tflops']:7.1f} TF  "
              f">={r['min_gbps']:6.0f} GB/s")
        total += r["us"]
    print(f"total (one call each): {total:.1f} us")
    print(json.dumps(rows))


if __name__ == "__main__":
    main()
