#!/usr/bin/env python3
"""Single-shape conv loop for PMC counter runs (rocprofv3 --pmc ... --
python tools/pmcprobe.py <shape> [iters]). Prints theoretical bytes per
call so FETCH_SIZE/WRITE_SIZE accounting (profiles: the streaming-bound
1x1+res shapes' next probe) reduces to a division."""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

import defer_amd.ops as ops

SHAPES = {
    "l1res": (64, 56, 56, 64, 256, 1, 1, 0, True),    # L1.c1x1_256+res
    "l2res": (64, 28, 28, 128, 512, 1, 1, 0, True),
    "l3res": (64, 14, 14, 256, 1024, 1, 1, 0, True),
    "l1c1": (64, 56, 56, 64, 64, 1, 1, 0, False),
}


def main():
    name = sys.argv[1] if len(sys.argv) > 1 else "l1res"
    iters = int(sys.argv[2]) if len(sys.argv) > 2 else 20
    N, H, W, Cin, Cout, R, stride, pad, res = SHAPES[name]
    x = torch.randn(N, H, W, Cin, device="cuda", dtype=torch.bfloat16)
    w = (torch.randn(Cout, R, R, Cin) * 0.05).to("cuda", torch.bfloat16)
    sc = torch.rand(Cout, device="cuda") + 0.5
    bi = torch.zeros(Cout, device="cuda")
    OH = (H + 2 * pad - R) // stride + 1
    r = (torch.randn(N, OH, OH, Cout, device="cuda",
                     dtype=torch.bfloat16) if res else None)
    for _ in range(3):
        y = ops.conv2d_bn_act(x, w, sc, bi, stride=stride, padding=pad,
                              act="relu", residual=r)
    torch.cuda.synchronize()
    for _ in range(iters):
        y = ops.conv2d_bn_act(x, w, sc, bi, stride=stride, padding=pad,
                              act="relu", residual=r)
    torch.cuda.synchronize()
    rd = x.numel() * 2 + w.numel() * 2 + (r.numel() * 2 if res else 0) \
        + Cout * 8
    wr = y.numel() * 2
    print(f"{name}: theory read {rd/1e6:.1f} MB/call, "
          f"write {wr/1e6:.1f} MB/call, iters={iters} (+3 warmup)")


if __name__ == "__main__":
    main()
