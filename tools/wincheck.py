#!/usr/bin/env python3
"""Numerics check for the window-reuse conv path (DEFER_CONV_VARIANT=w
forces it past the size gate): forced-win vs the fp32 CPU reference,
including edge tiles, residual fusion, and multi-cb shapes."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
os.environ.setdefault("DEFER_CONV_VARIANT", "w")

import torch

import defer_amd.ops as ops
from defer_amd.ops import reference as ref

SHAPES = [  # (B, H, W, Cin, Cout, res)
    (2, 16, 16, 64, 64, False),
    (2, 8, 16, 64, 128, True),
    (2, 8, 16, 128, 64, False),      # ncb=2, single tile per image
    (2, 8, 16, 192, 64, False),      # ncb=3
    (3, 24, 32, 128, 192, False),
    (2, 16, 48, 256, 64, True),
    (1, 56, 112, 64, 64, False),     # W%16=0, H%8=0
    (2, 56, 56, 64, 64, True),       # 56%16=8: column-masked edge tiles
    (2, 28, 28, 128, 128, False),    # 28%8=4 and 28%16=12: both masked
    (2, 14, 14, 128, 64, False),     # tiny, heavy masking
    (2, 12, 16, 64, 64, True),       # TH=4 tile path (OH%8 != 0)
    (3, 28, 28, 192, 128, False),    # TH=4 with masked cols
]

# small-Cin window (stem) shapes: (B, H, W, Cin, Cout, R, stride, pad)
SWIN = [
    (2, 16, 32, 3, 64, 3, 1, 1),     # VGG-stem class
    (2, 30, 28, 3, 64, 3, 1, 1),     # masked edge tiles
    (2, 24, 32, 4, 64, 3, 2, 1),     # Cin=4, stride 2
    (1, 224, 224, 3, 64, 3, 1, 1),   # full VGG b1.c1 shape
]

fails = 0
for (B, H, W, Cin, Cout, R, stride, pad) in SWIN:
    torch.manual_seed(1)
    x = torch.randn(B, H, W, Cin, dtype=torch.bfloat16)
    w = (torch.randn(Cout, R, R, Cin, dtype=torch.bfloat16) * 0.05)
    sc = torch.rand(Cout) + 0.5
    bi = torch.randn(Cout) * 0.1
    want = ref.conv2d_bn_act(x.float(), w.float(), sc, bi, stride, pad,
                             "relu", None)
    y = ops.conv2d_bn_act(x.cuda(), w.cuda(), sc.cuda(), bi.cuda(),
                          stride=stride, padding=pad, act="relu")
    got = y.float().cpu()
    rel = ((got - want).abs().max().item()
           / max(want.abs().max().item(), 1e-6))
    ok = rel < 0.05 and torch.isfinite(got).all()
    print(f"swin {(B,H,W,Cin,Cout,R,stride)}: maxrel {rel:.4f} "
          f"{'OK' if ok else 'FAIL'}")
    fails += not ok

for (B, H, W, Cin, Cout, res) in SHAPES:
    torch.manual_seed(0)
    x = torch.randn(B, H, W, Cin, dtype=torch.bfloat16)
    w = (torch.randn(Cout, 3, 3, Cin, dtype=torch.bfloat16) * 0.05)
    sc = torch.rand(Cout) + 0.5
    bi = torch.randn(Cout) * 0.1
    rt = (torch.randn(B, H, W, Cout, dtype=torch.bfloat16)
          if res else None)
    want = ref.conv2d_bn_act(x.float(), w.float(), sc, bi, 1, 1, "relu",
                             rt.float() if res else None)
    y = ops.conv2d_bn_act(x.cuda(), w.cuda(), sc.cuda(), bi.cuda(),
                          stride=1, padding=1, act="relu",
                          residual=rt.cuda() if res else None)
    got = y.float().cpu()
    err = (got - want).abs()
    rel = err.max().item() / max(want.abs().max().item(), 1e-6)
    ok = rel < 0.05 and torch.isfinite(got).all()
    print(f"{(B,H,W,Cin,Cout,res)}: maxrel {rel:.4f} "
          f"{'OK' if ok else 'FAIL'}")
    if not ok:
        # error map: which 8x16 tiles are wrong?
        e = (err / max(want.abs().max().item(), 1e-6)) > 0.05
        bad = e.any(dim=-1)   # [B,H,W]
        for b in range(B):
            rows = ["".join("X" if bad[b, h, wd] else "."
                            for wd in range(W)) for h in range(H)]
            print(f"  img{b}:"); [print("   ", r) for r in rows[:24]]
    fails += not ok
sys.exit(1 if fails else 0)
