#!/usr/bin/env python3
"""ZFP codec throughput on MI355X: encode/decode GB/s per boundary shape."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from defer_amd.ops import codec
import defer_amd._hip_ops as hip

SHAPES = [  # ResNet50 boundary activations at batch 64 (bf16)
    ("add_2 56x56x256", (64, 56, 56, 256)),
    ("add_8 28x28x512", (64, 28, 28, 512)),
    ("add_12 14x14x1024", (64, 14, 14, 1024)),
    ("add_16 7x7x2048", (64, 7, 7, 2048)),
]

def bench(shape, rate, iters=30):
    x = torch.randn(*shape, device="cuda", dtype=torch.bfloat16)
    w = codec.zfp_encode(x, rate)
    y = codec.zfp_decode(w, shape, rate, dtype=torch.bfloat16)
    torch.cuda.synchronize()
    t0, t1, t2 = (torch.cuda.Event(True) for _ in range(3))
    t0.record()
    for _ in range(iters):
        codec.zfp_encode(x, rate, out=w)
    t1.record()
    for _ in range(iters):
        y = codec.zfp_decode(w, shape, rate, dtype=torch.bfloat16)
    t2.record()
    torch.cuda.synchronize()
    enc_us = t0.elapsed_time(t1) * 1e3 / iters
    dec_us = t1.elapsed_time(t2) * 1e3 / iters
    # phase bisection: transform-only / serialize-only encodes
    ph = {}
    for phase in (1, 2, 4, 5):
        torch.cuda.synchronize()
        a, b = torch.cuda.Event(True), torch.cuda.Event(True)
        a.record()
        for _ in range(iters):
            hip.zfp_encode(x, rate, w, phase)
        b.record()
        torch.cuda.synchronize()
        ph[phase] = a.elapsed_time(b) * 1e3 / iters
    mb = x.numel() * 2 / 1e6
    return enc_us, dec_us, mb, w.numel() / 1e6, ph

def bench_fp8(shape, iters=50):
    x = torch.randn(*shape, device="cuda", dtype=torch.bfloat16)
    n = x.numel()
    w = torch.empty(n + 4, dtype=torch.uint8, device="cuda")
    hip.fp8_encode(x, w)
    y = hip.fp8_decode(w, list(shape))
    torch.cuda.synchronize()
    t0, t1, t2 = (torch.cuda.Event(True) for _ in range(3))
    t0.record()
    for _ in range(iters):
        hip.fp8_encode(x, w)
    t1.record()
    for _ in range(iters):
        y = hip.fp8_decode(w, list(shape))
    t2.record()
    torch.cuda.synchronize()
    enc = t0.elapsed_time(t1) * 1e3 / iters
    dec = t1.elapsed_time(t2) * 1e3 / iters
    mb = n * 2 / 1e6
    print(f"fp8  {str(shape):20s}: enc {enc:7.1f} us ({mb/enc*1e3:6.0f} GB/s) "
          f"dec {dec:7.1f} us ({mb/dec*1e3:6.0f} GB/s)  {mb:.0f}->{(n+4)/1e6:.0f} MB")


for name, shape in SHAPES:
    bench_fp8(shape)
for name, shape in SHAPES:
    for rate in (4, 8):
        e, d, mb, wmb, ph = bench(shape, rate)
        print(f"{name:20s} rate{rate:2d}: enc {e:7.1f} us ({mb/e*1e3:6.0f} GB/s) "
              f"dec {d:7.1f} us ({mb/d*1e3:6.0f} GB/s)  {mb:.0f}->{wmb:.0f} MB"
              f"  [p1 {ph[1]:.1f}us p2 {ph[2]:.1f}us stage {ph[4]:.1f}us lift {ph[5]:.1f}us]")
