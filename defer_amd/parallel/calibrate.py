"""Measured per-layer cost calibration for the auto-partitioner.

The reference leaves cut choice to the user (test/test.py:17-18); round 1
replaced that with a cost model whose kernel-class rates were hand-measured
constants (partitioner._EFF_* / _BW_*). This module replaces the constants
with a one-shot on-GPU per-layer timing pass ("provide a
simple per-layer-cost profiler to pick cuts"): each graph node's layer is
run standalone on its real input shape and timed with hipEvent pairs; the
result is cached as a JSON artifact (us per image per node) that
auto_partition consumes directly, so any kernel change re-measures rather
than silently invalidating the cuts.

Artifact format (defer_amd/calib/{model}.json or a user path):
    {"model": "resnet50", "batch": 64, "device": "...",
     "us_per_image": {node_name: float, ...}}
"""

import copy
import json
import os
import time
from typing import Dict, Optional

import torch

from defer_amd.graph import LayerGraph

_CALIB_DIR = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "calib")


def calibration_path(model_name: str) -> str:
    return os.path.join(_CALIB_DIR, f"{model_name}.json")


def load_calibration(path: str) -> Dict[str, float]:
    """Return {node_name: us_per_image} from a calibration artifact."""
    with open(path) as f:
        d = json.load(f)
    return {k: float(v) for k, v in d["us_per_image"].items()}


def find_calibration(model_name: str,
                     explicit: Optional[str] = None
                     ) -> Optional[Dict[str, float]]:
    """Explicit file wins; else the in-tree per-model artifact; else None
    (auto_partition falls back to the static cost model)."""
    path = explicit or calibration_path(model_name)
    if os.path.exists(path):
        return load_calibration(path)
    return None


@torch.no_grad()
def measure_layer_times(gm, input_shape=(64, 224, 224, 3),
                        device="cuda", dtype=torch.bfloat16,
                        iters: int = 30, warmup: int = 8,
                        reps: int = 3) -> Dict[str, float]:
    """Time each graph node's layer standalone on `device` at the given
    batch; returns {node_name: us per IMAGE}. GPU-only by intent (the
    partitioner's costs describe the HIP kernels); CPU works for tests
    with wall-clock timing.

    Each layer is timed as the MIN over `reps` independent event-timed
    runs: one-off multi-ms stalls (allocator slow paths, DPM dips on a
    fresh box) otherwise land in a single layer's mean and poison the
    cut choice — observed as ~47 us/img outliers on elementwise nodes,
    114x their real cost."""
    graph: LayerGraph = gm.graph
    dev = torch.device(device)
    cuda = dev.type == "cuda"
    if not cuda:
        dtype = torch.float32
    batch = int(input_shape[0])

    if cuda:
        # sustained load to ramp DPM clocks before any timed region
        xw = torch.randn(max(batch, 16), 56, 56, 64, device=dev,
                         dtype=dtype)
        ww = (torch.randn(64, 3, 3, 64) * 0.05).to(dev, dtype)
        from defer_amd import ops as _ops

        t0 = time.perf_counter()
        while time.perf_counter() - t0 < 0.6:
            for _ in range(50):
                _ops.conv2d_bn_act(xw, ww, None, None, stride=1,
                                   padding=1)
            torch.cuda.synchronize(dev)
        del xw, ww

    # shape trace on CPU to learn each node's input shapes
    shapes = {LayerGraph.INPUT: tuple(input_shape)}
    x = torch.zeros(1, *input_shape[1:])
    env = {LayerGraph.INPUT: x}
    for n in graph.nodes:
        args = [env[p] for p in n.inputs]
        env[n.name] = n.layer(*args, **n.kwargs)
        shapes[n.name] = (batch,) + tuple(env[n.name].shape[1:])

    times_us: Dict[str, float] = {}
    for n in graph.nodes:
        layer = copy.deepcopy(n.layer).to(dev)
        if cuda:
            for p in layer.parameters():
                if p.dim() >= 2:
                    p.data = p.data.to(dtype).contiguous()
                else:
                    p.data = p.data.float()
        ins = [torch.randn(*shapes[p], device=dev).to(dtype).contiguous()
               for p in n.inputs]
        for _ in range(warmup):
            layer(*ins, **n.kwargs)
        it = max(iters // reps, 1)
        best_ms = None
        for _ in range(reps):
            if cuda:
                torch.cuda.synchronize(dev)
                ev0 = torch.cuda.Event(True)
                ev1 = torch.cuda.Event(True)
                ev0.record()
                for _ in range(it):
                    layer(*ins, **n.kwargs)
                ev1.record()
                ev1.synchronize()
                ms = ev0.elapsed_time(ev1)
            else:
                t0 = time.perf_counter()
                for _ in range(it):
                    layer(*ins, **n.kwargs)
                ms = (time.perf_counter() - t0) * 1e3
            best_ms = ms if best_ms is None else min(best_ms, ms)
        times_us[n.name] = best_ms * 1e3 / it / batch
        del ins, layer
    return times_us


def calibrate_and_save(gm, path: Optional[str] = None,
                       input_shape=(64, 224, 224, 3), device="cuda",
                       dtype=torch.bfloat16) -> str:
    """Run the timing pass and write the artifact. Returns the path."""
    us = measure_layer_times(gm, input_shape, device, dtype)
    path = path or calibration_path(gm.model_name)
    os.makedirs(os.path.dirname(path), exist_ok=True)
    dev_name = (torch.cuda.get_device_name(0)
                if torch.device(device).type == "cuda"
                and torch.cuda.is_available() else str(device))
    with open(path, "w") as f:
        json.dump({"model": gm.model_name, "batch": int(input_shape[0]),
                   "device": dev_name,
                   "us_per_image": {k: round(v, 4)
                                    for k, v in us.items()}}, f, indent=1)
    return path
