"""Model partitioner — the reference's dag_util + DEFER._partition rebuilt.

`partition_model(model, cut_points)` splits a GraphModel (or any traceable
nn.Module) into N sequential stage models at named articulation layers —
the contract of construct_model/traverse (dag_util.py:27-31) enforced on an
explicit DAG.

`auto_partition(model, num_stages, ...)` is the cut chooser the reference
lacks (it leaves cut choice to the user, test/test.py:17-18; uneven stages
cap pipeline speedup). It picks cuts minimizing the pipeline bottleneck
max(stage_compute, hop_bytes/link_bw) from a static per-layer cost model,
because on MI355X the steady-state pipeline throughput is
max-over-stages(compute, xGMI relay).
"""

from typing import List, Tuple

import torch
import torch.nn as nn

from defer_amd.graph import GraphModel, LayerGraph, from_torch

# xGMI p2p: 7 links x ~153 GB/s per GPU; one dedicated link per pipeline
# hop, full duplex. Used only as a relative weight in the cut chooser.
XGMI_LINK_GBPS = 153.0
# Effective sustained compute for weighting conv FLOPs against hop bytes
# (relative units; ratio is what matters). Calibrated against the measured
# whole-model ResNet50 bf16 forward on MI355X (~1.83 ms / 64 images at
# 8.2 GFLOP/image ~= 287 TF effective — bench.py, tile policy v3 +
# staging rewrite; see profiles/README.md).
EFF_TFLOPS = 287.0


def as_graph_model(model) -> GraphModel:
    if isinstance(model, GraphModel):
        return model
    if isinstance(model, LayerGraph):
        return GraphModel(model)
    if isinstance(model, nn.Module):
        return GraphModel(from_torch(model), name=type(model).__name__)
    raise TypeError(f"cannot partition {type(model)!r}")


def partition_model(model, cut_points: List[str]) -> List[GraphModel]:
    """Split at named layers into len(cut_points)+1 stage models
    (part1..partN, dispatcher.py:27-42)."""
    gm = as_graph_model(model)
    return gm.stage_models(list(cut_points))


# --------------------------------------------------------------------------
# cost model
# --------------------------------------------------------------------------

def _trace_shapes(graph: LayerGraph, input_shape) -> dict:
    """Run the graph once on CPU meta-ish (tiny batch) to get each node's
    output shape."""
    shapes = {}
    x = torch.zeros(*input_shape)
    env = {LayerGraph.INPUT: x}
    for n in graph.nodes:
        args = [env[p] for p in n.inputs]
        with torch.no_grad():
            env[n.name] = n.layer(*args, **n.kwargs)
        shapes[n.name] = tuple(env[n.name].shape)
    return shapes


def node_costs(graph: LayerGraph, input_shape=(1, 224, 224, 3),
               bytes_per_elem: float = 2.0) -> Tuple[dict, dict]:
    """Per-node (flops, output_bytes) for one input of `input_shape`."""
    fl, ob, _ = node_times(graph, input_shape, bytes_per_elem)
    return fl, ob


# Measured kernel-class rates on MI355X (tools/convbench.py, tile policy
# v3 + window kernel; see profiles/). A node's time estimate is
# max(flops / EFF_class, streamed bytes / BW_class) — the 1x1 and
# elementwise classes are streaming-bound, the 3x3s MFMA/L2-bound.
_EFF_3X3 = 450e12       # window / RSC implicit-GEMM 3x3s
_EFF_1X1 = 300e12       # GEMM-mode 1x1s
_EFF_STEM = 150e12      # Cin<64 stem paths
_BW_STREAM = 4.5e12     # achieved streaming bound of the conv epilogues
_BW_STEM = 1.3e12       # stem/channel-pad paths (measured vs out bytes)
_BW_ELTW = 5.0e12       # pooling / elementwise kernels


def node_times(graph: LayerGraph, input_shape=(1, 224, 224, 3),
               bytes_per_elem: float = 2.0) -> Tuple[dict, dict, dict]:
    """Per-node (flops, output_bytes, est_time_us) for one input.

    The time model folds in each kernel class's measured efficiency so
    the cut chooser balances real stage times, not raw FLOPs (a 1x1 conv
    runs ~2-3x slower per FLOP than a 3x3 here — both models' hot
    kernels are documented in profiles/README.md).
    """
    from defer_amd.models import layers as L

    shapes = _trace_shapes(graph, input_shape)
    flops, out_bytes, time_us = {}, {}, {}
    env_shapes = {LayerGraph.INPUT: tuple(input_shape)}

    def nbytes(shape):
        return float(torch.tensor(shape).prod()) * bytes_per_elem

    for n in graph.nodes:
        out = shapes[n.name]
        f = 0.0
        lay = n.layer
        in_b = sum(nbytes(env_shapes[p]) for p in n.inputs
                   if p in env_shapes)
        out_b = nbytes(out)
        env_shapes[n.name] = out
        if isinstance(lay, L.ConvBNAct):
            f = float(out[0] * out[1] * out[2]) * lay.flops_per_pixel()
            if lay.cin < 64:
                t = max(f / _EFF_STEM, out_b / _BW_STEM)
            elif lay.kernel >= 3:
                t = max(f / _EFF_3X3, (in_b + out_b) / _BW_STREAM)
            else:
                t = max(f / _EFF_1X1, (in_b + out_b) / _BW_STREAM)
        elif isinstance(lay, nn.Conv2d):
            # FX-imported models carry plain torch convs (NCHW):
            # out = (B, C, H, W)
            kh, kw = lay.kernel_size
            f = (2.0 * float(out[0] * out[2] * out[3])
                 * kh * kw * (lay.in_channels // lay.groups)
                 * lay.out_channels)
            if lay.in_channels < 64:
                t = max(f / _EFF_STEM, out_b / _BW_STEM)
            elif max(kh, kw) >= 3:
                t = max(f / _EFF_3X3, (in_b + out_b) / _BW_STREAM)
            else:
                t = max(f / _EFF_1X1, (in_b + out_b) / _BW_STREAM)
        elif isinstance(lay, L.Dense):
            f = 2.0 * out[0] * lay.cin * lay.cout
            t = max(f / _EFF_1X1, (in_b + out_b) / _BW_STREAM)
        elif isinstance(lay, nn.Linear):
            f = 2.0 * float(out[0]) * lay.in_features * lay.out_features
            t = max(f / _EFF_1X1, (in_b + out_b) / _BW_STREAM)
        else:
            f = 2.0 * float(torch.tensor(out).prod())
            t = (in_b + out_b) / _BW_ELTW
        flops[n.name] = f
        out_bytes[n.name] = out_b
        time_us[n.name] = t * 1e6
    return flops, out_bytes, time_us


def auto_partition(model, num_stages: int, input_shape=(1, 224, 224, 3),
                   bytes_per_elem: float = 2.0,
                   eff_tflops: float = EFF_TFLOPS,
                   link_gbps: float = XGMI_LINK_GBPS,
                   measured_us: dict = None,
                   ) -> Tuple[List[str], List[GraphModel]]:
    """Choose num_stages-1 cuts minimizing the pipeline bottleneck.

    Bottleneck model (comm overlapped with compute on a side stream):
        stage_time_i = max(sum(flops_i)/eff, in_bytes_i/bw, out_bytes_i/bw)
    Minimized exactly by binary search on the bottleneck + greedy
    feasibility check over the valid articulation points.

    `measured_us` ({node_name: us per image}, from parallel.calibrate)
    replaces the static per-node time estimates with on-GPU measured
    ones; nodes absent from the dict keep their model estimate.
    """
    gm = as_graph_model(model)
    graph = gm.graph
    if num_stages <= 1:
        return [], [gm]
    flops, out_bytes, time_us = node_times(graph, input_shape,
                                           bytes_per_elem)
    if measured_us:
        time_us = {n: measured_us.get(n, t) for n, t in time_us.items()}
    cuts_avail = graph.valid_cut_points()
    if len(cuts_avail) < num_stages - 1:
        raise ValueError(
            f"graph has only {len(cuts_avail)} articulation points; cannot "
            f"make {num_stages} stages")
    pos = {n.name: i for i, n in enumerate(graph.nodes)}
    cut_pos = sorted(pos[c] for c in cuts_avail)
    names = [n.name for n in graph.nodes]
    # prefix sums of compute time (us per image) per node — the
    # kernel-class-aware estimates; eff_tflops rescales them only if a
    # caller overrides the default (relative balance is what matters)
    t_node = [time_us[n] * (EFF_TFLOPS / eff_tflops) for n in names]
    prefix = [0.0]
    for t in t_node:
        prefix.append(prefix[-1] + t)
    hop_t = {i: out_bytes[names[i]] / (link_gbps * 1e3)  # us per image
             for i in cut_pos}

    # Exact DP, lexicographic objective: minimize the pipeline bottleneck
    # max over stages of max(stage compute, in-hop, out-hop); tie-break on
    # sum of squared stage times (balance — empty stages are wasted GPUs
    # once codec/launch overheads enter).
    ncp = len(cut_pos)
    INF = float("inf")

    def stage_time(start_node: int, end_node: int) -> float:
        return prefix[end_node + 1] - prefix[start_node]

    # state: (number of cuts placed, index into cut_pos of last cut)
    # value: (bottleneck, sumsq); parent pointer for reconstruction
    best = {}
    for j, i in enumerate(cut_pos):
        t = stage_time(0, i)
        b = max(t, hop_t[i])
        best[(1, j)] = (b, t * t, None)
    for s in range(2, num_stages):
        for j, i in enumerate(cut_pos):
            cur = (INF, INF, None)
            for pj in range(j):
                if (s - 1, pj) not in best:
                    continue
                pb, psq, _ = best[(s - 1, pj)]
                t = stage_time(cut_pos[pj] + 1, i)
                b = max(pb, t, hop_t[i])
                cand = (b, psq + t * t, pj)
                if (cand[0], cand[1]) < (cur[0], cur[1]):
                    cur = cand
            if cur[0] < INF:
                best[(s, j)] = cur
    final = (INF, INF, None)
    for j, i in enumerate(cut_pos):
        if (num_stages - 1, j) not in best:
            continue
        pb, psq, _ = best[(num_stages - 1, j)]
        t = stage_time(i + 1, len(names) - 1)
        cand = (max(pb, t), psq + t * t, j)
        if (cand[0], cand[1]) < (final[0], final[1]):
            final = cand
    if final[2] is None:
        raise RuntimeError("auto_partition failed to find a feasible split")
    # reconstruct
    cuts_idx = []
    j = final[2]
    for s in range(num_stages - 1, 0, -1):
        cuts_idx.append(cut_pos[j])
        j = best[(s, j)][2]
    cuts_idx.reverse()
    cut_names = [names[i] for i in cuts_idx]
    return cut_names, gm.stage_models(cut_names)
