"""Partition / graph visualization — parity with the reference's
`plot_model(model, to_file="model_{ip}.png")` on each compute node
(reference: node.py:39). Graphviz is not a dependency here; the engine
emits DOT source (renderable with any graphviz) and a plain-text per-stage
summary instead, per SURVEY.md §2.2 ("per-stage text/DOT dump").
"""

from typing import List, Optional

import torch.nn as nn

from defer_amd.graph import GraphModel, LayerGraph


def _layer_label(layer) -> str:
    if isinstance(layer, nn.Module):
        return layer.__class__.__name__
    return getattr(layer, "__name__", "fn")


def _param_count(layer) -> int:
    if isinstance(layer, nn.Module):
        return sum(p.numel() for p in layer.parameters())
    return 0


def to_dot(graph: LayerGraph, name: str = "stage",
           cut_points: Optional[List[str]] = None) -> str:
    """DOT source for a LayerGraph. Cut-point nodes (stage boundaries)
    are drawn doubled, mirroring what the reference's per-node PNG shows
    about where the model was split."""
    cuts = set(cut_points or ())
    lines = [f'digraph "{name}" {{',
             "  rankdir=TB;",
             '  node [shape=box, fontname="monospace"];',
             f'  "{LayerGraph.INPUT}" [shape=ellipse];']
    for n in graph.nodes:
        shape = ', peripheries=2' if n.name in cuts else ""
        lines.append(
            f'  "{n.name}" [label="{n.name}\\n{_layer_label(n.layer)}"'
            f"{shape}];")
        for p in n.inputs:
            lines.append(f'  "{p}" -> "{n.name}";')
    lines.append(f'  "{graph.output}" -> "output";')
    lines.append('  "output" [shape=ellipse];')
    lines.append("}")
    return "\n".join(lines) + "\n"


def describe(graph: LayerGraph, name: str = "stage") -> str:
    """Plain-text stage summary: layer table + totals (the text analogue
    of the reference's model_{ip}.png, node.py:39)."""
    rows = []
    total = 0
    for n in graph.nodes:
        p = _param_count(n.layer)
        total += p
        rows.append((n.name, _layer_label(n.layer),
                     ",".join(n.inputs), p))
    w0 = max((len(r[0]) for r in rows), default=4)
    w1 = max((len(r[1]) for r in rows), default=5)
    w2 = max((len(r[2]) for r in rows), default=6)
    out = [f"# {name}: {len(graph.nodes)} layers, {total:,} params",
           f"{'name':<{w0}}  {'layer':<{w1}}  {'inputs':<{w2}}  params"]
    for r in rows:
        out.append(f"{r[0]:<{w0}}  {r[1]:<{w1}}  {r[2]:<{w2}}  {r[3]:,}")
    return "\n".join(out) + "\n"


def dump_partition(stages: List[GraphModel], directory: str,
                   cut_points: Optional[List[str]] = None) -> List[str]:
    """Write per-stage DOT + text dumps (stage_i.dot / stage_i.txt) plus a
    whole-partition overview — one artifact per stage like the reference's
    one PNG per compute node. Returns the written paths."""
    import os

    os.makedirs(directory, exist_ok=True)
    paths = []
    for i, s in enumerate(stages):
        nm = getattr(s, "model_name", f"stage{i}")
        for ext, fn in (("dot", to_dot), ("txt", describe)):
            p = os.path.join(directory, f"stage_{i}.{ext}")
            with open(p, "w") as f:
                f.write(fn(s.graph, name=nm))
            paths.append(p)
    ov = os.path.join(directory, "partition.txt")
    with open(ov, "w") as f:
        f.write(f"cuts: {list(cut_points or ())}\n")
        for i, s in enumerate(stages):
            total = sum(p.numel() for p in s.parameters())
            f.write(f"stage {i}: {len(s.graph.nodes)} layers, "
                    f"{total:,} params, output={s.graph.output}\n")
    paths.append(ov)
    return paths
