"""LayerGraph — the DAG IR that models are built in and partitioned on.

The reference partitions a Keras DAG by recursing backward from a named end
layer to a named start layer and re-applying layer objects
(dag_util.py:9-31). Here the DAG is explicit: a `LayerGraph` is a
topologically ordered list of named nodes, each applying an `nn.Module`
layer to the outputs of named parent nodes. That makes partitioning a list
split + an articulation-point check, makes stages directly executable, and
gives stable layer names for cut points (`add_2`, ... — the reference's
contract, test/test.py:18).

`GraphModel` wraps a LayerGraph as an nn.Module (owns the parameters).
`from_torch` converts an arbitrary traceable nn.Module into a LayerGraph
via torch.fx, for capability parity with "partition any Keras model".
"""

from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Union

import torch
import torch.nn as nn


@dataclass
class GraphNode:
    name: str
    layer: Union[nn.Module, Callable]
    inputs: List[str]                     # names of parent nodes (or "input")
    # kwargs passed to the layer call (constants only)
    kwargs: dict = field(default_factory=dict)


class LayerGraph:
    """A single-input single-output DAG of named layers.

    Node order must be topological. The node named "input" is implicit: it
    is the graph's input tensor.
    """

    INPUT = "input"

    def __init__(self, nodes: List[GraphNode], output: Optional[str] = None):
        self.nodes = list(nodes)
        self.by_name: Dict[str, GraphNode] = {}
        seen = {self.INPUT}
        for n in self.nodes:
            if n.name in self.by_name or n.name == self.INPUT:
                raise ValueError(f"duplicate node name {n.name!r}")
            for p in n.inputs:
                if p not in seen:
                    raise ValueError(
                        f"node {n.name!r} input {p!r} not defined before it "
                        "(graph must be topologically ordered)")
            self.by_name[n.name] = n
            seen.add(n.name)
        self.output = output or self.nodes[-1].name
        if self.output not in self.by_name:
            raise ValueError(f"output node {self.output!r} not in graph")

    # ------------------------------------------------------------------ run
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        env: Dict[str, torch.Tensor] = {self.INPUT: x}
        # Reference-count consumers so intermediates free eagerly (288 GB of
        # HBM is generous but a 16-block ResNet keeps nothing it must not).
        remaining: Dict[str, int] = {}
        for n in self.nodes:
            for p in n.inputs:
                remaining[p] = remaining.get(p, 0) + 1
        remaining[self.output] = remaining.get(self.output, 0) + 1
        for n in self.nodes:
            args = [env[p] for p in n.inputs]
            env[n.name] = n.layer(*args, **n.kwargs)
            for p in n.inputs:
                remaining[p] -= 1
                if remaining[p] == 0:
                    del env[p]
        return env[self.output]

    # ---------------------------------------------------------- partitioning
    def valid_cut_points(self) -> List[str]:
        """Names of nodes whose single output is the only tensor crossing
        from {nodes up to and incl. it} to {nodes after it} — the
        articulation contract the reference imposes on start/end layers
        (dag_util.py:27-31; ResNet50's add_N satisfy it, test/test.py:18)."""
        pos = {n.name: i for i, n in enumerate(self.nodes)}
        pos[self.INPUT] = -1
        # farthest-forward consumer of each node's output
        last_use = {name: i for name, i in pos.items()}
        for n in self.nodes:
            for p in n.inputs:
                last_use[p] = max(last_use[p], pos[n.name])
        cuts = []
        for i, n in enumerate(self.nodes[:-1]):
            # A cut after node i is valid iff no node <= i (except node i
            # itself) is consumed by a node > i.
            ok = all(
                last_use[m.name] <= i
                for m in self.nodes[:i]
            ) and last_use[self.INPUT] <= i
            if ok:
                cuts.append(n.name)
        return cuts

    def split(self, cut_points: List[str]) -> List["LayerGraph"]:
        """Split into len(cut_points)+1 stage graphs. Each cut point must be
        a valid articulation node (checked). Stage k's "input" is the
        previous stage's output — the reference's part1..partN contract
        (dispatcher.py:27-42)."""
        valid = set(self.valid_cut_points())
        pos = {n.name: i for i, n in enumerate(self.nodes)}
        for c in cut_points:
            if c not in pos:
                raise ValueError(f"cut point {c!r} is not a layer name")
            if c not in valid:
                raise ValueError(
                    f"cut point {c!r} is not a valid articulation point "
                    f"(a later layer consumes an earlier tensor across it)")
        idxs = sorted(pos[c] for c in cut_points)
        if len(set(idxs)) != len(idxs):
            raise ValueError("duplicate cut points")
        stages: List[LayerGraph] = []
        start = 0
        bounds = idxs + [len(self.nodes) - 1]
        for b in bounds:
            seg = self.nodes[start:b + 1]
            # remap references to the previous stage's output node -> input
            prev_out = self.nodes[start - 1].name if start > 0 else None
            remapped = [
                GraphNode(
                    n.name, n.layer,
                    [self.INPUT if p == prev_out else p for p in n.inputs],
                    dict(n.kwargs))
                for n in seg
            ]
            stages.append(LayerGraph(remapped, output=seg[-1].name))
            start = b + 1
        return stages

    def layer_names(self) -> List[str]:
        return [n.name for n in self.nodes]


class GraphModel(nn.Module):
    """nn.Module wrapper that owns a LayerGraph's modules as submodules."""

    def __init__(self, graph: LayerGraph, name: str = "model"):
        super().__init__()
        self.graph = graph
        self.model_name = name
        mods = {}
        for n in graph.nodes:
            if isinstance(n.layer, nn.Module):
                mods[n.name] = n.layer
        # ModuleDict keys cannot contain '.', graph names never do.
        self.layers = nn.ModuleDict(mods)

    def forward(self, x):
        return self.graph.forward(x)

    def stage_models(self, cut_points: List[str]) -> List["GraphModel"]:
        return [GraphModel(g, name=f"{self.model_name}.part{i+1}")
                for i, g in enumerate(self.graph.split(cut_points))]


def from_torch(model: nn.Module) -> LayerGraph:
    """FX-trace an arbitrary single-input nn.Module into a LayerGraph.

    Capability parity with partitioning a user-provided Keras model
    (dispatcher.py:107). Leaf modules become graph layers; call_function
    nodes (e.g. `x + y`, torch.relu) become function layers.
    """
    import torch.fx as fx

    class LeafTracer(fx.Tracer):
        def is_leaf_module(self, m, qualname):
            # defer_amd layer modules are opaque leaves
            if m.__class__.__module__.startswith("defer_amd"):
                return True
            return super().is_leaf_module(m, qualname)

    tracer = LeafTracer()
    g = tracer.trace(model)
    modules = dict(model.named_modules())
    nodes: List[GraphNode] = []
    name_map: Dict[str, str] = {}
    placeholder = None
    for node in g.nodes:
        if node.op == "placeholder":
            if placeholder is not None:
                raise ValueError("only single-input models are supported")
            placeholder = node.name
            name_map[node.name] = LayerGraph.INPUT
        elif node.op == "call_module":
            mod = modules[node.target]
            ins = [name_map[a.name] for a in node.args]
            nm = node.target.replace(".", "_")
            nodes.append(GraphNode(nm, mod, ins))
            name_map[node.name] = nm
        elif node.op == "call_function" or node.op == "call_method":
            fn = node.target
            if node.op == "call_method":
                meth = node.target

                def fn(x, *a, _m=meth, **k):
                    return getattr(x, _m)(*a, **k)

            # Tensor inputs can appear anywhere in the argument tree
            # (e.g. torch.cat([a, b], dim=-1)): replace every fx.Node
            # with a positional placeholder and rebuild at call time.
            class _Ref:
                __slots__ = ("i",)

                def __init__(self, i):
                    self.i = i

            refs: List[str] = []

            def scan(a):
                if isinstance(a, fx.Node):
                    refs.append(a.name)
                    return _Ref(len(refs) - 1)
                if isinstance(a, (list, tuple)):
                    return type(a)(scan(x) for x in a)
                return a

            sargs = tuple(scan(a) for a in node.args)
            skwargs = {k: scan(v) for k, v in node.kwargs.items()}
            ins = [name_map[r] for r in refs]
            nm = node.name

            def wrapped(*tensors, _fn=fn, _sa=sargs, _sk=skwargs,
                        _Ref=_Ref):
                def subst(a):
                    if isinstance(a, _Ref):
                        return tensors[a.i]
                    if isinstance(a, (list, tuple)):
                        return type(a)(subst(x) for x in a)
                    return a

                return _fn(*(subst(a) for a in _sa),
                           **{k: subst(v) for k, v in _sk.items()})

            nodes.append(GraphNode(nm, wrapped, ins))
            name_map[node.name] = nm
        elif node.op == "output":
            out_arg = node.args[0]
            return LayerGraph(nodes, output=name_map[out_arg.name])
        elif node.op == "get_attr":
            raise ValueError("get_attr nodes not supported; wrap constants "
                             "in a module")
    raise ValueError("graph had no output node")
