"""Property-based tests of the LayerGraph partitioner core (the dag_util
equivalent): on random skip-connected DAGs,

- `valid_cut_points` must match a brute-force articulation check, and
- splitting at ANY subset of valid cuts must reproduce the whole-graph
  forward bitwise (the reference's part1..partN contract,
  dispatcher.py:27-42).
"""

import torch
import torch.nn as nn
from hypothesis import given, settings, strategies as st

from defer_amd.graph import GraphModel, GraphNode, LayerGraph


def _build(n_nodes: int, skips: list) -> LayerGraph:
    """Chain of Linear(4,4) nodes; skips[i] = (src, dst) adds an extra
    edge via an add node after dst."""
    torch.manual_seed(n_nodes * 31 + len(skips))
    nodes = [GraphNode("n0", nn.Linear(4, 4), ["input"])]
    for i in range(1, n_nodes):
        nodes.append(GraphNode(f"n{i}", nn.Linear(4, 4), [f"n{i-1}"]))
    g_nodes = list(nodes)
    # splice add nodes (skip connections) after their dst position
    for j, (src, dst) in enumerate(skips):
        name = f"add{j}"
        # insert after dst's current position
        pos = max(i for i, n in enumerate(g_nodes)
                  if n.name == f"n{dst}")
        prev = g_nodes[pos].name
        g_nodes.insert(pos + 1,
                       GraphNode(name, lambda a, b: a + b,
                                 [prev, f"n{src}"]))
        # rewire the following node to consume the add
        if pos + 2 < len(g_nodes):
            nxt = g_nodes[pos + 2]
            g_nodes[pos + 2] = GraphNode(
                nxt.name, nxt.layer,
                [name if p == prev else p for p in nxt.inputs],
                nxt.kwargs)
    return LayerGraph(g_nodes)


def _brute_force_cuts(g: LayerGraph):
    """A cut after node i is valid iff no earlier node's output (or the
    input) is consumed after i."""
    pos = {n.name: i for i, n in enumerate(g.nodes)}
    cuts = []
    for i in range(len(g.nodes) - 1):
        ok = True
        for n in g.nodes:
            if pos[n.name] <= i:
                continue
            for p in n.inputs:
                if p == LayerGraph.INPUT or pos[p] < i:
                    ok = False
        if ok:
            cuts.append(g.nodes[i].name)
    return cuts


@st.composite
def graphs(draw):
    n = draw(st.integers(min_value=2, max_value=10))
    n_skips = draw(st.integers(min_value=0, max_value=3))
    skips = []
    for _ in range(n_skips):
        src = draw(st.integers(min_value=0, max_value=n - 2))
        dst = draw(st.integers(min_value=src + 1, max_value=n - 1))
        skips.append((src, dst))
    return _build(n, skips)


@settings(max_examples=60, deadline=None)
@given(graphs(), st.data())
def test_valid_cuts_match_brute_force_and_split_is_exact(g, data):
    assert g.valid_cut_points() == _brute_force_cuts(g)

    valid = g.valid_cut_points()
    if valid:
        k = data.draw(st.integers(min_value=1, max_value=len(valid)))
        idx = sorted(data.draw(
            st.lists(st.integers(0, len(valid) - 1), min_size=k,
                     max_size=k, unique=True)))
        cuts = [valid[i] for i in idx]
        stages = g.split(cuts)
        assert len(stages) == len(cuts) + 1
        x = torch.randn(3, 4)
        with torch.no_grad():
            want = g.forward(x)
            z = x
            for s in stages:
                z = s.forward(z)
        assert torch.equal(z, want)


@settings(max_examples=30, deadline=None)
@given(graphs())
def test_graphmodel_parameters_partition_cleanly(g):
    """Every parameter of the whole model appears in exactly one stage
    (weights ship once, dispatcher.py:57)."""
    gm = GraphModel(g)
    valid = g.valid_cut_points()
    if not valid:
        return
    stages = [GraphModel(s) for s in g.split([valid[len(valid) // 2]])]
    total = sum(p.numel() for p in gm.parameters())
    split_total = sum(p.numel() for s in stages for p in s.parameters())
    assert split_total == total


@settings(max_examples=40, deadline=None)
@given(st.integers(1, 4), st.integers(0, 2**31 - 1), st.booleans())
def test_fusion_preserves_forward_and_names(nblocks, seed, act_none):
    """fuse_residual_adds: conv(act=none)+AddAct pairs collapse into the
    fused module without changing the forward result or losing the
    add_N names (the partition cut points)."""
    from defer_amd.models.layers import AddAct, ConvBNAct
    from defer_amd.parallel.fusion import fuse_residual_adds

    torch.manual_seed(seed % (2**31))
    nodes = [GraphNode("c0", ConvBNAct(8, 16, 3, 1, 1, "relu"),
                       ["input"])]
    x_name, cin = "c0", 16
    for b in range(nblocks):
        conv = ConvBNAct(cin, cin, 3, 1, 1,
                         "none" if not act_none else "relu")
        nodes.append(GraphNode(f"b{b}_conv", conv, [x_name]))
        nodes.append(GraphNode(f"add_{b}",
                               AddAct("relu"),
                               [f"b{b}_conv", x_name]))
        x_name = f"add_{b}"
    g = LayerGraph(nodes)
    fused = fuse_residual_adds(g)
    names = [n.name for n in fused.nodes]
    for b in range(nblocks):
        assert f"add_{b}" in names          # cut names preserved
    x = torch.randn(2, 12, 12, 8)
    with torch.no_grad():
        a = g.forward(x)
        b = fused.forward(x)
    assert torch.allclose(a, b, atol=1e-5), float((a - b).abs().max())


@st.composite
def random_dag_models(draw):
    """A random single-input CNN DAG as a plain nn.Module: features are
    produced by conv / add / cat / relu ops over earlier features (same
    spatial size throughout so every combination is shape-legal)."""
    n_ops = draw(st.integers(min_value=2, max_value=8))
    chans = [3]                      # channels of feats[i]
    spec = []                        # (op, operand indices)
    conv_shapes = []
    for _ in range(n_ops):
        op = draw(st.sampled_from(["conv", "add", "cat", "relu"]))
        if op == "conv":
            src = draw(st.integers(0, len(chans) - 1))
            cout = draw(st.sampled_from([4, 6, 8]))
            conv_shapes.append((chans[src], cout))
            spec.append(("conv", (src, len(conv_shapes) - 1)))
            chans.append(cout)
        elif op == "add":
            src_a = draw(st.integers(0, len(chans) - 1))
            peers = [i for i, c in enumerate(chans)
                     if c == chans[src_a]]
            src_b = draw(st.sampled_from(peers))
            spec.append(("add", (src_a, src_b)))
            chans.append(chans[src_a])
        elif op == "cat":
            src_a = draw(st.integers(0, len(chans) - 1))
            src_b = draw(st.integers(0, len(chans) - 1))
            spec.append(("cat", (src_a, src_b)))
            chans.append(chans[src_a] + chans[src_b])
        else:
            src = draw(st.integers(0, len(chans) - 1))
            spec.append(("relu", (src,)))
            chans.append(chans[src])
    seed = draw(st.integers(0, 2**31 - 1))
    return spec, conv_shapes, seed


def _build_dag_module(spec, conv_shapes, seed):
    torch.manual_seed(seed)

    class RandNet(nn.Module):
        def __init__(self):
            super().__init__()
            self.convs = nn.ModuleList(
                [nn.Conv2d(ci, co, 3, padding=1)
                 for ci, co in conv_shapes])

        def forward(self, x):
            feats = [x]
            for op, args in spec:
                if op == "conv":
                    feats.append(self.convs[args[1]](feats[args[0]]))
                elif op == "add":
                    feats.append(feats[args[0]] + feats[args[1]])
                elif op == "cat":
                    feats.append(torch.cat([feats[args[0]],
                                            feats[args[1]]], dim=1))
                else:
                    feats.append(torch.relu(feats[args[0]]))
            return feats[-1].mean(dim=(2, 3))

    return RandNet().eval()


@settings(max_examples=40, deadline=None)
@given(random_dag_models())
def test_from_torch_random_dags_roundtrip_and_split(model_spec):
    """from_torch on arbitrary conv/add/cat/relu DAGs reproduces the
    module forward bitwise, and splitting at any valid cut point keeps
    it bitwise — the 'partition any model' contract, fuzzed."""
    from defer_amd.graph import GraphModel, from_torch

    spec, conv_shapes, seed = model_spec
    net = _build_dag_module(spec, conv_shapes, seed)
    g = from_torch(net)
    gm = GraphModel(g)
    torch.manual_seed(seed ^ 0x5EED)
    x = torch.randn(2, 3, 6, 6)
    with torch.no_grad():
        want = net(x)
        got = gm(x)
    assert torch.equal(got, want)

    cuts = g.valid_cut_points()
    if cuts:
        mid = cuts[len(cuts) // 2]
        with torch.no_grad():
            z = x
            for s in g.split([mid]):
                z = s.forward(z)
        assert torch.equal(z, want)

        # auto_partition must also handle the FX graph (cost model over
        # plain convs + cat/add/relu function layers)
        from defer_amd.parallel.partitioner import auto_partition

        _, stages = auto_partition(g, 2, input_shape=(1, 3, 6, 6))
        with torch.no_grad():
            z = x
            for s in stages:
                z = s(z)
        assert torch.equal(z, want)
