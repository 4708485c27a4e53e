"""LayerGraph IR: validation, articulation points, splitting."""

import pytest
import torch
import torch.nn as nn

from defer_amd.graph import GraphModel, GraphNode, LayerGraph, from_torch


def _lin(cin, cout):
    m = nn.Linear(cin, cout)
    return m


def simple_graph():
    # input -> a -> b -> (c, skip) -> add -> d
    nodes = [
        GraphNode("a", _lin(8, 8), ["input"]),
        GraphNode("b", _lin(8, 8), ["a"]),
        GraphNode("c", _lin(8, 8), ["b"]),
        GraphNode("add", lambda x, y: x + y, ["c", "b"]),
        GraphNode("d", _lin(8, 4), ["add"]),
    ]
    return LayerGraph(nodes)


def test_forward_matches_manual():
    g = simple_graph()
    x = torch.randn(3, 8)
    a = g.by_name["a"].layer(x)
    b = g.by_name["b"].layer(a)
    c = g.by_name["c"].layer(b)
    d = g.by_name["d"].layer(c + b)
    assert torch.allclose(g.forward(x), d)


def test_topological_order_enforced():
    with pytest.raises(ValueError, match="topologically"):
        LayerGraph([
            GraphNode("x", _lin(4, 4), ["y"]),
            GraphNode("y", _lin(4, 4), ["input"]),
        ])


def test_duplicate_names_rejected():
    with pytest.raises(ValueError, match="duplicate"):
        LayerGraph([
            GraphNode("x", _lin(4, 4), ["input"]),
            GraphNode("x", _lin(4, 4), ["x"]),
        ])


def test_valid_cut_points_exclude_skip_span():
    g = simple_graph()
    cuts = g.valid_cut_points()
    # b..add spans the residual: c is not an articulation point
    assert "c" not in cuts
    assert "a" in cuts and "b" in cuts and "add" in cuts
    # output node is never a cut
    assert "d" not in cuts


def test_split_at_invalid_cut_raises():
    g = simple_graph()
    with pytest.raises(ValueError, match="articulation"):
        g.split(["c"])
    with pytest.raises(ValueError, match="not a layer"):
        g.split(["nope"])


def test_split_equivalence():
    g = simple_graph()
    x = torch.randn(5, 8)
    want = g.forward(x)
    for cuts in (["a"], ["b"], ["add"], ["a", "add"], ["a", "b", "add"]):
        stages = g.split(cuts)
        assert len(stages) == len(cuts) + 1
        z = x
        for s in stages:
            z = s.forward(z)
        assert torch.allclose(z, want), cuts


def test_from_torch_fx_frontend():
    class Net(nn.Module):
        def __init__(self):
            super().__init__()
            self.l1 = nn.Linear(8, 8)
            self.l2 = nn.Linear(8, 8)
            self.l3 = nn.Linear(8, 4)

        def forward(self, x):
            a = torch.relu(self.l1(x))
            b = self.l2(a)
            return self.l3(a + b)

    net = Net()
    g = from_torch(net)
    x = torch.randn(2, 8)
    assert torch.allclose(g.forward(x), net(x))
    # the add spans a skip; l2 output is not an articulation point
    names = g.layer_names()
    assert any("add" in n for n in names)


def test_graph_model_stage_models():
    gm = GraphModel(simple_graph())
    x = torch.randn(2, 8)
    want = gm(x)
    parts = gm.stage_models(["b"])
    z = x
    for p in parts:
        z = p(z)
    assert torch.allclose(z, want)
    # parameters are owned (state_dict round-trips)
    sd = parts[0].state_dict()
    assert any("layers.a" in k for k in sd)


def test_from_torch_nested_tensor_args():
    """Tensor inputs inside list/tuple arguments (torch.cat) and keyword
    constants resolve correctly through the FX frontend."""
    class Net(nn.Module):
        def __init__(self):
            super().__init__()
            self.l = nn.Linear(4, 4)

        def forward(self, x):
            y = self.l(x)
            z = torch.cat([y, x], dim=-1)
            return torch.clamp(z, min=-1.0, max=1.0).sum(dim=-1)

    net = Net()
    g = from_torch(net)
    x = torch.randn(3, 4)
    assert torch.allclose(g.forward(x), net(x))
