"""Model zoo: ResNet50 / VGG19 shapes and partition parity (the
reference's 8-stage cut list, test/test.py:18)."""

import torch

from defer_amd.models import DEFER_8STAGE_CUTS, resnet50, vgg19
from defer_amd.parallel.partitioner import (auto_partition, node_costs,
                                            partition_model)


def test_resnet50_shapes():
    m = resnet50()
    x = torch.randn(2, 64, 64, 3)  # fully-conv until GAP: small input ok
    with torch.no_grad():
        y = m(x)
    assert y.shape == (2, 1000)
    assert torch.allclose(y.float().sum(dim=-1), torch.ones(2), atol=1e-4)


def test_resnet50_defer_cut_list_is_valid():
    m = resnet50()
    cuts = set(m.graph.valid_cut_points())
    for c in DEFER_8STAGE_CUTS:
        assert c in cuts
    stages = partition_model(m, DEFER_8STAGE_CUTS)
    assert len(stages) == 8


def test_resnet50_partition_equivalence():
    m = resnet50()
    x = torch.randn(1, 64, 64, 3)
    with torch.no_grad():
        want = m(x)
        z = x
        for s in partition_model(m, DEFER_8STAGE_CUTS):
            z = s(z)
    assert torch.equal(z, want)  # identical op order -> bitwise


def test_vgg19_shapes_and_partition():
    # faithful Keras VGG19 head: Flatten(7x7x512)->fc4096 needs the real
    # 224x224 input (fc1 is the published 25088x4096 GEMM)
    m = vgg19(num_classes=10)
    p = dict(m.named_parameters())
    assert p["layers.fc1.weight"].shape == (4096, 7 * 7 * 512)
    x = torch.randn(1, 224, 224, 3)
    with torch.no_grad():
        want = m(x)
    assert want.shape == (1, 10)
    cuts, stages = auto_partition(m, 4, input_shape=(1, 224, 224, 3))
    assert len(stages) == 4
    with torch.no_grad():
        z = x
        for s in stages:
            z = s(z)
    assert torch.equal(z, want)


def test_vgg19_gap_variant():
    from defer_amd.models import vgg19_gap

    m = vgg19_gap(num_classes=10)
    assert dict(m.named_parameters())["layers.fc1.weight"].shape == (4096, 512)
    x = torch.randn(1, 64, 64, 3)  # GAP head is input-size independent
    with torch.no_grad():
        y = m(x)
    assert y.shape == (1, 10)


def test_auto_partition_resnet_8():
    m = resnet50()
    cuts, stages = auto_partition(m, 8)
    assert len(cuts) == 7 and len(stages) == 8
    # bottleneck sanity: no stage is empty of compute
    fl, _ = node_costs(m.graph)
    for s in stages:
        assert sum(fl.get(n.name, 0.0) for n in s.graph.nodes) > 0


def test_node_costs_total_flops():
    m = resnet50()
    fl, ob = node_costs(m.graph)
    total = sum(fl.values())
    # ResNet50 @224 is ~8.2 GFLOP (2*MAC) per image
    assert 7.5e9 < total < 9.0e9
    assert ob["add_1"] == 56 * 56 * 256 * 2.0


def test_fusion_preserves_semantics():
    from defer_amd.graph import GraphModel
    from defer_amd.parallel.fusion import fuse_residual_adds

    m = resnet50()
    x = torch.randn(1, 64, 64, 3)
    with torch.no_grad():
        want = m(x)
    fused = GraphModel(fuse_residual_adds(m.graph))
    with torch.no_grad():
        got = fused(x)
    assert torch.allclose(got, want, atol=1e-5)
    # all 16 residual adds fused away
    kinds = [type(n.layer).__name__ for n in fused.graph.nodes]
    assert "AddAct" not in kinds
    assert kinds.count("FusedConvAddAct") == 16
    # cut names still valid
    assert "add_8" in fused.graph.valid_cut_points()


def test_stage_executor_applies_fusion_cpu():
    from defer_amd.parallel.pipeline import StageExecutor
    from defer_amd.graph import GraphModel

    m = resnet50()
    x = torch.randn(1, 64, 64, 3)
    with torch.no_grad():
        want = m(x)
    ex = StageExecutor(GraphModel(m.graph), "cpu", torch.float32)
    with torch.no_grad():
        got = ex.run(x)
    assert torch.allclose(got, want, atol=1e-5)


def test_resnet_deep_variants():
    """ResNet-101/152: same bottleneck blocks at greater depth; the
    partitioner handles them with auto cuts."""
    from defer_amd.models import resnet101, resnet152

    for mk, nadds in ((resnet101, 33), (resnet152, 50)):
        m = mk()
        names = m.graph.layer_names()
        assert f"add_{nadds}" in names and f"add_{nadds+1}" not in names
        x = torch.randn(1, 64, 64, 3)
        with torch.no_grad():
            y = m(x)
        assert y.shape == (1, 1000)
        cuts, stages = auto_partition(m, 8, input_shape=(1, 64, 64, 3))
        assert len(stages) == 8
        with torch.no_grad():
            z = x
            for s in stages:
                z = s(z)
        assert torch.allclose(z, y)


def test_densenet121_shapes_partition_and_cuts():
    """DenseNet-121: concat-DAG model family — cumulative concats are
    articulation points, so the partitioner can cut INSIDE dense blocks
    (the reference's dag_util handles multi-parent Keras joins the same
    way, dag_util.py:15-21)."""
    from defer_amd.models import densenet121

    torch.manual_seed(0)
    m = densenet121(num_classes=10)
    x = torch.randn(1, 64, 64, 3)
    with torch.no_grad():
        want = m(x)
    assert want.shape == (1, 10)
    assert torch.allclose(want.float().sum(dim=-1), torch.ones(1),
                          atol=1e-4)
    cuts = m.graph.valid_cut_points()
    assert "dense2_6cat" in cuts and "trans1_pool" in cuts
    auto_cuts, stages = auto_partition(m, 8, input_shape=(1, 64, 64, 3))
    assert len(stages) == 8
    assert any(c.startswith("dense") for c in auto_cuts)
    with torch.no_grad():
        z = x
        for s in stages:
            z = s(z)
    assert torch.equal(z, want)


def test_avgpool_and_concat_reference_ops():
    from defer_amd import ops

    x = torch.randn(2, 8, 8, 16)
    y = ops.avgpool2d(x, 2, 2, 0)
    assert y.shape == (2, 4, 4, 16)
    import torch.nn.functional as F

    want = F.avg_pool2d(x.permute(0, 3, 1, 2), 2, 2) \
        .permute(0, 2, 3, 1)
    assert torch.allclose(y, want, atol=1e-6)
    a, b = torch.randn(2, 4, 4, 8), torch.randn(2, 4, 4, 24)
    cat = ops.concat_channels([a, b])
    assert torch.equal(cat, torch.cat([a, b], dim=-1))
