"""DenseNet-121 as a defer_amd LayerGraph — a concat-DAG model family
beyond the reference's benchmarked ResNet50/VGG19 (the reference
partitions arbitrary Keras DAGs, dag_util.py:15-21 handles multi-parent
joins; here the dense connections exercise the same machinery in the
in-repo zoo: every cumulative concat is an articulation point, so the
auto-partitioner can cut inside dense blocks, not just at transitions).

Structure follows torchvision densenet121 (growth 32, bn_size 4, blocks
6/12/24/16, pre-activation BN-ReLU-Conv composites, 2x2/2 avg-pool
transitions): stem 7x7/2 -> [block, transition]x3 -> block -> norm5 ->
GAP -> fc1000 -> softmax. Channels: 64 -> 256/128 -> 512/256 -> 1024/512
-> 1024.

Layer names: denseB_Lx (x = c1 bottleneck, c2 3x3, cat), transB_*,
norm5, fc, softmax.
"""

from typing import List

from defer_amd.graph import GraphNode, LayerGraph, GraphModel
from defer_amd.models.layers import (AvgPool, BNAct, BNActConv, Concat,
                                     ConvBNAct, Dense, GlobalAvgPool,
                                     MaxPool, Softmax)

_BLOCKS = [6, 12, 24, 16]
_GROWTH = 32
_BN_SIZE = 4


def densenet121(num_classes: int = 1000) -> GraphModel:
    nodes: List[GraphNode] = []

    def N(name, layer, inputs):
        nodes.append(GraphNode(name, layer, inputs))
        return name

    # stem (post-activation, torchvision conv0/norm0/relu0/pool0)
    x = N("conv1", ConvBNAct(3, 64, 7, 2, 3, "relu"), ["input"])
    x = N("pool1", MaxPool(3, 2, 1), [x])
    c = 64

    for b, nlayers in enumerate(_BLOCKS, start=1):
        for j in range(1, nlayers + 1):
            # c1 owns BOTH folded BNs: norm1 on its input staging and
            # norm2 on its fp32 output accumulator (gemm_prebn_kernel),
            # so conv2 runs as a plain 3x3 — no standalone bn_act pass
            # anywhere in the dense layer
            f = N(f"dense{b}_{j}c1",
                  BNActConv(c, _BN_SIZE * _GROWTH, 1, 1, 0,
                            out_bn=True), [x])
            f = N(f"dense{b}_{j}c2",
                  ConvBNAct(_BN_SIZE * _GROWTH, _GROWTH, 3, 1, 1,
                            act="none", bn=False), [f])
            x = N(f"dense{b}_{j}cat", Concat(), [x, f])
            c += _GROWTH
        if b < len(_BLOCKS):
            t = N(f"trans{b}_conv", BNActConv(c, c // 2, 1, 1, 0), [x])
            x = N(f"trans{b}_pool", AvgPool(2, 2, 0), [t])
            c //= 2

    x = N("norm5", BNAct(c), [x])
    x = N("gap", GlobalAvgPool(), [x])
    x = N("fc", Dense(c, num_classes), [x])
    x = N("softmax", Softmax(), [x])
    return GraphModel(LayerGraph(nodes, output=x), name="densenet121")
