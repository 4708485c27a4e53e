"""VGG19 as a defer_amd LayerGraph (BASELINE.json config 5).

Larger boundary activations than ResNet50 (e.g. 112x112x128 after block2)
— stresses the xGMI send/recv relay and the ZFP/LZ4 codec path.
Layer names follow Keras VGG19 (the reference benchmarks published Keras
models, test/test.py:14): blockB_convC, blockB_pool, flatten, fc1/fc2,
predictions.

`vgg19` is the faithful architecture: Flatten(7x7x512 = 25088) ->
fc4096 -> fc4096 -> fc1000, so fc1 is the published 102.8 M-param
25088x4096 GEMM (a real large-K test for the MFMA GEMM path).
`vgg19_gap` keeps round 1's GlobalAvgPool(512)->fc4096 head as a compact
variant (2.1 M-param fc1).
"""

from typing import List

from defer_amd.graph import GraphNode, LayerGraph, GraphModel
from defer_amd.models.layers import (ConvBNAct, Dense, Flatten,
                                     GlobalAvgPool, MaxPool, Softmax)

_CFG = [(2, 64), (2, 128), (4, 256), (4, 512), (4, 512)]


def _vgg19(name: str, head: str, num_classes: int,
           include_top: bool) -> GraphModel:
    nodes: List[GraphNode] = []

    def N(node_name, layer, inputs):
        nodes.append(GraphNode(node_name, layer, inputs))
        return node_name

    x = "input"
    cin = 3
    for bi, (convs, width) in enumerate(_CFG, start=1):
        for ci in range(1, convs + 1):
            x = N(f"block{bi}_conv{ci}",
                  ConvBNAct(cin, width, 3, 1, 1, "relu", bn=False), [x])
            cin = width
        x = N(f"block{bi}_pool", MaxPool(2, 2, 0), [x])

    if include_top:
        if head == "flatten":
            # Keras VGG19 head: Flatten(7*7*512) -> fc4096
            x = N("flatten", Flatten(), [x])
            fc1_in = 7 * 7 * cin
        else:
            x = N("gap", GlobalAvgPool(), [x])
            fc1_in = cin
        x = N("fc1", Dense(fc1_in, 4096, act="relu"), [x])
        x = N("fc2", Dense(4096, 4096, act="relu"), [x])
        x = N("predictions", Dense(4096, num_classes), [x])
        x = N("softmax", Softmax(), [x])

    return GraphModel(LayerGraph(nodes, output=x), name=name)


def vgg19(num_classes: int = 1000, include_top: bool = True) -> GraphModel:
    return _vgg19("vgg19", "flatten", num_classes, include_top)


def vgg19_gap(num_classes: int = 1000,
              include_top: bool = True) -> GraphModel:
    return _vgg19("vgg19_gap", "gap", num_classes, include_top)
