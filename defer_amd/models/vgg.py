"""VGG19 as a defer_amd LayerGraph (BASELINE.json config 5).

Larger boundary activations than ResNet50 (e.g. 112x112x128 after block2)
— stresses the xGMI send/recv relay and the ZFP/LZ4 codec path.
Layer names follow Keras VGG19: blockB_convC, blockB_pool, fc1/fc2,
predictions.
"""

from typing import List

from defer_amd.graph import GraphNode, LayerGraph, GraphModel
from defer_amd.models.layers import (ConvBNAct, Dense, GlobalAvgPool,
                                     MaxPool, Softmax)

_CFG = [(2, 64), (2, 128), (4, 256), (4, 512), (4, 512)]


def vgg19(num_classes: int = 1000, include_top: bool = True) -> GraphModel:
    nodes: List[GraphNode] = []

    def N(name, layer, inputs):
        nodes.append(GraphNode(name, layer, inputs))
        return name

    x = "input"
    cin = 3
    for bi, (convs, width) in enumerate(_CFG, start=1):
        for ci in range(1, convs + 1):
            x = N(f"block{bi}_conv{ci}",
                  ConvBNAct(cin, width, 3, 1, 1, "relu", bn=False), [x])
            cin = width
        x = N(f"block{bi}_pool", MaxPool(2, 2, 0), [x])

    if include_top:
        # Keras VGG19 flattens 7x7x512 into two 4096 FC layers; global-avg
        # + dense keeps the graph single-tensor and the params sane for a
        # random-init throughput benchmark; fc sizes follow the reference.
        x = N("gap", GlobalAvgPool(), [x])
        x = N("fc1", Dense(cin, 4096, act="relu"), [x])
        x = N("fc2", Dense(4096, 4096, act="relu"), [x])
        x = N("predictions", Dense(4096, num_classes), [x])
        x = N("softmax", Softmax(), [x])

    return GraphModel(LayerGraph(nodes, output=x), name="vgg19")
