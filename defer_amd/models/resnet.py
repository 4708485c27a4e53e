"""ResNet-50/101/152 built as defer_amd LayerGraphs.

The reference's benchmark model is Keras ResNet50, include_top=True,
224x224x3 (test/test.py:14). The residual-add layers are named add_1 ..
add_16 so the reference's partition lists (add_2, add_4, ... add_14 —
test/test.py:18) name the same cut points here. The 101/152 variants
(deeper conv4_x/conv3_x stages, same bottleneck blocks) exercise the
partitioner and pipeline at greater depth.
"""

from typing import List

from defer_amd.graph import GraphNode, LayerGraph, GraphModel
from defer_amd.models.layers import (AddAct, ConvBNAct, Dense,
                                     GlobalAvgPool, MaxPool, Softmax)


def _resnet(depths, name, num_classes=1000, include_top=True) -> GraphModel:
    nodes: List[GraphNode] = []
    add_idx = [0]

    def N(name, layer, inputs):
        nodes.append(GraphNode(name, layer, inputs))
        return name

    # --- stem -------------------------------------------------------------
    x = N("conv1", ConvBNAct(3, 64, kernel=7, stride=2, padding=3,
                             act="relu"), ["input"])
    x = N("pool1", MaxPool(3, 2, 1), [x])

    # --- residual stages ----------------------------------------------------
    def bottleneck(x, stage, block, cin, width, stride):
        cout = width * 4
        p = f"res{stage}_{block}"
        a = N(f"{p}_conv1", ConvBNAct(cin, width, 1, 1, 0, "relu"), [x])
        b = N(f"{p}_conv2", ConvBNAct(width, width, 3, stride, 1, "relu"),
              [a])
        c = N(f"{p}_conv3", ConvBNAct(width, cout, 1, 1, 0, "none"), [b])
        if stride != 1 or cin != cout:
            sc = N(f"{p}_proj", ConvBNAct(cin, cout, 1, stride, 0, "none"),
                   [x])
        else:
            sc = x
        add_idx[0] += 1
        return N(f"add_{add_idx[0]}", AddAct("relu"), [c, sc]), cout

    cin = 64
    widths = (64, 128, 256, 512)
    for stage, (blocks, width, stride) in enumerate(
            [(depths[i], widths[i], 1 if i == 0 else 2)
             for i in range(4)], start=2):
        for b in range(blocks):
            x, cin = bottleneck(x, stage, b, cin, width,
                                stride if b == 0 else 1)

    # --- head ---------------------------------------------------------------
    if include_top:
        x = N("avg_pool", GlobalAvgPool(), [x])
        x = N("fc1000", Dense(cin, num_classes), [x])
        x = N("softmax", Softmax(), [x])

    return GraphModel(LayerGraph(nodes, output=x), name=name)


def resnet50(num_classes: int = 1000, include_top: bool = True):
    return _resnet((3, 4, 6, 3), "resnet50", num_classes, include_top)


def resnet101(num_classes: int = 1000, include_top: bool = True):
    return _resnet((3, 4, 23, 3), "resnet101", num_classes, include_top)


def resnet152(num_classes: int = 1000, include_top: bool = True):
    return _resnet((3, 8, 36, 3), "resnet152", num_classes, include_top)


# The reference's 8-stage partition (7 cuts at every other residual add,
# test/test.py:18).
DEFER_8STAGE_CUTS = ["add_2", "add_4", "add_6", "add_8", "add_10",
                     "add_12", "add_14"]
