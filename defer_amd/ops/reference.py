"""Pure-PyTorch fp32 reference implementations of every defer_amd op.

These are the ground truth that the HIP kernels are tested against
(numerics tests compare each gfx950 kernel to these at fp32), and the CPU
execution path for the plumbing config (BASELINE.json config 1 — the
reference's test/local_infer.py analogue).

All tensor ops use NHWC layout (the layout the HIP kernels use — NHWC makes
the implicit-GEMM K dimension (r, s, c) contiguous per (r, s) slice).
Weights for conv are [R, S, Cin, Cout] ("RSCK"); dense weights are
[Cin, Cout].
"""

from typing import Optional

import torch
import torch.nn.functional as F


def conv2d_bn_act(
    x: torch.Tensor,            # [N, H, W, C]   activation
    w: torch.Tensor,            # [K, R, S, C]   weights (OHWI)
    scale: Optional[torch.Tensor],  # [K] folded BN scale (None -> 1)
    bias: Optional[torch.Tensor],   # [K] folded BN shift / conv bias
    stride: int = 1,
    padding: int = 0,
    act: str = "none",          # "none" | "relu"
    residual: Optional[torch.Tensor] = None,  # [N, OH, OW, K] added pre-act
) -> torch.Tensor:
    """Fused conv + (folded) batchnorm + residual-add + activation.

    Mirrors what the reference gets from Keras Conv2D + BatchNormalization +
    Add + ReLU layers executed inside model.predict (node.py:106)."""
    xf = x.permute(0, 3, 1, 2).float()                 # NHWC -> NCHW
    wf = w.permute(0, 3, 1, 2).float()                 # OHWI -> OIHW
    y = F.conv2d(xf, wf, bias=None, stride=stride, padding=padding)
    if scale is not None:
        y = y * scale.float().view(1, -1, 1, 1)
    if bias is not None:
        y = y + bias.float().view(1, -1, 1, 1)
    y = y.permute(0, 2, 3, 1)                          # NCHW -> NHWC
    if residual is not None:
        y = y + residual.float()
    if act == "relu":
        y = torch.relu(y)
    return y.to(x.dtype)


def batchnorm_apply(x, scale, bias, act="none"):
    """Standalone inference batchnorm (y = x*scale + bias), NHWC.

    scale/bias are the folded gamma/sqrt(var+eps), beta - mean*scale."""
    y = x.float() * scale.float() + bias.float()
    if act == "relu":
        y = torch.relu(y)
    return y.to(x.dtype)


def add_act(a, b, act="relu"):
    """Residual add + activation (the reference's `add_N` layers +
    following ReLU — ResNet50 skip connections, test/test.py:18)."""
    y = a.float() + b.float()
    if act == "relu":
        y = torch.relu(y)
    return y.to(a.dtype)


def relu(x):
    return torch.relu(x)


def maxpool2d(x, kernel=3, stride=2, padding=1):
    """Max pooling, NHWC (ResNet50 stem)."""
    xf = x.permute(0, 3, 1, 2).float()
    y = F.max_pool2d(xf, kernel_size=kernel, stride=stride, padding=padding)
    return y.permute(0, 2, 3, 1).to(x.dtype)


def avgpool2d(x, kernel=2, stride=2, padding=0):
    """Average pooling, NHWC, valid taps only (count_include_pad=False;
    DenseNet transition layers)."""
    xf = x.permute(0, 3, 1, 2).float()
    y = F.avg_pool2d(xf, kernel_size=kernel, stride=stride,
                     padding=padding, count_include_pad=False)
    return y.permute(0, 2, 3, 1).to(x.dtype)


def concat_channels(xs):
    """Concat NHWC tensors on the channel dim (DenseNet)."""
    return torch.cat(list(xs), dim=-1)


def global_avg_pool(x):
    """[N, H, W, C] -> [N, C] (ResNet50 head)."""
    return x.float().mean(dim=(1, 2)).to(x.dtype)


def linear(x, w, bias=None):
    """[N, Cin] @ [Cout, Cin]^T + bias (classifier head)."""
    y = x.float() @ w.float().t()
    if bias is not None:
        y = y + bias.float()
    return y.to(x.dtype)


def softmax(x):
    """Row softmax over the last dim (classifier head)."""
    return F.softmax(x.float(), dim=-1).to(x.dtype)
