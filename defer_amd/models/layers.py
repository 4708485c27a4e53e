"""Layer modules for defer_amd models.

Each layer is a thin nn.Module over `defer_amd.ops` (HIP kernels on GPU,
fp32 PyTorch reference on CPU). These are the in-repo replacements for the
Keras layer zoo the reference executes via model.predict (node.py:106):
Conv2D/BatchNorm/ReLU -> ConvBNAct (BN folded at init), Add(+ReLU) ->
AddAct, MaxPooling2D -> MaxPool, GlobalAveragePooling2D -> GlobalAvgPool,
Dense -> Dense, softmax -> Softmax.

Activations are NHWC; conv weights are [Cout, R, S, Cin] (OHWI), chosen so
the implicit-GEMM K dimension (r, s, c) is contiguous per output channel:
the MFMA kernel's B^T fragment read is then a dense row read, and the
global->LDS staging of a [BN][BK] weight tile is fully coalesced.
"""

import math

import torch
import torch.nn as nn

from defer_amd import ops


class ConvBNAct(nn.Module):
    """Conv2d + folded inference BatchNorm + optional ReLU, fused.

    BN folding: y = conv(x) * scale + bias with scale = gamma/sqrt(var+eps),
    bias = beta - mean*scale. Random init keeps BN-ish statistics so
    activations stay O(1) through deep stacks (bench uses random-init
    weights per BASELINE.json).
    """

    def __init__(self, cin, cout, kernel=3, stride=1, padding=None,
                 act="relu", bn=True):
        super().__init__()
        self.cin, self.cout = cin, cout
        self.kernel, self.stride = kernel, stride
        self.padding = (kernel // 2) if padding is None else padding
        self.act = act
        fan_in = cin * kernel * kernel
        w = torch.randn(cout, kernel, kernel, cin) * math.sqrt(2.0 / fan_in)
        self.weight = nn.Parameter(w)
        self.scale = nn.Parameter(torch.ones(cout)) if bn else None
        self.bias = nn.Parameter(torch.zeros(cout))

    def forward(self, x, residual=None):
        return ops.conv2d_bn_act(
            x, self.weight.to(x.dtype), self.scale, self.bias,
            stride=self.stride, padding=self.padding, act=self.act,
            residual=residual)

    def extra_repr(self):
        return (f"{self.cin}->{self.cout} k{self.kernel} s{self.stride} "
                f"p{self.padding} act={self.act}")

    def flops_per_pixel(self):
        return 2 * self.kernel * self.kernel * self.cin * self.cout


class AddAct(nn.Module):
    """Residual add + activation — the reference's `add_N` cut layers
    (test/test.py:18) with the ReLU that follows them fused in."""

    def __init__(self, act="relu"):
        super().__init__()
        self.act = act

    def forward(self, a, b):
        return ops.add_act(a, b, act=self.act)


class MaxPool(nn.Module):
    def __init__(self, kernel=3, stride=2, padding=1):
        super().__init__()
        self.kernel, self.stride, self.padding = kernel, stride, padding

    def forward(self, x):
        return ops.maxpool2d(x, self.kernel, self.stride, self.padding)


class AvgPool(nn.Module):
    """Average pooling (DenseNet transition 2x2/2)."""

    def __init__(self, kernel=2, stride=2, padding=0):
        super().__init__()
        self.kernel, self.stride, self.padding = kernel, stride, padding

    def forward(self, x):
        return ops.avgpool2d(x, self.kernel, self.stride, self.padding)


class Concat(nn.Module):
    """Channel concat of NHWC inputs (DenseNet dense connections).
    GPU path is strided device copies, no torch compute kernels."""

    def forward(self, *xs):
        return ops.concat_channels(xs)


class BNActConv(nn.Module):
    """Pre-activation composite BN -> ReLU -> Conv (DenseNet layer
    ordering; torchvision DenseLayer norm/relu/conv). Inference BN is a
    folded per-channel scale/bias applied by the standalone bn_act HIP
    kernel; the conv runs without epilogue BN."""

    def __init__(self, cin, cout, kernel=1, stride=1, padding=0,
                 out_bn=False):
        super().__init__()
        self.cin, self.cout = cin, cout
        self.kernel, self.stride, self.padding = kernel, stride, padding
        fan_in = cin * kernel * kernel
        w = torch.randn(cout, kernel, kernel, cin) \
            * math.sqrt(2.0 / fan_in)
        self.weight = nn.Parameter(w)
        self.scale = nn.Parameter(torch.ones(cin))
        self.bias = nn.Parameter(torch.zeros(cin))
        # out_bn: this layer ALSO owns the next composite's folded
        # BN-ReLU, applied to the conv output (DenseNet norm2 folds
        # into c1's epilogue; conv2 then runs as a plain 3x3)
        self.out_bn = out_bn
        if out_bn:
            self.out_scale = nn.Parameter(torch.ones(cout))
            self.out_bias = nn.Parameter(torch.zeros(cout))

    def forward(self, x):
        if self.kernel == 1 and self.stride == 1 and self.padding == 0:
            # fused single-pass path (the standalone bn_act round-trips
            # were 57% of the DenseNet step before fusion)
            return ops.conv1x1_prebn(
                x, self.weight.to(x.dtype),
                self.scale.float(), self.bias.float(),
                self.out_scale.float() if self.out_bn else None,
                self.out_bias.float() if self.out_bn else None)
        z = ops.batchnorm_apply(x, self.scale, self.bias, act="relu")
        y = ops.conv2d_bn_act(z, self.weight.to(x.dtype), None, None,
                              stride=self.stride,
                              padding=self.padding, act="none")
        if self.out_bn:
            y = ops.batchnorm_apply(y, self.out_scale, self.out_bias,
                                    act="relu")
        return y

    def extra_repr(self):
        return (f"bn-relu-conv {self.cin}->{self.cout} k{self.kernel} "
                f"s{self.stride} p{self.padding}")

    def flops_per_pixel(self):
        return 2 * self.kernel * self.kernel * self.cin * self.cout


class BNAct(nn.Module):
    """Standalone folded-BN + activation (DenseNet final norm5)."""

    def __init__(self, c, act="relu"):
        super().__init__()
        self.act = act
        self.scale = nn.Parameter(torch.ones(c))
        self.bias = nn.Parameter(torch.zeros(c))

    def forward(self, x):
        return ops.batchnorm_apply(x, self.scale, self.bias, self.act)


class GlobalAvgPool(nn.Module):
    def forward(self, x):
        return ops.global_avg_pool(x)


class Flatten(nn.Module):
    """NHWC -> (N, H*W*C). Matches Keras Flatten under channels_last
    (the reference's VGG19 head flattens 7x7x512 -> 25088); a reshape of
    a contiguous NHWC tensor, so free on GPU."""

    def forward(self, x):
        return x.reshape(x.shape[0], -1)


class Dense(nn.Module):
    def __init__(self, cin, cout, bias=True, act="none"):
        super().__init__()
        self.cin, self.cout = cin, cout
        self.act = act
        self.weight = nn.Parameter(
            torch.randn(cout, cin) * math.sqrt(1.0 / cin))
        self.bias = nn.Parameter(torch.zeros(cout)) if bias else None

    def forward(self, x):
        y = ops.linear(x, self.weight.to(x.dtype), self.bias)
        if self.act == "relu":
            y = ops.relu(y)
        return y


class Softmax(nn.Module):
    def forward(self, x):
        return ops.softmax(x)
