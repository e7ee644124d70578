"""NHWC layer modules on top of cyclegan_amd.ops.

Parameters are fp32 masters (OHWI for convs); compute dtype follows the
activation dtype (bf16 on MI355X), with shadow casting handled inside the
op layer.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn

from .. import ops
from ..ops.norm import EPS_DEFAULT


def init_conv_weight_(w: torch.Tensor, kind: str = "normal002"):
    """Reference inits: RandomNormal(0, 0.02) for all convs except the
    generator head, which uses Keras' default glorot_uniform
    (/root/reference/cyclegan/model.py:10-11,165-166). w is OHWI."""
    cout, kh, kw, cin = w.shape
    with torch.no_grad():
        if kind == "normal002":
            w.normal_(0.0, 0.02)
        elif kind == "glorot":
            fan_in, fan_out = kh * kw * cin, kh * kw * cout
            limit = math.sqrt(6.0 / (fan_in + fan_out))
            w.uniform_(-limit, limit)
        else:
            raise ValueError(kind)


class ConvNHWC(nn.Module):
    def __init__(self, cin: int, cout: int, kernel: int, stride: int = 1,
                 padding="same", pad_mode: str = "zeros", bias: bool = False,
                 act: Optional[str] = None, slope: float = 0.2,
                 init: str = "normal002"):
        super().__init__()
        self.stride, self.padding, self.pad_mode = stride, padding, pad_mode
        self.act, self.slope = act, slope
        self.weight = nn.Parameter(torch.empty(cout, kernel, kernel, cin))
        init_conv_weight_(self.weight, init)
        if bias:
            self.bias = nn.Parameter(torch.zeros(cout))
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        return ops.conv2d(x, self.weight, self.bias, self.stride,
                          self.padding, self.pad_mode, self.act, self.slope)


class ConvTransposeNHWC(nn.Module):
    """TF-'SAME' stride-s transpose conv: out = in * stride."""

    def __init__(self, cin: int, cout: int, kernel: int, stride: int = 2,
                 bias: bool = False, act: Optional[str] = None,
                 init: str = "normal002"):
        super().__init__()
        self.stride, self.act = stride, act
        self.weight = nn.Parameter(torch.empty(cout, kernel, kernel, cin))
        init_conv_weight_(self.weight, init)
        if bias:
            self.bias = nn.Parameter(torch.zeros(cout))
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        return ops.conv_transpose2d(x, self.weight, self.bias, self.stride,
                                    self.act)


class InstanceNormNHWC(nn.Module):
    """tfa-style InstanceNorm: eps=1e-3, gamma~N(0,0.02), beta=0, with
    optionally fused activation and residual add."""

    def __init__(self, channels: int, eps: float = EPS_DEFAULT,
                 act: Optional[str] = None, slope: float = 0.2):
        super().__init__()
        self.eps, self.act, self.slope = eps, act, slope
        self.gamma = nn.Parameter(torch.empty(channels).normal_(0.0, 0.02))
        self.beta = nn.Parameter(torch.zeros(channels))

    def forward(self, x, residual: Optional[torch.Tensor] = None):
        return ops.instance_norm(x, self.gamma, self.beta, self.eps,
                                 self.act, self.slope, residual)


class ResBlock(nn.Module):
    """Reference residual block (/root/reference/cyclegan/model.py:36-74):
    reflect-pad(1) -> conv3x3 valid no-bias -> IN -> ReLU ->
    reflect-pad(1) -> conv3x3 valid no-bias -> IN -> (+input).
    Pads are folded into the convs; the add is fused into the second IN."""

    def __init__(self, channels: int):
        super().__init__()
        self.conv1 = ConvNHWC(channels, channels, 3, 1,
                              padding=(1, 1, 1, 1), pad_mode="reflect")
        self.norm1 = InstanceNormNHWC(channels, act="relu")
        self.conv2 = ConvNHWC(channels, channels, 3, 1,
                              padding=(1, 1, 1, 1), pad_mode="reflect")
        self.norm2 = InstanceNormNHWC(channels)

    def forward(self, x):
        h = self.norm1(self.conv1(x))
        return self.norm2(self.conv2(h), residual=x)
