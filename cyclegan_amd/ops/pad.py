"""Standalone NHWC ReflectionPad2d (reference ReflectionPadding2D,
/root/reference/cyclegan/model.py:14-33).

In the generator the pad is normally folded into the following conv
(ops.conv pad_mode='reflect' — a load-address transform, no materialized
padded tensor). The standalone op exists for API parity and for tests.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from . import backend


class _ReflectPadFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, pt, pb, pl, pr):
        ctx.pads = (pt, pb, pl, pr)
        return backend.ext().reflect_pad_fwd(x, pt, pb, pl, pr)

    @staticmethod
    def backward(ctx, dy):
        pt, pb, pl, pr = ctx.pads
        dx = backend.ext().reflect_pad_bwd(dy.contiguous(), pt, pb, pl, pr)
        return dx, None, None, None, None


def reflection_pad2d(x: torch.Tensor, padding=(1, 1)) -> torch.Tensor:
    """Pad H and W by (ph, pw) in REFLECT mode; x is [B,H,W,C]."""
    ph, pw = padding
    if backend.use_hip(x):
        return _ReflectPadFn.apply(x, ph, ph, pw, pw)
    xn = x.permute(0, 3, 1, 2)
    y = F.pad(xn, (pw, pw, ph, ph), mode="reflect")
    return y.permute(0, 2, 3, 1)
