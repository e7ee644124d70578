"""Fused InstanceNorm (NHWC) — per-(sample,channel) mean/var over H*W.

Replicates tfa.layers.InstanceNormalization (groups=-1) semantics used by the
reference (/root/reference/cyclegan/model.py:58-72): eps=1e-3 (tfa default,
NOT PyTorch's 1e-5), affine gamma/beta with gamma~N(0,0.02), beta=0.

MI355X fusion: the normalize pass optionally fuses the following ReLU /
LeakyReLU and/or a residual add (resblock tail, model.py:73) so the
activation tensor makes one HBM round trip instead of three.

gamma/beta stay fp32 (256 floats — L2-resident); x may be bf16 or fp32;
statistics always accumulate in fp32.
"""

from __future__ import annotations

from typing import Optional

import torch

from . import backend
from .conv import ACT_NONE, ACT_RELU, ACT_LRELU, _ACT

EPS_DEFAULT = 1e-3  # tfa InstanceNormalization default


def _in_ref(x, gamma, beta, eps, act, slope, residual):
    # compute in fp32 for bf16 inputs; keep native precision otherwise
    # (fp64 gradcheck needs the graph to stay fp64)
    cdt = torch.float32 if x.dtype == torch.bfloat16 else x.dtype
    xf = x.to(cdt)
    mean = xf.mean(dim=(1, 2), keepdim=True)
    var = xf.var(dim=(1, 2), unbiased=False, keepdim=True)
    y = (xf - mean) * torch.rsqrt(var + eps)
    y = y * gamma.to(cdt) + beta.to(cdt)
    y = y.to(x.dtype)
    if residual is not None:
        y = y + residual
    if act == ACT_RELU:
        y = torch.relu(y)
    elif act == ACT_LRELU:
        y = torch.nn.functional.leaky_relu(y, slope)
    return y


class _InstNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, eps, act, slope, residual):
        ext = backend.ext()
        y, mean, rstd = ext.instnorm_fwd(x, gamma.float(), beta.float(), eps, act, slope,
                                         residual)
        ctx.save_for_backward(x, gamma, mean, rstd, y)
        ctx.conf = (eps, act, slope, residual is not None)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, gamma, mean, rstd, y = ctx.saved_tensors
        eps, act, slope, has_res = ctx.conf
        ext = backend.ext()
        dy = dy.contiguous()
        if has_res and act != ACT_NONE:
            # rare combo: residual grad needs the transformed dy explicitly
            dy = ext.act_bwd(dy, y, act, slope)
            dres = dy
            dx, dgamma, dbeta = ext.instnorm_bwd(dy, x, gamma.float(), mean,
                                                 rstd, None, ACT_NONE, 0.0)
        else:
            dres = dy if has_res else None
            dx, dgamma, dbeta = ext.instnorm_bwd(
                dy, x, gamma.float(), mean, rstd,
                y if act != ACT_NONE else None, act, slope)
        return (dx, dgamma.to(gamma.dtype), dbeta, None, None, None, dres)


def instance_norm(
    x: torch.Tensor,
    gamma: torch.Tensor,
    beta: torch.Tensor,
    eps: float = EPS_DEFAULT,
    act: Optional[str] = None,
    slope: float = 0.2,
    residual: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    a = _ACT[act]
    if backend.use_hip(x, gamma):
        return _InstNormFn.apply(x, gamma, beta, eps, a, slope, residual)
    return _in_ref(x, gamma, beta, eps, a, slope, residual)
