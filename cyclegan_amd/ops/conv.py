"""NHWC convolution ops (implicit-GEMM MFMA kernels on GPU).

Layouts (MI355X-first):
- activations: ``[B, H, W, C]`` contiguous (channels innermost → coalesced
  bf16x8 loads, C is the GEMM K/N dimension for the matrix cores).
- weights: OHWI ``[Cout, kh, kw, Cin]`` contiguous — the transposed
  implicit-GEMM B operand ``B^T[N = Cout][K = kh*kw*Cin]`` row-major, so
  weight staging is a coalesced row copy AND the per-lane MFMA B-fragment
  (8 contiguous k at one n) is a single ds_read_b128. No transpose
  anywhere on the forward path.
- conv_transpose weights: OHWI ``[Cout, kh, kw, Cin]`` with Cin = source
  channels (our own convention; checkpoints are native to this framework).

Padding follows the TF 'SAME'/'VALID' conventions of the reference
(pad_before = total//2 — /root/reference/main.py uses Keras Conv2D 'same');
pads are explicit ``(pt, pb, pl, pr)`` everywhere. ``pad_mode='reflect'``
implements ReflectionPadding2D folded into the conv
(/root/reference/cyclegan/model.py:14-33).

Activation epilogues (none/relu/lrelu/tanh) are fused into the conv kernel
— on 8 TB/s HBM the win is not re-reading the output tensor.

Autograd: the Function receives the fp32 master weight; bf16 compute copies
come from ops.shadow. Weight grads are produced in fp32 (split-K
accumulation) and flow straight into the fp32 master's grad bucket.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn.functional as F

from . import backend
from .shadow import (compute_weight, compute_weight_t, compute_weight_p,
                     compute_weight_tp, compute_bias_p, compute_weight_packp,
                     compute_bias_packb, fp8_weight_shadow)

ACT_NONE, ACT_RELU, ACT_LRELU, ACT_TANH = 0, 1, 2, 3

_ACT = {None: ACT_NONE, "none": ACT_NONE, "relu": ACT_RELU, "lrelu": ACT_LRELU, "tanh": ACT_TANH}


def same_pads(in_h: int, in_w: int, kh: int, kw: int, stride: int) -> Tuple[int, int, int, int]:
    """TF 'SAME' pads (pad_before = total//2, extra goes after)."""
    out_h = -(-in_h // stride)
    out_w = -(-in_w // stride)
    ph = max(0, (out_h - 1) * stride + kh - in_h)
    pw = max(0, (out_w - 1) * stride + kw - in_w)
    return ph // 2, ph - ph // 2, pw // 2, pw - pw // 2


def _apply_act(y: torch.Tensor, act: int, slope: float) -> torch.Tensor:
    if act == ACT_RELU:
        return torch.relu(y)
    if act == ACT_LRELU:
        return F.leaky_relu(y, slope)
    if act == ACT_TANH:
        return torch.tanh(y)
    return y


def act_bwd_from_output(dy: torch.Tensor, y: torch.Tensor, act: int, slope: float) -> torch.Tensor:
    """Backward of the fused activation, computed from the *output* y."""
    if act == ACT_RELU:
        return dy * (y > 0).to(dy.dtype)
    if act == ACT_LRELU:
        # slope > 0 keeps sign(pre-act) == sign(post-act)
        return torch.where(y > 0, dy, dy * slope)
    if act == ACT_TANH:
        return dy * (1 - y.float() * y.float()).to(dy.dtype)
    return dy


def _conv_ref(x, w, bias, stride, pads, pad_mode):
    """Differentiable torch reference (CPU path / GPU oracle); w is OHWI."""
    pt, pb, pl, pr = pads
    xn = x.permute(0, 3, 1, 2)
    if pt or pb or pl or pr:
        mode = "reflect" if pad_mode == "reflect" else "constant"
        xn = F.pad(xn, (pl, pr, pt, pb), mode=mode)
    wn = w.permute(0, 3, 1, 2)  # OHWI -> OIHW
    y = F.conv2d(xn.contiguous(), wn.contiguous(), bias, stride=stride)
    return y.permute(0, 2, 3, 1)


def _convt_ref(x, w, bias, stride, pt, pl, out_h, out_w):
    """Transpose conv = adjoint of the TF-'SAME' strided conv (gather form:
    out[i] += in[o]*w[k] where i = s*o + k - pt); w is OHWI."""
    xn = x.permute(0, 3, 1, 2)
    wn = w.permute(3, 0, 1, 2)  # OHWI -> (Cin, Cout, kh, kw)
    y = F.conv_transpose2d(xn.contiguous(), wn.contiguous(), bias, stride=stride)
    y = y[:, :, pt : pt + out_h, pl : pl + out_w]
    return y.permute(0, 2, 3, 1)


def _pad_channels(t: torch.Tensor, c_to: int = 8) -> torch.Tensor:
    """Zero-pad the channel (last) dim of an NHWC tensor up to c_to."""
    c = t.shape[-1]
    if c >= c_to:
        return t
    return F.pad(t, (0, c_to - c))


class _ConvFn(torch.autograd.Function):
    """HIP implicit-GEMM conv with fused pad + bias + activation.

    Channel dims < 8 (the RGB stem / 1-3 channel heads) are zero-padded to
    8 so every shape runs the fast glds kernels; padded input channels see
    zero activations (grads exactly zero) and padded output channels are
    sliced off."""

    @staticmethod
    def forward(ctx, x, w, bias, stride, pads, reflect, act, slope):
        ext = backend.ext()
        cout = w.shape[0]
        xp = _pad_channels(x).contiguous()
        wc = compute_weight_p(w, x)
        bc = compute_bias_p(bias, x) if bias is not None else None
        y = ext.conv2d_fwd(xp, wc, bc, stride, *pads, reflect, act, slope)
        if cout < 8:
            y = y[..., :cout].contiguous()
        ctx.save_for_backward(xp, w, y)
        ctx.conf = (stride, pads, reflect, act, slope, bias is not None,
                    x.shape[3])
        return y

    @staticmethod
    def backward(ctx, dy):
        xp, w, y = ctx.saved_tensors
        stride, pads, reflect, act, slope, has_bias, cin = ctx.conf
        cout = w.shape[0]
        pt, pb, pl, pr = pads
        ext = backend.ext()
        dy = dy.contiguous()
        if act != ACT_NONE:
            dy = ext.act_bwd(dy, y, act, slope)
        dyp = _pad_channels(dy).contiguous()
        db = None
        if has_bias and ctx.needs_input_grad[2]:
            db = ext.channel_sum(dyp)[: dy.shape[-1]]
        dx = dw = None
        if ctx.needs_input_grad[0]:
            wt = compute_weight_tp(w, xp)
            dx = ext.conv2d_dgrad(dyp, wt, xp.shape[1], xp.shape[2], stride,
                                  pt, pb, pl, pr, reflect)
            if cin < 8:
                dx = dx[..., :cin].contiguous()
        if ctx.needs_input_grad[1]:
            dw = ext.conv2d_wgrad(xp, dyp, w.shape[1], w.shape[2], stride,
                                  pt, pl, reflect)
            if cout < 8 or cin < 8:
                dw = dw[:cout, :, :, :cin].contiguous()
            if dw.dtype != w.dtype:
                dw = dw.to(w.dtype)
        return dx, dw, db, None, None, None, None, None


def _head_packable(x, w, stride, pads, reflect) -> bool:
    """Generator-head geometry (7x7 s1 reflect-3, Cout <= 8, Cin % 8 == 0):
    runs as the 8-pixel-packed conv — 8 adjacent output pixels fold into
    the MFMA N dimension (Cout' = 64) so the matrix cores run dense
    instead of 8/16-wide, and the 49x input re-read of the skinny path
    collapses into a 7x2-block window over the channel-folded image."""
    import os
    if os.environ.get("CYG_NO_PACKHEAD") == "1":
        return False
    return (reflect and stride == 1 and pads == (3, 3, 3, 3)
            and w.shape[0] <= 8 and w.shape[1] == 7 and w.shape[2] == 7
            and w.shape[3] % 8 == 0 and w.shape[3] >= 8
            and x.shape[2] % 8 == 0 and x.shape[1] >= 7 and x.shape[2] >= 16)


def _fold_packed_dw(dwp: torch.Tensor, wshape) -> torch.Tensor:
    """[64, KH, 2, 8*I] packed wgrad -> OHWI [O, KH, KW, I]."""
    O, KH, KW, I = wshape
    d6 = dwp.view(8, 8, KH, 2, 8, I)                    # [d,co,ty,blk,pos,ci]
    d7 = d6.permute(0, 3, 4, 1, 2, 5).reshape(8, 16, 8, KH, I)  # [d,a,co,ty,ci]
    dev = dwp.device
    idx = (torch.arange(KW, device=dev).view(1, -1)
           + torch.arange(8, device=dev).view(-1, 1))   # a = tx + d
    idx = idx.view(8, KW, 1, 1, 1).expand(8, KW, 8, KH, I)
    dw = d7.gather(1, idx).sum(0)                       # [tx, co, ty, ci]
    return dw.permute(1, 2, 0, 3)[:O].contiguous()      # [co,ty,tx,ci]


class _ConvHeadPackedFn(torch.autograd.Function):
    """8-pixel-packed head conv (see _head_packable). Forward and wgrad
    run the packed dense-N GEMM; dgrad keeps the unpacked formulation
    (its packed form would do 2.4x the FLOPs on zero weights)."""

    @staticmethod
    def forward(ctx, x, w, bias, act, slope):
        ext = backend.ext()
        B, H, W, Cin = x.shape
        cout = w.shape[0]
        wpk = compute_weight_packp(w, x)
        bpk = compute_bias_packb(bias, x) if bias is not None else None
        xp = ext.reflect_pad_fwd(x.contiguous(), 3, 3, 3, 5)
        xf = xp.view(B, H + 6, (W + 8) // 8, 8 * Cin)
        yp = ext.conv2d_fwd(xf, wpk, bpk, 1, 0, 0, 0, 0, False, act, slope)
        y = yp.view(B, H, W, 8)
        if cout < 8:
            y = y[..., :cout].contiguous()
        ctx.save_for_backward(xp, w, y)
        ctx.conf = (act, slope, bias is not None, cout, Cin, H, W)
        return y

    @staticmethod
    def backward(ctx, dy):
        xp, w, y = ctx.saved_tensors
        act, slope, has_bias, cout, Cin, H, W = ctx.conf
        ext = backend.ext()
        dy = dy.contiguous()
        if act != ACT_NONE:
            dy = ext.act_bwd(dy, y, act, slope)
        dyp = _pad_channels(dy).contiguous()
        db = None
        if has_bias and ctx.needs_input_grad[2]:
            db = ext.channel_sum(dyp)[: dy.shape[-1]]
        dx = dw = None
        B = dyp.shape[0]
        if ctx.needs_input_grad[0]:
            wt = compute_weight_tp(w, xp)
            dx = ext.conv2d_dgrad(dyp, wt, H, W, 1, 3, 3, 3, 3, True)
        if ctx.needs_input_grad[1]:
            xf = xp.view(B, H + 6, (W + 8) // 8, 8 * Cin)
            dy_pk = dyp.view(B, H, W // 8, 64)
            dwp = ext.conv2d_wgrad(xf, dy_pk, 7, 2, 1, 0, 0, False)
            dw = _fold_packed_dw(dwp, tuple(w.shape))
            if dw.dtype != w.dtype:
                dw = dw.to(w.dtype)
        return dx, dw, db, None, None


_FP8_MODE = False


def set_fp8_mode(on: bool):
    """Enable the CDNA4 fp8 (e4m3) forward-conv path (BASELINE config 5):
    per-tensor-scaled fp8 weights+activations on the fp8 MFMA, fp32
    accumulate, bf16 outputs; backward runs on the saved bf16 activations."""
    global _FP8_MODE
    _FP8_MODE = bool(on)


def fp8_mode() -> bool:
    return _FP8_MODE


class _ConvFp8Fn(torch.autograd.Function):
    """fp8 e4m3 forward conv with delayed per-tensor activation scaling
    (ops.fp8_state): quantization uses last step's amax — no separate
    amax reduction in the hot path — and the dequant factor is computed
    inside the conv epilogue from the amax/sw scalars."""

    @staticmethod
    def forward(ctx, x, w, bias, stride, pads, reflect, act, slope):
        from . import fp8_state
        ext = backend.ext()
        cout = w.shape[0]
        xp = _pad_channels(x).contiguous()
        wq, sw = fp8_weight_shadow(w)
        prev, cur, fresh = fp8_state.slots_for(w)
        if fresh:  # bootstrap (eager, once per layer, pre-capture)
            prev.copy_(xp.detach().abs().amax().float().clamp(min=1e-12))
        xq = ext.quant_fp8_d(xp, prev, cur)
        bc = compute_bias_p(bias, x) if bias is not None else None
        y = ext.conv2d_fp8_fwd(xq, wq, prev, sw, bc, stride, *pads,
                               reflect, act, slope)
        if cout < 8:
            y = y[..., :cout].contiguous()
        ctx.save_for_backward(xp, w, y)
        ctx.conf = (stride, pads, reflect, act, slope, bias is not None,
                    x.shape[3])
        return y

    backward = _ConvFn.backward


class _ConvTFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, bias, stride, pt, pl, out_h, out_w, act, slope):
        ext = backend.ext()
        wc = compute_weight(w, x)
        bc = compute_weight(bias, x) if bias is not None else None
        y = ext.convt2d_fwd(x, wc, bc, stride, pt, pl, out_h, out_w, act, slope)
        ctx.save_for_backward(x, w, y)
        ctx.conf = (stride, pt, pl, act, slope, bias is not None)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, y = ctx.saved_tensors
        stride, pt, pl, act, slope, has_bias = ctx.conf
        ext = backend.ext()
        dy = dy.contiguous()
        if act != ACT_NONE:
            dy = ext.act_bwd(dy, y, act, slope)
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            # adjoint of the gather is the forward strided conv of dy
            wt = compute_weight_t(w, x)
            dx = ext.convt2d_dgrad(dy, wt, x.shape[1], x.shape[2], stride, pt, pl)
        if ctx.needs_input_grad[1]:
            # convT wgrad == conv wgrad with roles swapped (x := dy, dy := x),
            # then a channel transpose back to OHWI
            dwc = ext.conv2d_wgrad(dy, x, w.shape[1], w.shape[2], stride,
                                   pt, pl, False)
            dw = dwc.permute(3, 1, 2, 0).contiguous()
            if dw.dtype != w.dtype:
                dw = dw.to(w.dtype)
        if has_bias and ctx.needs_input_grad[2]:
            db = dy.float().sum(dim=(0, 1, 2))
        return dx, dw, db, None, None, None, None, None, None, None


def conv2d(
    x: torch.Tensor,
    w: torch.Tensor,
    bias: Optional[torch.Tensor] = None,
    stride: int = 1,
    padding: str | Tuple[int, int, int, int] = "valid",
    pad_mode: str = "zeros",
    act: Optional[str] = None,
    slope: float = 0.2,
) -> torch.Tensor:
    """NHWC conv2d. ``padding`` is 'valid', 'same' (TF semantics) or explicit
    (pt, pb, pl, pr). ``pad_mode`` 'zeros'|'reflect'. ``act`` fused epilogue."""
    kh, kw = w.shape[1], w.shape[2]
    if padding == "valid":
        pads = (0, 0, 0, 0)
    elif padding == "same":
        pads = same_pads(x.shape[1], x.shape[2], kh, kw, stride)
    else:
        pads = tuple(padding)
    a = _ACT[act]
    if backend.use_hip(x, w):
        # packed head first: skinny-Cout convs gain more from dense-N
        # bf16 packing than from fp8's rate (and skip fp8's quant cost)
        if _head_packable(x, w, stride, pads, pad_mode == "reflect"):
            return _ConvHeadPackedFn.apply(x, w, bias, a, slope)
        # fp8 only where the 16x16x128 fp8 kernel + quant beats the bf16
        # glds kernel (tools/fp8_micro.py on MI355X): the wide-channel
        # stride-1 3x3 family at batch >= 12 (K3: 999 vs 715 TF/s) and
        # the 4x4 s1 wide discriminator conv. Elsewhere the per-tensor
        # quantization overhead cancels the MFMA-rate gain.
        if (_FP8_MODE and x.dtype == torch.bfloat16
                and w.shape[3] % 16 == 0 and w.shape[3] >= 256
                and stride == 1
                and ((w.shape[1] == 3 and x.shape[0] >= 12)
                     or w.shape[1] == 4)):
            return _ConvFp8Fn.apply(x, w, bias, stride, pads,
                                    pad_mode == "reflect", a, slope)
        return _ConvFn.apply(x, w, bias, stride, pads, pad_mode == "reflect", a, slope)
    wc = w if w.dtype == x.dtype else w.to(x.dtype)
    bc = bias if (bias is None or bias.dtype == x.dtype) else bias.to(x.dtype)
    y = _conv_ref(x, wc, bc, stride, pads, pad_mode)
    return _apply_act(y, a, slope)


def conv_transpose2d(
    x: torch.Tensor,
    w: torch.Tensor,
    bias: Optional[torch.Tensor] = None,
    stride: int = 2,
    act: Optional[str] = None,
    slope: float = 0.2,
) -> torch.Tensor:
    """NHWC transpose conv with TF-'SAME' semantics: out = in*stride.

    Defined as the adjoint of conv2d(..., stride, 'same') mapping
    (in*stride) -> in, i.e. out[i] += in[o] * w[k] with i = s*o + k - pt.
    """
    kh, kw = w.shape[1], w.shape[2]
    out_h, out_w = x.shape[1] * stride, x.shape[2] * stride
    pt, _, pl, _ = same_pads(out_h, out_w, kh, kw, stride)
    a = _ACT[act]
    if backend.use_hip(x, w):
        return _ConvTFn.apply(x, w, bias, stride, pt, pl, out_h, out_w, a, slope)
    wc = w if w.dtype == x.dtype else w.to(x.dtype)
    bc = bias if (bias is None or bias.dtype == x.dtype) else bias.to(x.dtype)
    y = _convt_ref(x, wc, bc, stride, pt, pl, out_h, out_w)
    return _apply_act(y, a, slope)
