"""Independent validation of the CPU reference path (the kernel oracle)
against torch.nn.functional convolutions.

The GPU kernels are tested against ops.conv2d's CPU path; this file
closes the loop by checking that CPU path (NHWC/OHWI layouts, TF 'SAME'
pad split, reflection folding, conv-transpose conventions) against
torch's own NCHW convolutions on hypothesis-driven shapes.
"""

import pytest
import torch
import torch.nn.functional as F
from hypothesis import given, settings, strategies as st

from cyclegan_amd import ops
from cyclegan_amd.ops.conv import same_pads


def _nchw(x):  # NHWC -> NCHW
    return x.permute(0, 3, 1, 2).contiguous()


def _torch_conv(x, w, bias, stride, pads, pad_mode):
    # pads = (pt, pb, pl, pr); F.pad order = (left, right, top, bottom)
    xc = _nchw(x)
    if any(pads):
        xc = F.pad(xc, (pads[2], pads[3], pads[0], pads[1]),
                   mode="reflect" if pad_mode == "reflect" else "constant")
    wc = w.permute(0, 3, 1, 2)  # OHWI -> OIHW
    y = F.conv2d(xc, wc, bias, stride=stride)
    return y.permute(0, 2, 3, 1)


@settings(max_examples=30, deadline=None)
@given(st.integers(1, 2), st.integers(1, 3), st.integers(1, 4),
       st.integers(1, 5), st.integers(1, 7), st.integers(1, 2),
       st.booleans(), st.sampled_from(["valid", "same", "explicit"]))
def test_conv2d_cpu_matches_torch(b, cin, cout, kh, kw, stride, reflect,
                                  padding):
    g = torch.Generator().manual_seed(b * 1000 + cin * 100 + kh * 10 + kw)
    h = kh + 3 + stride
    w_ = kw + 2 + stride
    x = torch.randn(b, h, w_, cin, generator=g)
    wt = torch.randn(cout, kh, kw, cin, generator=g) * 0.3
    bias = torch.randn(cout, generator=g)
    if padding == "explicit":
        pads = (kh - 1, kh // 2, kw - 1, kw // 2)
    elif padding == "same":
        pads = same_pads(h, w_, kh, kw, stride)
    else:
        pads = (0, 0, 0, 0)
    if reflect and (pads[0] >= h or pads[1] >= h or pads[2] >= w_
                    or pads[3] >= w_):
        return  # reflect pad must be < dim
    mode = "reflect" if reflect else "zeros"
    y = ops.conv2d(x, wt, bias, stride=stride, padding=pads, pad_mode=mode)
    ref = _torch_conv(x, wt, bias, stride, pads, mode)
    assert y.shape == ref.shape
    assert torch.allclose(y, ref, atol=1e-4, rtol=1e-4), \
        (y - ref).abs().max()


@settings(max_examples=20, deadline=None)
@given(st.integers(1, 2), st.integers(1, 3), st.integers(1, 3),
       st.integers(2, 4))
def test_conv_transpose2d_cpu_matches_torch(b, cin, cout, k):
    """Stride-2 'same' transpose conv (the generator upsample shape
    family) vs torch: output 2x input, TF output_padding semantics."""
    g = torch.Generator().manual_seed(b * 97 + cin * 13 + cout * 7 + k)
    h = 6
    x = torch.randn(b, h, h, cin, generator=g)
    wt = torch.randn(cout, k, k, cin, generator=g) * 0.3
    y = ops.conv_transpose2d(x, wt, None, stride=2)
    # torch equivalent: full (uncropped) transpose conv, then the TF
    # 'SAME' crop — pad_total = k - stride split before = total//2,
    # after = the rest (asymmetric for odd k)
    wc = wt.permute(3, 0, 1, 2)  # OHWI(cout,k,k,cin) -> (cin,cout,k,k)
    full = F.conv_transpose2d(_nchw(x), wc, None, stride=2)
    pt = max(0, (k - 2) // 2)
    ref = full[:, :, pt:pt + 2 * h, pt:pt + 2 * h].permute(0, 2, 3, 1)
    assert y.shape == ref.shape
    assert torch.allclose(y, ref, atol=1e-4, rtol=1e-4), \
        (y - ref).abs().max()


@settings(max_examples=20, deadline=None)
@given(st.integers(1, 3), st.integers(1, 6), st.integers(2, 9),
       st.booleans())
def test_instance_norm_cpu_matches_torch(b, c, hw, with_residual):
    """IN oracle vs torch.nn.functional.instance_norm at tfa's eps=1e-3
    (per-(sample,channel) stats, affine)."""
    g = torch.Generator().manual_seed(b * 31 + c * 7 + hw)
    x = torch.randn(b, hw, hw, c, generator=g)
    gamma = torch.randn(c, generator=g) * 0.02
    beta = torch.randn(c, generator=g) * 0.1
    res = torch.randn(b, hw, hw, c, generator=g) if with_residual else None
    y = ops.instance_norm(x, gamma, beta, eps=1e-3, act="relu",
                          residual=res)
    ref = F.instance_norm(_nchw(x), weight=gamma, bias=beta,
                          eps=1e-3).permute(0, 2, 3, 1)
    if res is not None:
        ref = ref + res
    ref = torch.relu(ref)
    assert torch.allclose(y, ref, atol=1e-5, rtol=1e-4), \
        (y - ref).abs().max()
