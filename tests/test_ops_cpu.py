"""Reference-path op semantics on CPU (the same implementations serve as
the GPU kernels' oracles)."""

import pytest
import torch
import torch.nn.functional as F

from cyclegan_amd import ops
from cyclegan_amd.ops.norm import _in_ref
from cyclegan_amd.ops.conv import act_bwd_from_output, ACT_RELU, ACT_LRELU, ACT_TANH


def test_conv2d_valid_matches_torch():
    torch.manual_seed(0)
    x = torch.randn(2, 10, 10, 3)
    w = torch.randn(8, 3, 3, 3)  # OHWI
    y = ops.conv2d(x, w, padding="valid")
    ref = F.conv2d(x.permute(0, 3, 1, 2), w.permute(0, 3, 1, 2))
    assert torch.allclose(y, ref.permute(0, 2, 3, 1), atol=1e-5)


def test_conv2d_same_stride2_output_shape():
    x = torch.randn(1, 256, 256, 4)
    w = torch.randn(8, 3, 3, 4)
    assert ops.conv2d(x, w, stride=2, padding="same").shape == (1, 128, 128, 8)
    w4 = torch.randn(8, 4, 4, 4)
    assert ops.conv2d(x, w4, stride=2, padding="same").shape == (1, 128, 128, 8)
    assert ops.conv2d(x, w4, stride=1, padding="same").shape == (1, 256, 256, 8)


def test_conv2d_reflect_pad_equals_explicit_pad():
    torch.manual_seed(1)
    x = torch.randn(2, 8, 8, 4)
    w = torch.randn(4, 3, 3, 4)
    y1 = ops.conv2d(x, w, padding=(1, 1, 1, 1), pad_mode="reflect")
    xp = ops.reflection_pad2d(x, (1, 1))
    y2 = ops.conv2d(xp, w, padding="valid")
    assert torch.allclose(y1, y2, atol=1e-5)


def test_conv_transpose_is_adjoint_of_conv():
    """<conv(x), y> == <x, convT(y)> with the TF-'SAME' pad pairing."""
    torch.manual_seed(2)
    s = 2
    x = torch.randn(1, 16, 16, 6)
    w = torch.randn(4, 3, 3, 6)  # OHWI conv: 6 -> 4 channels, 16 -> 8 spatial
    y = torch.randn(1, 8, 8, 4)
    cx = ops.conv2d(x, w, stride=s, padding="same")
    # adjoint maps 8 -> 16 with the transposed channel order
    wt = w.permute(3, 1, 2, 0).contiguous()  # OHWI with O=6, I=4
    aty = ops.conv_transpose2d(y, wt, stride=s)
    assert aty.shape == x.shape
    lhs = (cx * y).sum()
    rhs = (x * aty).sum()
    assert torch.allclose(lhs, rhs, rtol=1e-4)


def test_conv_transpose_shape_tf_same():
    x = torch.randn(2, 64, 64, 8)
    w = torch.randn(4, 3, 3, 8)
    assert ops.conv_transpose2d(x, w, stride=2).shape == (2, 128, 128, 4)


def test_reflection_pad_matches_torch():
    x = torch.randn(2, 6, 6, 3)
    y = ops.reflection_pad2d(x, (3, 3))
    ref = F.pad(x.permute(0, 3, 1, 2), (3, 3, 3, 3), mode="reflect")
    assert torch.allclose(y, ref.permute(0, 2, 3, 1))


def test_instance_norm_semantics():
    torch.manual_seed(3)
    x = torch.randn(2, 7, 9, 5)
    gamma = torch.randn(5) * 0.02
    beta = torch.randn(5) * 0.1
    y = ops.instance_norm(x, gamma, beta, eps=1e-3)
    # manual: per (b, c) stats over H, W
    xm = x.mean(dim=(1, 2), keepdim=True)
    xv = x.var(dim=(1, 2), unbiased=False, keepdim=True)
    ref = (x - xm) / torch.sqrt(xv + 1e-3) * gamma + beta
    assert torch.allclose(y, ref, atol=1e-5)


def test_instance_norm_eps_is_1e3_by_default():
    """tfa default eps=1e-3; a silent 1e-5 default would diverge on
    low-variance channels."""
    x = torch.zeros(1, 4, 4, 2)
    x[..., 0] = torch.linspace(0, 1e-3, 16).view(4, 4)
    g = torch.ones(2)
    b = torch.zeros(2)
    y3 = ops.instance_norm(x, g, b)          # default eps
    y5 = ops.instance_norm(x, g, b, eps=1e-5)
    assert not torch.allclose(y3, y5, atol=1e-3)


def test_instance_norm_fused_relu_and_residual():
    torch.manual_seed(4)
    x = torch.randn(2, 5, 5, 3)
    res = torch.randn(2, 5, 5, 3)
    g, b = torch.randn(3), torch.randn(3)
    y = ops.instance_norm(x, g, b, act="relu")
    base = ops.instance_norm(x, g, b)
    assert torch.allclose(y, torch.relu(base))
    y2 = ops.instance_norm(x, g, b, residual=res)
    assert torch.allclose(y2, base + res, atol=1e-6)


def test_losses_per_sample():
    torch.manual_seed(5)
    a = torch.randn(3, 4, 4, 2)
    b = torch.randn(3, 4, 4, 2)
    mae = ops.MAE(a, b)
    assert mae.shape == (3,)
    assert torch.allclose(mae, (a - b).abs().mean(dim=(1, 2, 3)))
    mse = ops.MSE(a, b)
    assert torch.allclose(mse, ((a - b) ** 2).mean(dim=(1, 2, 3)))
    msec = ops.MSE_const(b, 1.0)
    assert torch.allclose(msec, ((b - 1) ** 2).mean(dim=(1, 2, 3)))


def test_act_bwd_from_output():
    torch.manual_seed(6)
    x = torch.randn(100, requires_grad=True)
    dy = torch.randn(100)
    for act, fn in ((ACT_RELU, torch.relu),
                    (ACT_LRELU, lambda t: F.leaky_relu(t, 0.2)),
                    (ACT_TANH, torch.tanh)):
        x.grad = None
        y = fn(x)
        y.backward(dy)
        got = act_bwd_from_output(dy, y.detach(), act, 0.2)
        assert torch.allclose(got, x.grad, atol=1e-6), act


def test_conv_backward_through_ref():
    torch.manual_seed(7)
    x = torch.randn(1, 8, 8, 3, requires_grad=True)
    w = torch.randn(4, 3, 3, 3, requires_grad=True)
    y = ops.conv2d(x, w, stride=2, padding="same", act="relu")
    y.sum().backward()
    assert x.grad is not None and w.grad is not None
    assert x.grad.shape == x.shape and w.grad.shape == w.shape


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_conv_dtype_paths(dtype):
    x = torch.randn(1, 8, 8, 3).to(dtype)
    w = torch.randn(4, 3, 3, 3)  # fp32 master
    y = ops.conv2d(x, w, padding="same")
    assert y.dtype == dtype


def test_gradcheck_conv_small():
    """fp64 gradcheck of the conv op (reference path) on tiny shapes."""
    torch.manual_seed(8)
    x = torch.randn(1, 5, 5, 2, dtype=torch.float64, requires_grad=True)
    w = torch.randn(3, 3, 3, 2, dtype=torch.float64, requires_grad=True)
    assert torch.autograd.gradcheck(
        lambda a, b: ops.conv2d(a, b, None, 1, (1, 1, 1, 1), "reflect"),
        (x, w), eps=1e-6, atol=1e-4)


def test_gradcheck_conv_transpose_small():
    torch.manual_seed(9)
    x = torch.randn(1, 4, 4, 3, dtype=torch.float64, requires_grad=True)
    w = torch.randn(2, 3, 3, 3, dtype=torch.float64, requires_grad=True)
    assert torch.autograd.gradcheck(
        lambda a, b: ops.conv_transpose2d(a, b, stride=2), (x, w),
        eps=1e-6, atol=1e-4)


def test_gradcheck_instance_norm_small():
    torch.manual_seed(10)
    x = torch.randn(2, 4, 4, 3, dtype=torch.float64, requires_grad=True)
    g = torch.randn(3, dtype=torch.float64, requires_grad=True)
    b = torch.randn(3, dtype=torch.float64, requires_grad=True)
    assert torch.autograd.gradcheck(
        lambda a, gg, bb: ops.instance_norm(a, gg, bb, eps=1e-3), (x, g, b),
        eps=1e-6, atol=1e-4)


def test_bce_matches_tf_semantics():
    """BCE (reference dead-code parity): per-sample mean, prob clamp."""
    import torch
    from cyclegan_amd.ops import BCE
    g = torch.Generator().manual_seed(0)
    p = torch.rand(3, 5, 5, 2, generator=g)
    t = (torch.rand(3, 5, 5, 2, generator=g) > 0.5).float()
    out = BCE(t, p)
    assert out.shape == (3,)
    ref = torch.nn.functional.binary_cross_entropy(
        p, t, reduction="none").mean(dim=(1, 2, 3))
    assert torch.allclose(out, ref, atol=1e-5)
    # clamp keeps exact 0/1 predictions finite (TF semantics)
    assert torch.isfinite(BCE(torch.ones(1, 4), torch.ones(1, 4))).all()
    assert torch.isfinite(BCE(torch.ones(1, 4), torch.zeros(1, 4))).all()
