"""Production-shape kernel numerics: every conv/IN shape the 256² bench
dispatches, verified against the plain-PyTorch fp32 oracle ON GPU
(MIOpen/rocBLAS fp32 — an independent implementation), with PER-ELEMENT
relative bounds (a systematic bias cannot hide behind a max-of-tensor
norm).

A dispatch-recorder test then runs a real train step at the bench config
and asserts every conv shape it dispatches is in the covered list.

All tests @pytest.mark.gpu.
"""

import pytest
import torch

pytestmark = pytest.mark.gpu

from cyclegan_amd import ops
from cyclegan_amd.ops.conv import _conv_ref, _convt_ref, same_pads
from cyclegan_amd.ops.norm import _in_ref

DEV = "cuda:0"
B = 4           # bench per-GPU batch; generator calls run 2B and 3B


def mk(shape, seed=0, scale=1.0):
    g = torch.Generator().manual_seed(seed)
    t = (torch.rand(shape, generator=g) * 2 - 1) * scale
    return t.to(DEV, torch.bfloat16)


def check_elem(got, ref, rtol, name=""):
    """Per-element: |err| <= rtol * (|ref| + rms(ref)). The rms floor keeps
    near-zero elements from demanding absolute bf16 precision while still
    bounding them relative to the tensor's signal level."""
    got = got.float()
    ref = ref.float()
    assert got.shape == ref.shape, f"{name}: {got.shape} vs {ref.shape}"
    rms = ref.pow(2).mean().sqrt()
    bound = rtol * (ref.abs() + rms)
    err = (got - ref).abs()
    bad = (err > bound).sum().item()
    worst = (err / bound.clamp(min=1e-30)).max().item()
    assert bad == 0, f"{name}: {bad}/{ref.numel()} elements out of bound, worst ratio {worst:.2f}"


# Every distinct (Cin,Cout,K,stride,padding,pad_mode,bias,act) x (H,W,B)
# the 256² bench dispatches. Batch: G runs at 2B (cat x,y), 3B (F's fused
# call) and B; D runs at B. We test the largest batch per shape.
FULL_CASES = [
    # name,                 B,  H,   W,  Cin, Cout, K, s, padding,     pad_mode, bias,  act
    ("G_stem7x7_256",      12, 256, 256,   3,  64, 7, 1, (3, 3, 3, 3), "reflect", False, None),
    ("G_down1_3x3s2",      12, 256, 256,  64, 128, 3, 2, "same",       "zeros",  False, None),
    ("G_down2_3x3s2",      12, 128, 128, 128, 256, 3, 2, "same",       "zeros",  False, None),
    ("K3_resblock_B12",    12,  64,  64, 256, 256, 3, 1, (1, 1, 1, 1), "reflect", False, None),
    ("K3_resblock_B8",      8,  64,  64, 256, 256, 3, 1, (1, 1, 1, 1), "reflect", False, None),
    ("G_head7x7_256",      12, 256, 256,  64,   3, 7, 1, (3, 3, 3, 3), "reflect", True,  "tanh"),
    ("D_stem4x4s2_256",     4, 256, 256,   3,  64, 4, 2, "same",       "zeros",  True,  "lrelu"),
    ("D_down1_4x4s2",       4, 128, 128,  64, 128, 4, 2, "same",       "zeros",  False, "lrelu"),
    ("D_down2_4x4s2",       4,  64,  64, 128, 256, 4, 2, "same",       "zeros",  False, "lrelu"),
    ("D_wide4x4s1",         4,  32,  32, 256, 512, 4, 1, "same",       "zeros",  False, "lrelu"),
    ("D_head4x4s1",         4,  32,  32, 512,   1, 4, 1, "same",       "zeros",  True,  None),
]

CONVT_CASES = [
    ("G_up1_convT", 12, 64, 64, 256, 128, 3, 2),
    ("G_up2_convT", 12, 128, 128, 128, 64, 3, 2),
]


def _oracle_conv(x, w, b, s, pads, pm, act):
    y = _conv_ref(x.float(), w.float(), b.float() if b is not None else None,
                  s, pads, pm)
    if act:
        y = {"relu": torch.relu,
             "lrelu": lambda t: torch.nn.functional.leaky_relu(t, 0.2),
             "tanh": torch.tanh}[act](y)
    return y


@pytest.mark.parametrize("case", FULL_CASES, ids=[c[0] for c in FULL_CASES])
def test_conv_full_shape_fwd_bwd(case):
    name, b_, H, W, Cin, Cout, K, s, padding, pm, bias, act = case
    x = mk((b_, H, W, Cin), seed=101)
    w32 = mk((Cout, K, K, Cin), seed=102, scale=0.2).float()
    b32 = mk((Cout,), seed=103, scale=0.1).float() if bias else None

    xg = x.clone().requires_grad_(True)
    wg = w32.clone().requires_grad_(True)
    bg = b32.clone().requires_grad_(True) if bias else None
    y = ops.conv2d(xg, wg, bg, s, padding, pm, act)
    dy = mk(y.shape, seed=104)
    y.backward(dy)

    pads = same_pads(H, W, K, K, s) if padding == "same" else tuple(padding)
    xr = x.float().requires_grad_(True)
    wr = w32.clone().requires_grad_(True)
    br = b32.clone().requires_grad_(True) if bias else None
    yr = _oracle_conv(xr, wr, br, s, pads, pm, act)
    yr.backward(dy.float())

    check_elem(y, yr, 0.035, f"{name}:y")
    check_elem(xg.grad, xr.grad, 0.05, f"{name}:dx")
    check_elem(wg.grad, wr.grad, 0.05, f"{name}:dw")
    if bias:
        check_elem(bg.grad, br.grad, 0.05, f"{name}:db")


@pytest.mark.parametrize("case", CONVT_CASES, ids=[c[0] for c in CONVT_CASES])
def test_convt_full_shape_fwd_bwd(case):
    name, b_, IH, IW, Cin, Cout, K, s = case
    x = mk((b_, IH, IW, Cin), seed=111)
    w32 = mk((Cout, K, K, Cin), seed=112, scale=0.2).float()

    xg = x.clone().requires_grad_(True)
    wg = w32.clone().requires_grad_(True)
    y = ops.conv_transpose2d(xg, wg, stride=s)
    dy = mk(y.shape, seed=113)
    y.backward(dy)

    pt, _, pl, _ = same_pads(IH * s, IW * s, K, K, s)
    xr = x.float().requires_grad_(True)
    wr = w32.clone().requires_grad_(True)
    yr = _convt_ref(xr, wr, None, s, pt, pl, IH * s, IW * s)
    yr.backward(dy.float())

    check_elem(y, yr, 0.035, f"{name}:y")
    check_elem(xg.grad, xr.grad, 0.05, f"{name}:dx")
    check_elem(wg.grad, wr.grad, 0.05, f"{name}:dw")


@pytest.mark.parametrize("conf", [("stem", 12, 256, 256, 64, "relu", False),
                                  ("res", 12, 64, 64, 256, "relu", False),
                                  ("res_add", 12, 64, 64, 256, None, True),
                                  ("up", 12, 256, 256, 64, "relu", False)])
def test_instnorm_full_shape(conf):
    name, b_, H, W, C, act, use_res = conf
    from cyclegan_amd.ops.conv import _ACT
    x = mk((b_, H, W, C), seed=121)
    g32 = (torch.randn(C, generator=torch.Generator().manual_seed(122)) * 0.5).to(DEV)
    b32 = (torch.randn(C, generator=torch.Generator().manual_seed(123)) * 0.1).to(DEV)
    res = mk((b_, H, W, C), seed=124) if use_res else None

    xg = x.clone().requires_grad_(True)
    gg = g32.clone().requires_grad_(True)
    bg = b32.clone().requires_grad_(True)
    rg = res.clone().requires_grad_(True) if use_res else None
    y = ops.instance_norm(xg, gg, bg, act=act, residual=rg)
    dy = mk(y.shape, seed=125)
    y.backward(dy)

    xr = x.float().requires_grad_(True)
    gr = g32.float().requires_grad_(True)
    br = b32.float().requires_grad_(True)
    rr = res.float().requires_grad_(True) if use_res else None
    yr = _in_ref(xr, gr, br, 1e-3, _ACT[act], 0.2, rr)
    yr.backward(dy.float())

    check_elem(y, yr, 0.04, f"in:{name}:y")
    check_elem(xg.grad, xr.grad, 0.06, f"in:{name}:dx")
    check_elem(gg.grad, gr.grad, 0.05, f"in:{name}:dgamma")
    check_elem(bg.grad, br.grad, 0.05, f"in:{name}:dbeta")


def test_bench_dispatch_is_covered(tmp_path):
    """Run a real train step at the bench config (B=4, 256², 9 resblocks)
    with the conv entry points instrumented; every dispatched conv shape
    must be in FULL_CASES/CONVT_CASES (so the oracle suite above covers
    exactly what the bench runs)."""
    import argparse
    from cyclegan_amd.parallel import DistContext
    from cyclegan_amd.trainer import CycleGAN
    from cyclegan_amd.ops import conv as convmod

    recorded = set()
    orig_conv, orig_convt = convmod.conv2d, convmod.conv_transpose2d

    def rec_conv(x, w, bias=None, stride=1, padding="valid",
                 pad_mode="zeros", act=None, slope=0.2):
        recorded.add(("conv", x.shape[0], x.shape[1], x.shape[2], x.shape[3],
                      w.shape[0], w.shape[1], stride))
        return orig_conv(x, w, bias, stride, padding, pad_mode, act, slope)

    def rec_convt(x, w, bias=None, stride=2, act=None, slope=0.2):
        recorded.add(("convt", x.shape[0], x.shape[1], x.shape[2], x.shape[3],
                      w.shape[0], w.shape[1], stride))
        return orig_convt(x, w, bias, stride, act, slope)

    import cyclegan_amd.models.layers as L
    a = argparse.Namespace(output_dir=str(tmp_path), batch_size=B,
                           global_batch_size=B, num_residual_blocks=9,
                           compute_dtype=torch.bfloat16)
    ctx = DistContext(device=torch.device(DEV))
    gan = CycleGAN(a, ctx)
    x = mk((B, 256, 256, 3), seed=131)
    y = mk((B, 256, 256, 3), seed=132)
    try:
        L.ops.conv2d = rec_conv
        L.ops.conv_transpose2d = rec_convt
        gan.train_step(x, y)
        torch.cuda.synchronize()
    finally:
        L.ops.conv2d = orig_conv
        L.ops.conv_transpose2d = orig_convt

    covered = set()
    for (_, b_, H, W, Cin, Cout, K, s, *_rest) in FULL_CASES:
        for bb in (B, 2 * B, 3 * B):
            covered.add(("conv", bb, H, W, Cin, Cout, K, s))
    for (_, b_, IH, IW, Cin, Cout, K, s) in CONVT_CASES:
        for bb in (B, 2 * B, 3 * B):
            covered.add(("convt", bb, IH, IW, Cin, Cout, K, s))

    missing = recorded - covered
    assert not missing, f"bench dispatches uncovered shapes: {sorted(missing)}"
