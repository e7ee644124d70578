"""HIP kernel numerics vs the plain-PyTorch fp32 reference (oracle = the
same _conv_ref/_in_ref implementations the CPU path runs, in fp32 on CPU).

All tests @pytest.mark.gpu — run on an MI355X with
    python -m pytest tests -m gpu -x -q
"""

import pytest
import torch

pytestmark = pytest.mark.gpu

from cyclegan_amd.ops import backend
from cyclegan_amd.ops.conv import _conv_ref, _convt_ref, same_pads
from cyclegan_amd.ops.norm import _in_ref
from cyclegan_amd import ops

DEV = "cuda:0"


def ext():
    return backend.ext()


def mk(shape, seed=0, scale=1.0):
    g = torch.Generator().manual_seed(seed)
    t = (torch.rand(shape, generator=g) * 2 - 1) * scale
    return t.to(DEV, torch.bfloat16)


def check(got, ref, tol=0.05, name=""):
    got = got.float().cpu()
    ref = ref.float().cpu()
    assert got.shape == ref.shape, f"{name}: {got.shape} vs {ref.shape}"
    scale = ref.abs().max().item() + 1e-3
    err = (got - ref).abs().max().item()
    assert err <= tol * scale, f"{name}: max err {err:.4f} vs scale {scale:.4f}"


def test_mfma_layout_probe():
    """Validates the assumed A/B/D fragment layouts of
    mfma_f32_16x16x32_bf16 (asymmetric operands catch transposes)."""
    a = mk((16, 32), seed=1)
    bt = mk((16, 32), seed=2)
    c = ext().mfma_probe(a, bt)
    ref = a.float().cpu() @ bt.float().cpu().T
    check(c, ref, 0.02, "mfma_probe")


CONV_CASES = [
    # (name, B,H,W,Cin,Cout,K,stride,padding,pad_mode,bias,act)
    ("resblock3x3", 2, 16, 16, 256, 256, 3, 1, (1, 1, 1, 1), "reflect", False, None),
    ("resblock_relu", 1, 8, 8, 64, 64, 3, 1, (1, 1, 1, 1), "reflect", False, "relu"),
    ("down3x3s2", 2, 32, 32, 64, 128, 3, 2, "same", "zeros", False, None),
    ("disc4x4s2", 2, 32, 32, 64, 128, 4, 2, "same", "zeros", False, "lrelu"),
    ("disc_stem", 2, 32, 32, 3, 64, 4, 2, "same", "zeros", True, "lrelu"),
    ("stem7x7", 1, 16, 16, 3, 64, 7, 1, (3, 3, 3, 3), "reflect", False, None),
    ("head7x7", 1, 16, 16, 64, 3, 7, 1, (3, 3, 3, 3), "reflect", True, "tanh"),
    ("disc_head", 1, 16, 16, 512, 1, 4, 1, "same", "zeros", True, None),
]


@pytest.mark.parametrize("case", CONV_CASES, ids=[c[0] for c in CONV_CASES])
def test_conv2d_fwd_vs_oracle(case):
    name, B, H, W, Cin, Cout, K, s, padding, pm, bias, act = case
    x = mk((B, H, W, Cin), seed=3)
    w = mk((Cout, K, K, Cin), seed=4, scale=0.2)
    b = mk((Cout,), seed=5) if bias else None
    y = ops.conv2d(x, w, b, s, padding, pm, act)
    pads = same_pads(H, W, K, K, s) if padding == "same" else tuple(padding)
    ref = _conv_ref(x.float().cpu(), w.float().cpu(),
                    b.float().cpu() if bias else None, s, pads, pm)
    if act:
        ref = {"relu": torch.relu, "lrelu": lambda t: torch.nn.functional.leaky_relu(t, 0.2),
               "tanh": torch.tanh}[act](ref)
    check(y, ref, 0.05, name)


@pytest.mark.parametrize("case", CONV_CASES, ids=[c[0] for c in CONV_CASES])
def test_conv2d_grads_vs_oracle(case):
    name, B, H, W, Cin, Cout, K, s, padding, pm, bias, act = case
    x = mk((B, H, W, Cin), seed=6)
    w32 = mk((Cout, K, K, Cin), seed=7, scale=0.2).float()
    b32 = mk((Cout,), seed=8).float() if bias else None

    xg = x.clone().requires_grad_(True)
    wg = w32.clone().requires_grad_(True)
    bg = b32.clone().requires_grad_(True) if bias else None
    y = ops.conv2d(xg, wg, bg, s, padding, pm, act)
    dy = mk(y.shape, seed=9)
    y.backward(dy)

    # oracle in fp32 on CPU (same bf16-rounded inputs)
    xr = x.float().cpu().requires_grad_(True)
    wr = w32.cpu().requires_grad_(True)
    br = b32.cpu().requires_grad_(True) if bias else None
    pads = same_pads(H, W, K, K, s) if padding == "same" else tuple(padding)
    yr = _conv_ref(xr, wr, br, s, pads, pm)
    if act:
        yr = {"relu": torch.relu, "lrelu": lambda t: torch.nn.functional.leaky_relu(t, 0.2),
              "tanh": torch.tanh}[act](yr)
    yr.backward(dy.float().cpu())

    check(xg.grad, xr.grad, 0.08, f"{name}:dx")
    check(wg.grad, wr.grad, 0.08, f"{name}:dw")
    if bias:
        check(bg.grad, br.grad, 0.08, f"{name}:db")


def test_convt2d_fwd_and_grads_vs_oracle():
    B, IH, IW, Cin, Cout, K, s = 2, 8, 8, 256, 128, 3, 2
    x = mk((B, IH, IW, Cin), seed=10)
    w32 = mk((Cout, K, K, Cin), seed=11, scale=0.2).float()

    xg = x.clone().requires_grad_(True)
    wg = w32.clone().requires_grad_(True)
    y = ops.conv_transpose2d(xg, wg, stride=s, act="relu")
    assert y.shape == (B, IH * s, IW * s, Cout)
    dy = mk(y.shape, seed=12)
    y.backward(dy)

    xr = x.float().cpu().requires_grad_(True)
    wr = w32.cpu().requires_grad_(True)
    pt, _, pl, _ = same_pads(IH * s, IW * s, K, K, s)
    yr = _convt_ref(xr, wr, None, s, pt, pl, IH * s, IW * s)
    yr = torch.relu(yr)
    yr.backward(dy.float().cpu())

    check(y, yr, 0.05, "convt:y")
    check(xg.grad, xr.grad, 0.08, "convt:dx")
    check(wg.grad, wr.grad, 0.08, "convt:dw")


@pytest.mark.parametrize("conf", [
    ("plain", None, False), ("relu", "relu", False),
    ("lrelu", "lrelu", False), ("residual", None, True)])
def test_instnorm_vs_oracle(conf):
    name, act, use_res = conf
    B, H, W, C = 2, 16, 16, 128
    x = mk((B, H, W, C), seed=13)
    g32 = (torch.randn(C, generator=torch.Generator().manual_seed(14)) * 0.5).to(DEV)
    b32 = torch.randn(C, generator=torch.Generator().manual_seed(15)).to(DEV) * 0.1
    res = mk((B, H, W, C), seed=16) if use_res else None

    xg = x.clone().requires_grad_(True)
    gg = g32.clone().requires_grad_(True)
    bg = b32.clone().requires_grad_(True)
    rg = res.clone().requires_grad_(True) if use_res else None
    y = ops.instance_norm(xg, gg, bg, act=act, residual=rg)
    dy = mk(y.shape, seed=17)
    y.backward(dy)

    from cyclegan_amd.ops.conv import _ACT
    xr = x.float().cpu().requires_grad_(True)
    gr = g32.float().cpu().requires_grad_(True)
    br = b32.float().cpu().requires_grad_(True)
    rr = res.float().cpu().requires_grad_(True) if use_res else None
    yr = _in_ref(xr, gr, br, 1e-3, _ACT[act], 0.2, rr)
    yr.backward(dy.float().cpu())

    check(y, yr, 0.05, f"in:{name}:y")
    check(xg.grad, xr.grad, 0.1, f"in:{name}:dx")
    check(gg.grad, gr.grad, 0.08, f"in:{name}:dgamma")
    check(bg.grad, br.grad, 0.08, f"in:{name}:dbeta")
    if use_res:
        check(rg.grad, rr.grad, 0.05, f"in:{name}:dres")


def test_reflection_pad_vs_oracle():
    x = mk((2, 8, 8, 64), seed=18)
    xg = x.clone().requires_grad_(True)
    y = ops.reflection_pad2d(xg, (3, 3))
    dy = mk(y.shape, seed=19)
    y.backward(dy)

    xr = x.float().cpu().requires_grad_(True)
    yr = torch.nn.functional.pad(xr.permute(0, 3, 1, 2), (3, 3, 3, 3),
                                 mode="reflect").permute(0, 2, 3, 1)
    yr.backward(dy.float().cpu())
    check(y, yr, 0.02, "rpad:y")
    check(xg.grad, xr.grad, 0.02, "rpad:dx")


def test_losses_vs_oracle():
    a = mk((3, 16, 16, 3), seed=20)
    b = mk((3, 16, 16, 3), seed=21)
    for fn, tfn in ((ops.MAE, lambda d: d.abs()), (ops.MSE, lambda d: d * d)):
        ag = a.clone().requires_grad_(True)
        bg = b.clone().requires_grad_(True)
        out = fn(ag, bg)
        out.sum().backward()
        ref = tfn(b.float().cpu() - a.float().cpu()).mean(dim=(1, 2, 3))
        check(out, ref, 0.03, "loss")
        ar = a.float().cpu().requires_grad_(True)
        br = b.float().cpu().requires_grad_(True)
        tfn(br - ar).mean(dim=(1, 2, 3)).sum().backward()
        check(ag.grad, ar.grad, 0.05, "loss:ga")
        check(bg.grad, br.grad, 0.05, "loss:gb")

    # const-target variant
    bg = b.clone().requires_grad_(True)
    out = ops.MSE_const(bg, 1.0)
    out.sum().backward()
    d = b.float().cpu() - 1.0
    check(out, (d * d).mean(dim=(1, 2, 3)), 0.03, "mse_const")
    br = b.float().cpu().requires_grad_(True)
    ((br - 1.0) ** 2).mean(dim=(1, 2, 3)).sum().backward()
    check(bg.grad, br.grad, 0.05, "mse_const:g")


def test_adam_step_vs_cpu_formula():
    from cyclegan_amd.ops.adam import FusedAdam
    n = 10000
    g = torch.Generator().manual_seed(22)
    p0 = torch.randn(n, generator=g)
    gr = torch.randn(n, generator=g)

    pg = p0.clone().to(DEV)
    gg = gr.clone().to(DEV)
    opt_gpu = FusedAdam(pg, gg)
    pc = p0.clone()
    opt_cpu = FusedAdam(pc, gr.clone())
    for _ in range(3):
        opt_gpu.step()
        opt_cpu.step()
    assert torch.allclose(pg.cpu(), pc, atol=1e-6)
    assert torch.allclose(opt_gpu.m.cpu(), opt_cpu.m, atol=1e-6)
    assert torch.allclose(opt_gpu.v.cpu(), opt_cpu.v, atol=1e-6)


def test_native_ext_is_loaded():
    """Guards against silent eager fallback: the HIP .so must be loaded and
    used for GPU tensors."""
    assert backend.load_ext() is not None
    x = mk((1, 8, 8, 64), seed=23)
    w = mk((64, 3, 3, 64), seed=24)
    assert backend.use_hip(x, w)


def test_fp8_conv_fwd_vs_oracle():
    """fp8 e4m3 forward conv (per-tensor scales) vs fp32 oracle, fp8
    tolerance."""
    from cyclegan_amd.ops import conv as convmod
    convmod.set_fp8_mode(True)
    try:
        x = mk((12, 16, 16, 256), seed=30)
        w = mk((128, 3, 3, 256), seed=31, scale=0.2).float()
        y = ops.conv2d(x, w, None, 1, (1, 1, 1, 1), "reflect", "relu")
        ref = _conv_ref(x.float().cpu(), w.cpu(), None, 1, (1, 1, 1, 1), "reflect")
        ref = torch.relu(ref)
        check(y, ref, 0.15, "fp8:y")
        # backward still flows (bf16 kernels)
        xg = x.clone().requires_grad_(True)
        wg = w.clone().requires_grad_(True)
        y2 = ops.conv2d(xg, wg, None, 1, (1, 1, 1, 1), "reflect", "relu")
        y2.sum().backward()
        assert xg.grad is not None and wg.grad is not None
        assert torch.isfinite(xg.grad.float()).all()
    finally:
        convmod.set_fp8_mode(False)


def test_fp8_train_step(tmp_path):
    import argparse
    from cyclegan_amd.parallel import DistContext
    from cyclegan_amd.trainer import CycleGAN
    from cyclegan_amd.ops import conv as convmod
    a = argparse.Namespace()
    a.output_dir = str(tmp_path)
    a.batch_size = 1
    a.global_batch_size = 1
    a.num_residual_blocks = 1
    a.compute_dtype = torch.bfloat16
    a.fp8 = True
    torch.manual_seed(0)
    try:
        ctx = DistContext(device=torch.device("cuda", 0))
        gan = CycleGAN(a, ctx)
        x = torch.rand(1, 64, 64, 3, device=ctx.device, dtype=torch.bfloat16)
        r = gan.train_step(x, x)
        torch.cuda.synchronize()
        for k, v in r.items():
            assert torch.isfinite(v), k
    finally:
        convmod.set_fp8_mode(False)


@pytest.mark.parametrize("C", [64, 256, 512])
def test_instnorm_channel_widths(C):
    B, H, W = 2, 12, 12
    x = mk((B, H, W, C), seed=40 + C)
    g = torch.randn(C, generator=torch.Generator().manual_seed(41)).to(DEV) * 0.5
    bt = torch.randn(C, generator=torch.Generator().manual_seed(42)).to(DEV) * 0.1
    y = ops.instance_norm(x, g, bt, act="relu")
    ref = _in_ref(x.float().cpu(), g.float().cpu(), bt.float().cpu(), 1e-3, 1, 0.2, None)
    check(y, ref, 0.05, f"in_c{C}")


def test_conv_batch8_and_odd_spatial():
    # batched-call shapes (B=8) and non-power-of-two spatial
    x = mk((8, 24, 24, 64), seed=50)
    w = mk((128, 3, 3, 64), seed=51, scale=0.2)
    y = ops.conv2d(x, w, None, 2, "same", "zeros")
    ref = _conv_ref(x.float().cpu(), w.float().cpu(), None, 2,
                    same_pads(24, 24, 3, 3, 2), "zeros")
    check(y, ref, 0.05, "b8odd")


def test_conv_512_spatial_shape():
    x = mk((1, 128, 128, 64), seed=52)
    w = mk((64, 7, 7, 64), seed=53, scale=0.1)
    y = ops.conv2d(x, w, None, 1, (3, 3, 3, 3), "reflect")
    ref = _conv_ref(x.float().cpu(), w.float().cpu(), None, 1,
                    (3, 3, 3, 3), "reflect")
    check(y, ref, 0.05, "large_spatial")


def test_fp8_delayed_scaling_across_steps():
    """Delayed per-tensor scaling: the first call bootstraps amax; later
    calls quantize with the rolled previous-step amax. With shrinking
    activation ranges the stale scale only costs quantization headroom,
    so every step must stay within fp8 tolerance of the oracle."""
    from cyclegan_amd.ops import fp8_state
    from cyclegan_amd.ops import conv as convmod
    convmod.set_fp8_mode(True)
    try:
        w = mk((64, 3, 3, 256), seed=70, scale=0.2).float()
        for i, s in enumerate((1.0, 0.5, 0.25)):
            x = mk((12, 16, 16, 256), seed=71 + i) * s
            y = ops.conv2d(x, w, None, 1, "same", "zeros")
            ref = _conv_ref(x.float().cpu(), w.cpu(), None, 1,
                            same_pads(16, 16, 3, 3, 1), "zeros")
            check(y, ref, 0.15, f"fp8_delayed_{i}")
            fp8_state.roll()
            torch.cuda.synchronize()
    finally:
        convmod.set_fp8_mode(False)
