"""Per-op microbenchmark on the hot shapes (run on the GPU box).

    python tools/perf_micro.py
"""
import torch
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from cyclegan_amd.ops import backend

e = backend.ext()
DEV = "cuda:0"
torch.manual_seed(0)


def timeit(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    t = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    t.record()
    torch.cuda.synchronize()
    return s.elapsed_time(t) / iters * 1000  # us


def mk(*shape):
    return (torch.rand(*shape, device=DEV, dtype=torch.bfloat16) - 0.5)


B = 4
cases = []

# K3 resblock conv: 64x64x256, reflect pad 1
x = mk(B, 64, 64, 256)
w = mk(256, 3, 3, 256) * 0.1
wt = w.permute(3, 1, 2, 0).contiguous()
dy = mk(B, 64, 64, 256)
fl = 2 * (B * 64 * 64) * 256 * 2304
cases += [
    ("K3 fwd      ", fl, lambda: e.conv2d_fwd(x, w, None, 1, 1, 1, 1, 1, True, 0, 0.2)),
    ("K3 dgrad    ", fl, lambda: e.conv2d_dgrad(dy, wt, 64, 64, 1, 1, 1, 1, 1, True)),
    ("K3 wgrad    ", fl, lambda: e.conv2d_wgrad(x, dy, 3, 3, 1, 1, 1, True)),
]

# down conv 3x3 s2 128->256 @128^2
x2 = mk(B, 128, 128, 128)
w2 = mk(256, 3, 3, 128) * 0.1
wt2 = w2.permute(3, 1, 2, 0).contiguous()
dy2 = mk(B, 64, 64, 256)
fl2 = 2 * (B * 64 * 64) * 256 * (9 * 128)
cases += [
    ("down fwd    ", fl2, lambda: e.conv2d_fwd(x2, w2, None, 2, 0, 1, 0, 1, False, 0, 0.2)),
    ("down dgrad  ", fl2, lambda: e.conv2d_dgrad(dy2, wt2, 128, 128, 2, 0, 1, 0, 1, False)),
    ("down wgrad  ", fl2, lambda: e.conv2d_wgrad(x2, dy2, 3, 3, 2, 0, 0, False)),
]

# upsample convT 3x3 s2 256->128: in 64^2 -> out 128^2
x3 = mk(B, 64, 64, 256)
w3 = mk(128, 3, 3, 256) * 0.1
fl3 = 2 * (B * 64 * 64) * 128 * (9 * 256) // 4 * 4  # raw gather count incl. waste
cases += [
    ("convT fwd   ", fl3, lambda: e.convt2d_fwd(x3, w3, None, 2, 0, 0, 128, 128, 0, 0.2)),
]

# stem 7x7 (padded to 8ch) @256^2
x4 = mk(B, 256, 256, 8)
w4 = mk(64, 7, 7, 8) * 0.1
fl4 = 2 * (B * 256 * 256) * 64 * (49 * 8)
cases += [
    ("stem fwd    ", fl4, lambda: e.conv2d_fwd(x4, w4, None, 1, 3, 3, 3, 3, True, 0, 0.2)),
]
dy4 = mk(B, 256, 256, 64)
cases += [
    ("stem wgrad  ", fl4, lambda: e.conv2d_wgrad(x4, dy4, 7, 7, 1, 3, 3, True)),
]

# generator head 7x7 64->3 (padded to 8) @256^2, reflect pad 3 (BN16 tile path)
x5 = mk(B, 256, 256, 64)
w5 = mk(8, 7, 7, 64) * 0.1
wt5 = w5.permute(3, 1, 2, 0).contiguous()
dy5 = mk(B, 256, 256, 8)
fl5 = 2 * (B * 256 * 256) * 8 * (49 * 64)
cases += [
    ("head fwd    ", fl5, lambda: e.conv2d_fwd(x5, w5, None, 1, 3, 3, 3, 3, True, 0, 0.2)),
    ("head dgrad  ", fl5, lambda: e.conv2d_dgrad(dy5, wt5, 256, 256, 1, 3, 3, 3, 3, True)),
    ("head wgrad  ", fl5, lambda: e.conv2d_wgrad(x5, dy5, 7, 7, 1, 3, 3, True)),
]

# packed head (8-pixel N-packing): fwd via reflect_pad + folded conv,
# wgrad on the folded image (ops.conv._ConvHeadPackedFn path)
from cyclegan_amd.ops.shadow import _pack_head_weight_torch
w5pk = _pack_head_weight_torch(w5.float()).to(torch.bfloat16).contiguous()
xp5 = e.reflect_pad_fwd(x5, 3, 3, 3, 5)
xf5 = xp5.view(B, 262, 33, 512)
dypk5 = dy5.view(B, 256, 32, 64)


def head_packed_fwd():
    xp = e.reflect_pad_fwd(x5, 3, 3, 3, 5)
    xf = xp.view(B, 262, 33, 512)
    return e.conv2d_fwd(xf, w5pk, None, 1, 0, 0, 0, 0, False, 0, 0.2)


cases += [
    ("headP fwd   ", fl5, head_packed_fwd),
    ("headP wgrad ", fl5, lambda: e.conv2d_wgrad(xf5, dypk5, 7, 2, 1, 0, 0, False)),
]

# InstanceNorm 64^2 x 256
g = torch.rand(256, device=DEV)
bta = torch.rand(256, device=DEV)
xin = mk(B, 64, 64, 256)
res = mk(B, 64, 64, 256)
byts = B * 64 * 64 * 256 * 2
xin2 = mk(B, 256, 256, 64)
g64 = torch.rand(64, device=DEV)
b64 = torch.rand(64, device=DEV)
cases += [
    ("IN fwd      ", None, lambda: e.instnorm_fwd(xin, g, bta, 1e-3, 1, 0.2, None)),
    ("IN fwd 256^2", None, lambda: e.instnorm_fwd(xin2, g64, b64, 1e-3, 1, 0.2, None)),
]
mean_ = torch.rand(B, 256, device=DEV)
rstd_ = torch.rand(B, 256, device=DEV) + 0.5
cases += [
    ("IN bwd      ", None, lambda: e.instnorm_bwd(dy, xin, g, mean_, rstd_, None, 0, 0.2)),
]

for name, fl, fn in cases:
    us = timeit(fn)
    tf = (fl / (us * 1e-6) / 1e12) if fl else 0
    print(f"{name} {us:9.1f} us   {tf:7.1f} TF/s")

# fp8 K3 conv
fl8 = 2 * (B * 64 * 64) * 256 * 2304
sx = torch.tensor([1.0], device=DEV)
xq = e.quant_fp8(x, sx)
wq = e.quant_fp8(w, sx)
dqs = torch.tensor([1.0], device=DEV)
us8 = timeit(lambda: e.conv2d_fp8_fwd(xq, wq, dqs, None, None, 1, 1, 1, 1, 1, True, 0, 0.2))
print(f"K3 fp8 fwd   {us8:9.1f} us   {fl8/(us8*1e-6)/1e12:7.1f} TF/s")
