"""fp8-vs-bf16 per-shape A/B at train batch sizes (GPU box).

    python tools/fp8_micro.py
"""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
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
    return s.elapsed_time(t) / iters * 1000


def mk(*shape):
    return (torch.rand(*shape, device=DEV, dtype=torch.bfloat16) - 0.5)


sx = torch.tensor([1.0], device=DEV)
ap = torch.tensor([0.5], device=DEV)
ac = torch.zeros(1, device=DEV)

for B in (4, 8, 12):
    x = mk(B, 64, 64, 256)
    w = mk(256, 3, 3, 256) * 0.1
    fl = 2 * (B * 64 * 64) * 256 * 2304
    us_bf = timeit(lambda: e.conv2d_fwd(x, w, None, 1, 1, 1, 1, 1, True, 0, 0.2))
    xq = e.quant_fp8(x, sx)
    wq = e.quant_fp8(w, sx)
    us_f8 = timeit(lambda: e.conv2d_fp8_fwd(xq, wq, sx, None, None, 1, 1, 1, 1, 1, True, 0, 0.2))
    us_q = timeit(lambda: e.quant_fp8_d(x, ap, ac))
    print(f"K3 B={B:2d}: bf16 {us_bf:7.1f}us ({fl/us_bf/1e6:.0f} TF/s)  "
          f"fp8 {us_f8:7.1f}us ({fl/us_f8/1e6:.0f} TF/s)  quant {us_q:6.1f}us  "
          f"fp8+q vs bf16: {us_f8+us_q-us_bf:+7.1f}us")

# down conv 3x3 s2 at train batch
for B in (8, 12):
    x = mk(B, 128, 128, 128)
    w = mk(256, 3, 3, 128) * 0.1
    fl = 2 * (B * 64 * 64) * 256 * (9 * 128)
    us_bf = timeit(lambda: e.conv2d_fwd(x, w, None, 2, 0, 1, 0, 1, False, 0, 0.2))
    xq = e.quant_fp8(x, sx)
    wq = e.quant_fp8(w, sx)
    us_f8 = timeit(lambda: e.conv2d_fp8_fwd(xq, wq, sx, None, None, 2, 0, 1, 0, 1, False, 0, 0.2))
    us_q = timeit(lambda: e.quant_fp8_d(x, ap, ac))
    print(f"down B={B:2d}: bf16 {us_bf:7.1f}us  fp8 {us_f8:7.1f}us  quant {us_q:6.1f}us  "
          f"fp8+q vs bf16: {us_f8+us_q-us_bf:+7.1f}us")

# disc 4x4 s2 256->512-ish wide at 32^2 (s1) and 64^2 (s2)
for name, B, H, Cin, Cout, K, s, pads in (
        ("D_wide", 4, 32, 256, 512, 4, 1, (1, 2, 1, 2)),
        ("D_down2", 4, 64, 128, 256, 4, 2, (1, 2, 1, 2))):
    x = mk(B, H, H, Cin)
    w = mk(Cout, K, K, Cin) * 0.1
    OH = (H + pads[0] + pads[1] - K) // s + 1
    fl = 2 * (B * OH * OH) * Cout * (K * K * Cin)
    us_bf = timeit(lambda: e.conv2d_fwd(x, w, None, s, *pads, False, 0, 0.2))
    xq = e.quant_fp8(x, sx)
    wq = e.quant_fp8(w, sx)
    us_f8 = timeit(lambda: e.conv2d_fp8_fwd(xq, wq, sx, None, None, s, *pads, False, 0, 0.2))
    us_q = timeit(lambda: e.quant_fp8_d(x, ap, ac))
    print(f"{name} B={B}: bf16 {us_bf:7.1f}us  fp8 {us_f8:7.1f}us  quant {us_q:6.1f}us  "
          f"fp8+q vs bf16: {us_f8+us_q-us_bf:+7.1f}us")
