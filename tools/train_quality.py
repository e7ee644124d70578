"""Training-quality validation.

Runs a multi-thousand-step CycleGAN training at the headline config on
synthetic unpaired data (no network -> no real horse2zebra; data noted as
synthetic) and writes the full loss trajectory to a CSV artifact.

Two modes compared offline:
  python tools/train_quality.py --steps 3000 --tag bf16            (HIP bf16)
  python tools/train_quality.py --steps 600 --dtype fp32 --tag fp32 (torch fp32 oracle path)

The fp32 run uses the stock-PyTorch oracle kernels (CYGAN_FORCE_TORCH=1)
— an independent numerics baseline: if bf16+TF-Adam+eps-1e-3 IN drifted,
the curves would separate within a few hundred steps.

Checks printed at the end:
  - cycle losses strictly decreasing (smoothed) over the run
  - discriminator losses near the LSGAN equilibrium (0.25) without
    collapse to 0 or 0.5
  - generator adversarial loss bounded away from 0 and 1 long-term
"""
import argparse
import csv
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=3000)
    ap.add_argument("--batch_size", type=int, default=4)
    ap.add_argument("--image_size", type=int, default=256)
    ap.add_argument("--dtype", default="bf16", choices=["bf16", "fp32", "fp8"])
    ap.add_argument("--pool", type=int, default=128, help="synthetic pool size")
    ap.add_argument("--log_every", type=int, default=25)
    ap.add_argument("--tag", default="run")
    ap.add_argument("--out", default="gpurun_out/train_quality")
    a = ap.parse_args()

    if a.dtype == "fp32":
        os.environ["CYGAN_FORCE_TORCH"] = "1"  # independent oracle path

    from cyclegan_amd.parallel import DistContext
    from cyclegan_amd.trainer import CycleGAN

    a.output_dir = "/tmp/tq"
    a.global_batch_size = a.batch_size
    a.num_residual_blocks = 9
    a.compute_dtype = torch.float32 if a.dtype == "fp32" else torch.bfloat16
    a.fp8 = a.dtype == "fp8"
    torch.manual_seed(1234)
    ctx = DistContext(device=torch.device("cuda", 0))
    gan = CycleGAN(a, ctx)

    # synthetic unpaired pool with image-like statistics: smooth random
    # fields (low-frequency) rather than white noise, two distinct
    # "domains" (different blur + bias) so the translation task is
    # non-trivial
    g = torch.Generator().manual_seed(99)

    def make(domain):
        base = torch.rand(a.pool, 3, a.image_size // 8, a.image_size // 8,
                          generator=g)
        up = torch.nn.functional.interpolate(
            base, size=(a.image_size, a.image_size), mode="bilinear",
            align_corners=False)
        if domain == 1:
            up = 1.0 - up * 0.8          # domain shift
        return (up.permute(0, 2, 3, 1) * 2 - 1).contiguous()

    A = make(0).to(ctx.device, gan.compute_dtype)
    B = make(1).to(ctx.device, gan.compute_dtype)
    gi = torch.Generator().manual_seed(7)

    os.makedirs(a.out, exist_ok=True)
    path = os.path.join(a.out, f"trajectory_{a.tag}.csv")
    keys = list(gan._TRAIN_KEYS)
    rows = []
    for i in range(a.steps):
        ia = torch.randint(0, a.pool - a.batch_size, (1,), generator=gi).item()
        ib = torch.randint(0, a.pool - a.batch_size, (1,), generator=gi).item()
        r = gan.train_step(A[ia:ia + a.batch_size], B[ib:ib + a.batch_size])
        if i % a.log_every == 0 or i == a.steps - 1:
            torch.cuda.synchronize()
            vals = [float(r[k]) for k in keys]
            rows.append([i] + vals)
            if i % (a.log_every * 10) == 0:
                print(f"step {i:5d} " +
                      " ".join(f"{k.split('/')[-1]}={v:.3f}"
                               for k, v in zip(keys, vals)))
    with open(path, "w", newline="") as f:
        wcsv = csv.writer(f)
        wcsv.writerow(["step"] + keys)
        wcsv.writerows(rows)
    print("wrote", path)

    # ---- health checks ----
    import statistics
    col = {k: [r[1 + keys.index(k)] for r in rows] for k in keys}
    n = len(rows)
    q = max(2, n // 4)

    def mean(v):
        return statistics.fmean(v)

    first_cy = mean(col["loss_G/cycle"][:q])
    last_cy = mean(col["loss_G/cycle"][-q:])
    dx = mean(col["loss_X/loss"][-q:])
    dy_ = mean(col["loss_Y/loss"][-q:])
    gl = mean(col["loss_G/loss"][-q:])
    print(f"cycle first-quarter {first_cy:.3f} -> last-quarter {last_cy:.3f} "
          f"({'DECREASING ok' if last_cy < first_cy else 'NOT DECREASING'})")
    print(f"D losses last quarter: X {dx:.3f} Y {dy_:.3f} "
          f"({'equilibrium ok' if 0.03 < dx < 0.49 and 0.03 < dy_ < 0.49 else 'SUSPECT'})")
    print(f"G adversarial last quarter: {gl:.3f} "
          f"({'ok' if 0.05 < gl < 3.0 else 'SUSPECT'})")
    bad = any(not (v == v and abs(v) < 1e6) for vs in col.values() for v in vs)
    print("finite:", "ok" if not bad else "NON-FINITE VALUES")


if __name__ == "__main__":
    main()
