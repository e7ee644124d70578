"""Benchmark contract for the driver.

Measures the BASELINE.json north-star metric: images/sec (whole node) for
CycleGAN horse2zebra-shaped training at 256x256, bf16 compute, 9-resblock
generator, per-GPU batch 4, synthetic data + random-init weights.

``images`` counts every image consumed by a train step: a step processes
``global_batch`` horse images AND ``global_batch`` zebra images, so
images/step = 2 * global_batch (pairs/sec = value / 2, also reported).

Usage:  python bench.py [--gpus N] [--steps K] [--warmup W]
Multi-GPU (driver):  python -m torch.distributed.run --nnodes=1
    --nproc-per-node N --master-addr 127.0.0.1 bench.py --gpus N ...
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch

from cyclegan_amd.parallel import DistContext
from cyclegan_amd.trainer import CycleGAN


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=100)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--batch_size", type=int, default=4, help="per-GPU batch")
    ap.add_argument("--image_size", type=int, default=256)
    ap.add_argument("--num_residual_blocks", type=int, default=9)
    ap.add_argument("--dtype", default="bf16", choices=["bf16", "fp32", "fp8"])
    ap.add_argument("--output_dir", default="/tmp/cyclegan_bench")
    args = ap.parse_args()

    ctx = DistContext()
    on_gpu = ctx.device.type == "cuda"
    if not on_gpu:
        # plumbing-only CPU fallback (BASELINE config 1); the measured
        # configuration requires the MI355X box.
        args.image_size = 64
        args.num_residual_blocks = 1
        args.batch_size = 1
        args.dtype = "fp32"
        args.steps = min(args.steps, 20)
        args.warmup = min(args.warmup, 5)

    torch.manual_seed(1234)
    args.global_batch_size = ctx.world_size * args.batch_size
    args.compute_dtype = (torch.float32 if args.dtype == "fp32"
                          else torch.bfloat16)
    args.fp8 = args.dtype == "fp8"
    os.makedirs(args.output_dir, exist_ok=True)

    gan = CycleGAN(args, ctx)

    # synthetic input pool, generated on-device (no host pipeline in the
    # timed region; data shape = the horse2zebra 256x256 config)
    g = torch.Generator(device="cpu").manual_seed(4321 + ctx.rank)
    pool = []
    for _ in range(4):
        x = torch.rand(args.batch_size, args.image_size, args.image_size, 3,
                       generator=g) * 2 - 1
        y = torch.rand(args.batch_size, args.image_size, args.image_size, 3,
                       generator=g) * 2 - 1
        pool.append((x.to(ctx.device, args.compute_dtype),
                     y.to(ctx.device, args.compute_dtype)))

    def sync():
        ctx.barrier()
        if on_gpu:
            torch.cuda.synchronize()

    for i in range(args.warmup):
        gan.train_step(*pool[i % len(pool)])

    # steady-state step as ONE hip graph (multi-rank capture is opt-in,
    # CYG_GRAPH_DIST=1 — proven with RCCL at 1 rank) — every replay still
    # copies the step's input batch in and runs the full
    # forward/backward/optimizer.
    step = gan.train_step
    if on_gpu and not os.environ.get("CYG_NO_GRAPH"):
        if os.environ.get("CYG_SEGMENTED") == "1":
            # five RCCL-free graphs + eager all-reduces: measured ==
            # full-graph at N=1 no-dist (324.8 vs 324.3) but == eager
            # under a communicator (312.9 vs 314.1 at 1-rank force-dist:
            # the eager AR host overhead, not launch gaps, is the dist
            # cost) — so it stays opt-in
            from cyclegan_amd.trainer import SegmentedGraphedStep
            step = SegmentedGraphedStep(gan, *pool[0])
        elif (ctx.world_size == 1
                or os.environ.get("CYG_GRAPH_DIST") == "1"):
            from cyclegan_amd.trainer import GraphedStep
            step = GraphedStep(gan, *pool[0])
    sync()

    t0 = time.perf_counter()
    for i in range(args.steps):
        step(*pool[i % len(pool)])
    if on_gpu:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0
    ctx.barrier()

    # MAX elapsed over ranks -> whole-job throughput (NCCL needs a device
    # tensor)
    t = torch.tensor([elapsed], dtype=torch.float64,
                     device=ctx.device if on_gpu else "cpu")
    if ctx.distributed:
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
    elapsed = t.item()

    images_per_step = 2 * args.global_batch_size
    value = images_per_step * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if ctx.is_main:
        print(json.dumps({
            "metric": "images/sec (whole node) horse2zebra 256×256",
            "value": round(value, 3),
            "unit": "images/s",
            "n_gpus": ctx.world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {
                "model": f"cyclegan-resnet{args.num_residual_blocks}-patchgan",
                "global_batch": args.global_batch_size,
                "per_gpu_batch": args.batch_size,
                "image_size": args.image_size,
                "images_per_step": images_per_step,
                "pairs_per_sec": round(value / 2, 3),
                "parallelism": f"dp{ctx.world_size}",
            },
        }))


if __name__ == "__main__":
    main()
