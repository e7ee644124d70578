"""CycleGAN training CLI — same contract as the reference
(/root/reference/main.py:405-413: --output_dir --epochs --batch_size
--verbose --clear_output_dir; --batch_size is PER-REPLICA, global batch =
world_size * batch_size).

Single GPU / CPU:    python main.py --output_dir runs
Multi-GPU (DP/RCCL): python -m torch.distributed.run --nnodes=1
                     --nproc-per-node 8 --master-addr 127.0.0.1 main.py ...

Extra flags cover the BASELINE.json configs only (synthetic data source,
image size, resblock count, dtype).
"""

from __future__ import annotations

import argparse
import os
from shutil import rmtree
from time import time

import numpy as np
import torch
from tqdm import tqdm

from cyclegan_amd.parallel import DistContext
from cyclegan_amd.trainer import CycleGAN
from cyclegan_amd.data import Pipeline, DevicePrefetcher
from cyclegan_amd import utils


def train(args, pipe, gan, summary, epoch: int):
    results = {}
    # steady-state step as ONE hip graph at N=1 (same path bench.py
    # measures); short final batches and multi-rank training run eager
    use_graph = (gan.device.type == "cuda" and gan.ctx.world_size == 1
                 and not os.environ.get("CYG_NO_GRAPH"))
    it = DevicePrefetcher(pipe.train_epoch(epoch), gan.device, gan.compute_dtype)
    for x, y in tqdm(it, desc="Train", total=pipe.train_steps,
                     disable=args.verbose == 0 or not gan.ctx.is_main):
        if use_graph and x.shape[0] == args.batch_size:
            if gan.graphed is None:
                from cyclegan_amd.trainer import GraphedStep
                gan.graphed = GraphedStep(gan, x, y, preserve_state=True)
            result = gan.graphed.call_cloned(x, y)
        else:
            result = gan.train_step(x, y)
        utils.append_dict(results, result)
    reduced = gan.reduce_results(results)
    if summary is not None:
        for key, value in reduced.items():
            summary.scalar(key, value, step=epoch, training=True)


def test(args, pipe, gan, summary, epoch: int):
    results = {}
    it = DevicePrefetcher(pipe.test_epoch(), gan.device, gan.compute_dtype)
    for x, y in tqdm(it, desc="Test", total=pipe.test_steps,
                     disable=args.verbose == 0 or not gan.ctx.is_main):
        result = gan.test_step(x, y)
        utils.append_dict(results, result)
    reduced = gan.reduce_results(results)
    if summary is not None:
        for key, value in reduced.items():
            summary.scalar(key, value, step=epoch, training=False)
    return reduced


def main(args):
    ctx = DistContext()
    if ctx.is_main:
        if args.clear_output_dir and os.path.exists(args.output_dir):
            rmtree(args.output_dir)
        os.makedirs(args.output_dir, exist_ok=True)
    ctx.barrier()

    # reference seeds np/tf with 1234 (main.py:366-367); --seed overrides
    np.random.seed(args.seed)
    torch.manual_seed(args.seed)

    args.global_batch_size = ctx.world_size * args.batch_size
    if args.dtype:
        args.compute_dtype = {"fp32": torch.float32, "bf16": torch.bfloat16,
                              "fp8": torch.bfloat16}[args.dtype]
        args.fp8 = args.dtype == "fp8"
    else:
        args.compute_dtype = None
    print(f"Number of devices: {ctx.world_size}")

    summary = utils.Summary(args.output_dir) if ctx.is_main else None

    pipe = Pipeline(args, ctx, image_size=args.image_size)
    args.train_steps, args.test_steps = pipe.train_steps, pipe.test_steps

    gan = CycleGAN(args, ctx)
    gan.load_checkpoint()

    for epoch in range(args.epochs):
        if ctx.is_main:
            print(f"Epoch {epoch + 1:03d}/{args.epochs:03d}")

        start = time()
        train(args, pipe, gan, summary, epoch)
        results = test(args, pipe, gan, summary, epoch)
        end = time()
        if summary is not None:
            summary.scalar("elapse", end - start, step=epoch)
            if args.verbose == 2:
                # extension: RCCL all-reduce GPU-time per epoch (hip events
                # around each flat-grad all-reduce, parallel/dp.py)
                summary.scalar("comm/all_reduce_ms", gan.sync.pop_comm_ms(),
                               step=epoch)

        if ctx.is_main and results:
            # (the reference console print swaps two labels, main.py:394-397
            #  — corrected here; TB values are identical)
            print(f'MAE(X, F(G(X))): {results["error/MAE(X, F(G(X)))"]:.04f}\t\t'
                  f'MAE(Y, G(F(Y))): {results["error/MAE(Y, G(F(Y)))"]:.04f}\n'
                  f'MAE(X, F(X)): {results["error/MAE(X, F(X))"]:.04f}\t\t'
                  f'MAE(Y, G(Y)): {results["error/MAE(Y, G(Y))"]:.04f}\n'
                  f'Elapse: {end - start:.02f}s\n')

        if epoch % 10 == 0 or epoch == args.epochs - 1:
            gan.save_checkpoint()
            if ctx.is_main:
                utils.plot_cycle(pipe.plot_pairs(), gan, summary, epoch)
            ctx.barrier()

    if summary is not None:
        summary.close()


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("--output_dir", default="runs")
    parser.add_argument("--epochs", default=200, type=int)
    parser.add_argument("--batch_size", default=1, type=int,
                        help="per-replica batch size")
    parser.add_argument("--verbose", default=1, type=int, choices=[0, 1, 2])
    parser.add_argument("--clear_output_dir", action="store_true")
    # MI355X/BASELINE extras
    parser.add_argument("--data_dir", default=None,
                        help="trainA/trainB/testA/testB image folders; "
                             "default: synthetic horse2zebra-shaped data")
    parser.add_argument("--image_size", default=256, type=int)
    parser.add_argument("--num_residual_blocks", default=9, type=int)
    parser.add_argument("--dtype", default=None, choices=[None, "fp32", "bf16", "fp8"])
    parser.add_argument("--num_train_samples", default=None, type=int)
    parser.add_argument("--num_test_samples", default=None, type=int)
    parser.add_argument("--seed", default=1234, type=int)

    main(parser.parse_args())
