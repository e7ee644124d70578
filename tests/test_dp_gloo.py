"""DP equivalence without a cluster: 2 gloo processes on CPU must produce
the same update as 1 process with the same global batch (MirroredStrategy
mirror semantics)."""

import argparse
import os
import tempfile

import torch
import torch.multiprocessing as mp
import pytest

from cyclegan_amd.trainer import CycleGAN


def _make_args(outdir, batch, global_batch):
    a = argparse.Namespace()
    a.output_dir = outdir
    a.batch_size = batch
    a.global_batch_size = global_batch
    a.num_residual_blocks = 1
    a.compute_dtype = torch.float32
    return a


def _data():
    g = torch.Generator().manual_seed(99)
    x = torch.rand(2, 16, 16, 3, generator=g)
    y = torch.rand(2, 16, 16, 3, generator=g)
    return x, y


def _worker(rank, world, port, outdir):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    from cyclegan_amd.parallel import DistContext
    torch.manual_seed(1234)  # same init on both; broadcast also enforces it
    ctx = DistContext(device=torch.device("cpu"))
    gan = CycleGAN(_make_args(outdir, 1, 2), ctx)
    x, y = _data()
    xs = x[rank:rank + 1]
    ys = y[rank:rank + 1]
    r = gan.train_step(xs, ys)
    if rank == 0:
        torch.save({
            "flat_G": gan.groups["G"].flat_param.detach().clone(),
            "flat_X": gan.groups["X"].flat_param.detach().clone(),
            "loss_G_total": r["loss_G/total"].item(),
        }, os.path.join(outdir, "rank0_result.pt"))
    torch.distributed.destroy_process_group()


@pytest.mark.timeout(300)
def test_two_rank_matches_single_rank(tmp_path):
    port = 29511
    ctxq = mp.get_context("spawn")
    procs = [ctxq.Process(target=_worker, args=(r, 2, port, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    got = torch.load(os.path.join(str(tmp_path), "rank0_result.pt"),
                     weights_only=True)

    # single-process, global batch 2
    os.environ.update(RANK="0", WORLD_SIZE="1", LOCAL_RANK="0")
    from cyclegan_amd.parallel import DistContext
    torch.manual_seed(1234)
    ctx = DistContext(device=torch.device("cpu"))
    gan = CycleGAN(_make_args(str(tmp_path), 2, 2), ctx)
    x, y = _data()
    r = gan.train_step(x, y)

    assert torch.allclose(gan.groups["G"].flat_param, got["flat_G"], atol=1e-6)
    assert torch.allclose(gan.groups["X"].flat_param, got["flat_X"], atol=1e-6)
    # rank-0 partial loss + rank-1 partial = global mean; compare after manual sum
    # (the 2-rank run reports the SUM-all-reduced value only at epoch end; here
    #  we just check single-rank loss is finite and comparable in magnitude)
    assert abs(r["loss_G/total"].item()) < 100


def _worker_short_batch(rank, world, port, outdir):
    """Final global batch of 1 sample: rank 0 gets it, rank 1's slice is
    empty — both must finish the epoch with IDENTICAL replicas."""
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    from cyclegan_amd.parallel import DistContext
    torch.manual_seed(1234)
    ctx = DistContext(device=torch.device("cpu"))
    gan = CycleGAN(_make_args(outdir, 1, 2), ctx)
    x, y = _data()
    # step 1: full batch (1 sample per rank); step 2: short (rank 1 empty)
    gan.train_step(x[rank:rank + 1], y[rank:rank + 1])
    if rank == 0:
        r = gan.train_step(x[1:2], y[1:2])
    else:
        r = gan.train_step(x[:0], y[:0])
    for v in r.values():
        assert torch.isfinite(v)
    te = gan.test_step(x[:0], y[:0]) if rank == 1 else gan.test_step(x[:1], y[:1])
    assert set(te) == set(gan._TEST_KEYS)
    torch.save({"flat_G": gan.groups["G"].flat_param.detach().clone(),
                "t": gan.optimizers["G"].t},
               os.path.join(outdir, f"short_rank{rank}.pt"))
    torch.distributed.destroy_process_group()


@pytest.mark.timeout(300)
def test_short_final_batch_keeps_replicas_synced(tmp_path):
    port = 29517
    ctxq = mp.get_context("spawn")
    procs = [ctxq.Process(target=_worker_short_batch,
                          args=(r, 2, port, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
        assert p.exitcode == 0
    r0 = torch.load(os.path.join(str(tmp_path), "short_rank0.pt"))
    r1 = torch.load(os.path.join(str(tmp_path), "short_rank1.pt"))
    assert r0["t"] == r1["t"] == 2
    assert torch.equal(r0["flat_G"], r1["flat_G"])


def _worker4(rank, world, port, outdir):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    from cyclegan_amd.parallel import DistContext
    torch.manual_seed(1234)
    ctx = DistContext(device=torch.device("cpu"))
    gan = CycleGAN(_make_args(outdir, 1, 4), ctx)
    g = torch.Generator().manual_seed(99)
    x = torch.rand(4, 16, 16, 3, generator=g)
    y = torch.rand(4, 16, 16, 3, generator=g)
    gan.train_step(x[rank:rank + 1], y[rank:rank + 1])
    if rank == 0:
        torch.save({"flat_G": gan.groups["G"].flat_param.detach().clone()},
                   os.path.join(outdir, "r4_result.pt"))
    torch.distributed.destroy_process_group()


@pytest.mark.timeout(300)
def test_four_rank_matches_single_rank(tmp_path):
    """4-way DP (the driver runs up to 8): 4 gloo ranks with per-rank
    batch 1 must equal 1 process with global batch 4."""
    port = 29523
    ctxq = mp.get_context("spawn")
    procs = [ctxq.Process(target=_worker4, args=(r, 4, port, str(tmp_path)))
             for r in range(4)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    got = torch.load(os.path.join(str(tmp_path), "r4_result.pt"),
                     weights_only=True)

    os.environ.update(RANK="0", WORLD_SIZE="1", LOCAL_RANK="0")
    from cyclegan_amd.parallel import DistContext
    torch.manual_seed(1234)
    ctx = DistContext(device=torch.device("cpu"))
    gan = CycleGAN(_make_args(str(tmp_path), 4, 4), ctx)
    g = torch.Generator().manual_seed(99)
    x = torch.rand(4, 16, 16, 3, generator=g)
    y = torch.rand(4, 16, 16, 3, generator=g)
    gan.train_step(x, y)
    assert torch.allclose(gan.groups["G"].flat_param, got["flat_G"], atol=1e-6)
