"""CPU tests for the auxiliary tooling (translate CLI, profiler)."""

import argparse
import subprocess
import sys
import os

import torch


def test_translate_cli_cpu(tmp_path):
    # make a checkpoint with the CPU trainer, then run translate.py on it
    from cyclegan_amd.parallel import DistContext
    from cyclegan_amd.trainer import CycleGAN
    a = argparse.Namespace()
    a.output_dir = str(tmp_path)
    a.batch_size = 1
    a.global_batch_size = 1
    a.num_residual_blocks = 1
    a.compute_dtype = torch.float32
    ctx = DistContext(device=torch.device("cpu"))
    gan = CycleGAN(a, ctx)
    gan.save_checkpoint()

    out = tmp_path / "translated"
    r = subprocess.run(
        [sys.executable, "translate.py",
         "--checkpoint", str(tmp_path / "checkpoints" / "checkpoint.pt"),
         "--synthetic", "2", "--image_size", "32",
         "--num_residual_blocks", "1", "--output_dir", str(out)],
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    assert sorted(os.listdir(out)) == ["00000.png", "00001.png"]


def test_step_timer():
    from cyclegan_amd.utils.profiler import StepTimer
    t = StepTimer()
    t.start("a")
    t.stop("a")
    t.start("a")
    t.stop("a")
    s = t.summary()
    assert "a" in s and s["a"] >= 0
    assert t.counts["a"] == 2
