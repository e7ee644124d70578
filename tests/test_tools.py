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


def test_main_cli_end_to_end(tmp_path):
    """Reference CLI contract (main.py:405-413): one tiny epoch on CPU
    writes train+test event files, a checkpoint, and auto-resumes."""
    out = tmp_path / "run"
    cmd = [sys.executable, "main.py", "--output_dir", str(out),
           "--epochs", "1", "--batch_size", "2", "--verbose", "0",
           "--image_size", "32", "--num_residual_blocks", "1",
           "--num_train_samples", "4", "--num_test_samples", "2",
           "--dtype", "fp32"]
    cwd = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(cmd, cwd=cwd, capture_output=True, text=True,
                       timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    assert os.path.exists(out / "checkpoints" / "checkpoint.pt")
    import glob as g
    assert g.glob(str(out / "events.out.tfevents.*"))
    assert g.glob(str(out / "test" / "events.out.tfevents.*"))
    # auto-resume: second run must report loading the checkpoint
    r2 = subprocess.run(cmd, cwd=cwd, capture_output=True, text=True,
                        timeout=600)
    assert r2.returncode == 0, r2.stderr[-2000:]
    assert "restor" in (r2.stdout + r2.stderr).lower() or \
           "resum" in (r2.stdout + r2.stderr).lower() or \
           "loaded" in (r2.stdout + r2.stderr).lower()


def test_bench_json_contract(tmp_path):
    """The driver parses bench.py's final line as JSON with a fixed
    schema; guard every required key and the value semantics."""
    import json
    r = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--output_dir", str(tmp_path)],
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    d = json.loads(r.stdout.strip().splitlines()[-1])
    for k in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
              "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
              "dtype", "data", "config"):
        assert k in d, k
    assert d["metric"].startswith("images/sec")
    assert d["higher_is_better"] is True and d["scaling"] == "weak"
    assert d["data"] == "synthetic" and d["n_gpus"] == 1
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["config"]["global_batch"] == d["n_gpus"] * d["config"]["per_gpu_batch"]


def test_fused_adam_graph_mode_lr_math(tmp_path):
    """advance_lr must track the eager TF bias-correction schedule."""
    import math
    from cyclegan_amd.ops.adam import FusedAdam
    p = torch.zeros(4)
    g = torch.zeros(4)
    opt = FusedAdam(p, g)
    opt.prepare_graph()
    assert opt.graph_mode and opt._lr_t_dev is not None
    for t in range(1, 5):
        opt.advance_lr()
        want = opt.lr * math.sqrt(1 - opt.b2 ** t) / (1 - opt.b1 ** t)
        assert opt.t == t
        assert abs(opt._lr_t_dev.item() - want) < 1e-9 * (1 + abs(want))
