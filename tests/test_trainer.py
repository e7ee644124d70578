import argparse
import os

import torch
import pytest

from cyclegan_amd.parallel import DistContext
from cyclegan_amd.trainer import CycleGAN


def make_args(tmp_path, batch=2):
    a = argparse.Namespace()
    a.output_dir = str(tmp_path)
    a.batch_size = batch
    a.global_batch_size = batch
    a.num_residual_blocks = 1
    a.compute_dtype = torch.float32
    return a


@pytest.fixture
def ctx():
    return DistContext(device=torch.device("cpu"))


def test_train_step_runs_and_learns(tmp_path, ctx):
    torch.manual_seed(0)
    args = make_args(tmp_path)
    gan = CycleGAN(args, ctx)
    x = torch.rand(2, 32, 32, 3) * 2 - 1
    y = torch.rand(2, 32, 32, 3) * 2 - 1
    r0 = gan.train_step(x, y)
    assert set(r0) == {
        "loss_G/loss", "loss_G/cycle", "loss_G/identity", "loss_G/total",
        "loss_F/loss", "loss_F/cycle", "loss_F/identity", "loss_F/total",
        "loss_X/loss", "loss_Y/loss"}
    for _ in range(15):
        r = gan.train_step(x, y)
    # supervised-ish components must decrease on a fixed batch
    assert r["loss_G/cycle"] < r0["loss_G/cycle"]
    assert r["loss_F/cycle"] < r0["loss_F/cycle"]


def test_grads_stay_flat_views(tmp_path, ctx):
    torch.manual_seed(0)
    args = make_args(tmp_path)
    gan = CycleGAN(args, ctx)
    x = torch.rand(2, 16, 16, 3)
    y = torch.rand(2, 16, 16, 3)
    gan.train_step(x, y)
    for g in gan.groups.values():
        assert g.check_views()
        assert g.flat_grad.abs().sum() > 0


def test_gradient_group_isolation(tmp_path, ctx):
    """G's backward must not leave gradients in the discriminators
    (reference var_list semantics, main.py:249-260)."""
    torch.manual_seed(0)
    args = make_args(tmp_path)
    gan = CycleGAN(args, ctx)
    x = torch.rand(2, 16, 16, 3)
    y = torch.rand(2, 16, 16, 3)
    for g in gan.groups.values():
        g.zero_grad()
    fake_y = gan.G(x)
    G_loss = gan.generator_loss(gan.Y(fake_y))
    torch.autograd.backward(G_loss, inputs=gan.groups["G"].params)
    assert gan.groups["G"].flat_grad.abs().sum() > 0  # flows through frozen D
    assert gan.groups["Y"].flat_grad.abs().sum() == 0


def test_test_step_keys(tmp_path, ctx):
    args = make_args(tmp_path)
    gan = CycleGAN(args, ctx)
    r = gan.test_step(torch.rand(1, 16, 16, 3), torch.rand(1, 16, 16, 3))
    for k in ("error/MAE(X, F(G(X)))", "error/MAE(Y, G(F(Y)))",
              "error/MAE(X, F(X))", "error/MAE(Y, G(Y))"):
        assert k in r


def test_checkpoint_roundtrip(tmp_path, ctx):
    torch.manual_seed(0)
    args = make_args(tmp_path)
    gan = CycleGAN(args, ctx)
    x = torch.rand(1, 16, 16, 3)
    y = torch.rand(1, 16, 16, 3)
    gan.train_step(x, y)
    gan.save_checkpoint()
    ref = {n: g.flat_param.clone() for n, g in gan.groups.items()}
    ref_m = gan.optimizers["G"].m.clone()

    gan2 = CycleGAN(args, ctx)
    assert gan2.load_checkpoint()
    for n, g in gan2.groups.items():
        assert torch.equal(g.flat_param, ref[n]), n
    assert torch.equal(gan2.optimizers["G"].m, ref_m)
    assert gan2.optimizers["G"].t == 1
    # training continues bit-identically after resume
    r1 = gan.train_step(x, y)
    r2 = gan2.train_step(x, y)
    assert torch.equal(gan.groups["G"].flat_param, gan2.groups["G"].flat_param)


def test_no_checkpoint_returns_false(tmp_path, ctx):
    args = make_args(tmp_path)
    gan = CycleGAN(args, ctx)
    assert gan.load_checkpoint() is False


def test_loss_scaling_by_global_batch(tmp_path, ctx):
    """sum/global_batch semantics (main.py:172-174): with global_batch=4 and
    local batch 2, local loss must be half the batch-2 mean."""
    torch.manual_seed(0)
    args = make_args(tmp_path)
    args.global_batch_size = 4
    gan = CycleGAN(args, ctx)
    x = torch.rand(2, 16, 16, 3)
    y = torch.rand(2, 16, 16, 3)
    r = gan.test_step(x, y)
    gan.global_batch_size = 2
    r2 = gan.test_step(x, y)
    assert abs(r["loss_G/total"].item() * 2 - r2["loss_G/total"].item()) < 1e-5


def test_train_step_grads_match_reference_structure(tmp_path, ctx):
    """The shared-discriminator-node / batched-F train_step must produce
    the same gradients as the reference's explicit-recompute structure
    (/root/reference/main.py:207-262: separate F calls, re-discriminated
    detached fakes)."""
    torch.manual_seed(3)
    args = make_args(tmp_path)
    gan = CycleGAN(args, ctx)
    x = torch.rand(2, 32, 32, 3) * 2 - 1
    y = torch.rand(2, 32, 32, 3) * 2 - 1

    # --- reference-structure gradients ---
    fake_y = gan.G(x)
    same_y = gan.G(y)
    fake_x = gan.F(y)
    same_x = gan.F(x)
    G_total = (gan.generator_loss(gan.Y(fake_y))
               + gan.cycle_loss(y, gan.G(fake_x))
               + gan.identity_loss(y, same_y))
    F_total = (gan.generator_loss(gan.X(fake_x))
               + gan.cycle_loss(x, gan.F(fake_y))
               + gan.identity_loss(x, same_x))
    dx = gan.X(torch.cat([x, fake_x.detach()]))
    dy_ = gan.Y(torch.cat([y, fake_y.detach()]))
    X_loss = gan.discriminator_loss(dx[:2], dx[2:])
    Y_loss = gan.discriminator_loss(dy_[:2], dy_[2:])
    want = {}
    for name, loss in (("G", G_total), ("F", F_total),
                       ("X", X_loss), ("Y", Y_loss)):
        gs = torch.autograd.grad(loss, gan.groups[name].params,
                                 retain_graph=True)
        want[name] = torch.cat([g.reshape(-1) for g in gs])

    # --- production train_step (optimizer step leaves grads in place) ---
    gan.train_step(x, y)
    for name in ("G", "F", "X", "Y"):
        got = gan.groups[name].flat_grad
        assert torch.allclose(got, want[name], rtol=1e-5, atol=1e-7), name
