import argparse

import torch

from cyclegan_amd.data import Pipeline
from cyclegan_amd.data.pipeline import (ShuffleBuffer, preprocess_test,
                                        preprocess_train, synthetic_images)


def make_args(gb=2, b=2, n_train=10, n_test=4):
    a = argparse.Namespace()
    a.global_batch_size = gb
    a.batch_size = b
    a.num_train_samples = n_train
    a.num_test_samples = n_test
    a.data_dir = None
    a.seed = 1234
    return a


def test_pipeline_steps_and_shapes(local_ctx):
    p = Pipeline(make_args(), local_ctx, image_size=64)
    assert p.train_steps == 5 and p.test_steps == 2
    batches = list(p.train_epoch(0))
    assert len(batches) == 5
    x, y = batches[0]
    assert x.shape == (2, 64, 64, 3) and y.shape == (2, 64, 64, 3)
    assert x.min() >= -1.0 and x.max() <= 1.0


def test_pipeline_short_final_batch(local_ctx):
    p = Pipeline(make_args(n_train=5), local_ctx, image_size=64)
    assert p.train_steps == 3
    batches = list(p.train_epoch(0))
    assert batches[-1][0].shape[0] == 1  # 5 = 2+2+1


def test_epoch_shuffling_differs(local_ctx):
    p = Pipeline(make_args(), local_ctx, image_size=64)
    e0 = torch.cat([b[0] for b in p.train_epoch(0)])
    e1 = torch.cat([b[0] for b in p.train_epoch(1)])
    assert not torch.equal(e0, e1)
    # deterministic given the epoch
    e0b = torch.cat([b[0] for b in p.train_epoch(0)])
    assert torch.equal(e0, e0b)


def test_rank_slicing(local_ctx):
    p0 = Pipeline(make_args(gb=4, b=2), local_ctx, image_size=64)
    b0 = next(iter(p0.train_epoch(3)))
    local_ctx.rank = 1
    b1 = next(iter(p0.train_epoch(3)))
    assert b0[0].shape[0] == 2 and b1[0].shape[0] == 2
    assert not torch.equal(b0[0], b1[0])
    local_ctx.rank = 0


def test_shuffle_buffer_is_permutation():
    g = torch.Generator().manual_seed(0)
    items = list(range(100))
    out = list(ShuffleBuffer(items, 16, g))
    assert sorted(out) == items
    assert out != items


def test_preprocess_shapes():
    img = synthetic_images(1, 0, hw=(100, 120))[0]
    g = torch.Generator().manual_seed(0)
    t = preprocess_train(img, g, (286, 286), (256, 256))
    assert t.shape == (256, 256, 3)
    assert -1.0 <= t.min() and t.max() <= 1.0
    v = preprocess_test(img, (256, 256))
    assert v.shape == (256, 256, 3)


def test_plot_pairs(local_ctx):
    p = Pipeline(make_args(), local_ctx, image_size=64)
    pairs = list(p.plot_pairs())
    assert len(pairs) == 4  # min(5, num_test)
    assert pairs[0][0].shape == (1, 64, 64, 3)


def test_device_prefetcher_cpu_passthrough(local_ctx):
    import torch
    from cyclegan_amd.data import DevicePrefetcher
    p = Pipeline(make_args(), local_ctx, image_size=64)
    it = DevicePrefetcher(p.train_epoch(0), torch.device("cpu"), torch.float32)
    batches = list(it)
    assert len(batches) == p.train_steps
    ref = list(p.train_epoch(0))
    assert torch.equal(batches[0][0], ref[0][0])


def test_folder_images_loading(tmp_path):
    """--data_dir image-folder loading (PIL decode, RGB, uint8 HWC)."""
    import numpy as np
    import PIL.Image
    from cyclegan_amd.data.pipeline import folder_images
    d = tmp_path / "trainA"
    d.mkdir()
    for i, size in enumerate([(10, 12), (7, 7)]):
        a = (np.random.rand(size[0], size[1], 3) * 255).astype("uint8")
        PIL.Image.fromarray(a).save(str(d / f"im{i}.png"))
    (d / "notes.txt").write_text("ignored")
    imgs = folder_images(str(d))
    assert len(imgs) == 2
    assert imgs[0].shape == (10, 12, 3) and imgs[0].dtype == torch.uint8
    assert imgs[1].shape == (7, 7, 3)
