"""Input pipeline (reference get_datasets, /root/reference/main.py:18-83).

Semantics replicated:
- two unpaired domains; epoch length = ceil(min(|A|,|B|) / global_batch)
  (horse2zebra: 1067 train / 120 test pairs-equivalent);
- train preprocessing: random flip-LR -> bilinear resize 286 -> random crop
  256 -> scale to [-1,1]; applied ONCE then cached (the reference calls
  .map().cache(), freezing each sample's augmentation for the run), then a
  256-slot shuffle buffer per epoch;
- test preprocessing: resize 256 -> normalize, cached;
- each rank takes its per-replica slice of the global batch (the
  experimental_distribute_dataset split, main.py:80-81); a short final
  global batch may leave some ranks with 0 samples — handled upstream;
- plot set: first 5 test samples of each domain, batch 1 (rank 0 only).

Sources: horse2zebra is a tfds dataset and this environment has no network,
so the default source is a deterministic synthetic clone with the same
split sizes and image shapes; ``--data_dir`` can point at a directory with
trainA/trainB/testA/testB image folders (PIL-decodable) for real data.
"""

from __future__ import annotations

import math
import os
from typing import List, Optional, Tuple

import torch
import torch.nn.functional as F

SPLITS = {"trainA": 1067, "trainB": 1334, "testA": 120, "testB": 140}


def normalize_image(img: torch.Tensor) -> torch.Tensor:
    return img.float() / 127.5 - 1.0


def resize_bilinear(img: torch.Tensor, size: Tuple[int, int]) -> torch.Tensor:
    """img [H,W,C] float -> [h,w,C]; matches tf.image.resize bilinear
    (half-pixel centers, no antialias)."""
    x = img.permute(2, 0, 1).unsqueeze(0)
    y = F.interpolate(x, size=size, mode="bilinear", align_corners=False,
                      antialias=False)
    return y.squeeze(0).permute(1, 2, 0)


def preprocess_train(img_u8: torch.Tensor, gen: torch.Generator,
                     image_shape=(286, 286), crop=(256, 256)) -> torch.Tensor:
    img = img_u8.float()
    if torch.rand((), generator=gen).item() < 0.5:
        img = img.flip(1)  # left-right
    img = resize_bilinear(img, image_shape)
    ch, cw = crop
    max_y, max_x = img.shape[0] - ch, img.shape[1] - cw
    oy = int(torch.randint(0, max_y + 1, (), generator=gen))
    ox = int(torch.randint(0, max_x + 1, (), generator=gen))
    img = img[oy:oy + ch, ox:ox + cw, :]
    return img / 127.5 - 1.0


def preprocess_test(img_u8: torch.Tensor, size=(256, 256)) -> torch.Tensor:
    img = resize_bilinear(img_u8.float(), size)
    return img / 127.5 - 1.0


def synthetic_images(n: int, seed: int, hw: Tuple[int, int] = (256, 256)) -> List[torch.Tensor]:
    g = torch.Generator().manual_seed(seed)
    return [torch.randint(0, 256, (hw[0], hw[1], 3), generator=g,
                          dtype=torch.uint8) for _ in range(n)]


def folder_images(path: str) -> List[torch.Tensor]:
    import PIL.Image
    import numpy as np
    out = []
    for name in sorted(os.listdir(path)):
        if name.lower().endswith((".jpg", ".jpeg", ".png")):
            with PIL.Image.open(os.path.join(path, name)) as im:
                # np.asarray of a PIL image is read-only; copy so the
                # tensor owns writable memory (torch warns otherwise)
                out.append(torch.from_numpy(
                    np.array(im.convert("RGB"), dtype="uint8")))
    return out


class ShuffleBuffer:
    """tf.data-style fixed-size shuffle buffer (size 256 in the reference)."""

    def __init__(self, items: list, size: int, gen: torch.Generator):
        self.items, self.size, self.gen = items, size, gen

    def __iter__(self):
        buf = []
        for it in self.items:
            buf.append(it)
            if len(buf) >= self.size:
                j = int(torch.randint(0, len(buf), (), generator=self.gen))
                buf[j], buf[-1] = buf[-1], buf[j]
                yield buf.pop()
        while buf:
            j = int(torch.randint(0, len(buf), (), generator=self.gen))
            buf[j], buf[-1] = buf[-1], buf[j]
            yield buf.pop()


class Pipeline:
    def __init__(self, args, ctx, image_size: int = 256):
        self.ctx = ctx
        self.global_batch = args.global_batch_size
        self.per_replica = args.batch_size
        self.image_size = image_size
        self.seed = getattr(args, "seed", 1234)
        data_dir = getattr(args, "data_dir", None)
        n_train = getattr(args, "num_train_samples", None)
        n_test = getattr(args, "num_test_samples", None)

        gen = torch.Generator().manual_seed(self.seed)
        if data_dir:
            raw = {s: folder_images(os.path.join(data_dir, s)) for s in SPLITS}
        else:
            raw = {s: synthetic_images(
                n if (n := {"trainA": n_train, "trainB": n_train,
                            "testA": n_test, "testB": n_test}[s]) else SPLITS[s],
                seed=self.seed + i, hw=(image_size, image_size))
                for i, s in enumerate(SPLITS)}

        self.num_train = min(len(raw["trainA"]), len(raw["trainB"]))
        self.num_test = min(len(raw["testA"]), len(raw["testB"]))
        self.train_steps = math.ceil(self.num_train / self.global_batch)
        self.test_steps = math.ceil(self.num_test / self.global_batch)

        ih = (286 * image_size) // 256  # scale the 286/256 ratio with size
        ishape, cshape = (ih, ih), (image_size, image_size)
        # .map().cache(): augmentation frozen per-sample for the run
        self.trainA = [preprocess_train(im, gen, ishape, cshape)
                       for im in raw["trainA"][: self.num_train]]
        self.trainB = [preprocess_train(im, gen, ishape, cshape)
                       for im in raw["trainB"][: self.num_train]]
        self.testA = [preprocess_test(im, cshape) for im in raw["testA"][: self.num_test]]
        self.testB = [preprocess_test(im, cshape) for im in raw["testB"][: self.num_test]]

    def _rank_slice(self, batch: List[torch.Tensor]) -> torch.Tensor:
        r, b = self.ctx.rank, self.per_replica
        part = batch[r * b:(r + 1) * b]
        if not part:
            s = self.image_size
            return torch.empty(0, s, s, 3)
        return torch.stack(part)

    def train_epoch(self, epoch: int):
        """Yields (x, y) local batches; same shuffle series on every rank."""
        g = torch.Generator().manual_seed(self.seed * 100003 + epoch)
        a = iter(ShuffleBuffer(self.trainA, 256, g))
        b = iter(ShuffleBuffer(self.trainB, 256, g))
        for _ in range(self.train_steps):
            xs, ys = [], []
            for _ in range(self.global_batch):
                xa, xb = next(a, None), next(b, None)
                if xa is None or xb is None:
                    break
                xs.append(xa)
                ys.append(xb)
            yield self._rank_slice(xs), self._rank_slice(ys)

    def test_epoch(self):
        for s in range(self.test_steps):
            xs = self.testA[s * self.global_batch:(s + 1) * self.global_batch]
            ys = self.testB[s * self.global_batch:(s + 1) * self.global_batch]
            yield self._rank_slice(xs), self._rank_slice(ys)

    def plot_pairs(self, n: int = 5):
        for i in range(min(n, self.num_test)):
            yield self.testA[i].unsqueeze(0), self.testB[i].unsqueeze(0)
