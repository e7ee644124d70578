"""Cycle plotting (reference utils.plot_cycle, utils.py:112-145)."""

from __future__ import annotations

import numpy as np
import torch

from .summary import Summary, append_dict


def plot_cycle(plot_pairs, gan, summary: Summary, epoch: int):
    samples = {}
    for x, y in plot_pairs:
        fake_x, fake_y, cycle_x, cycle_y = gan.cycle_step(x, y, training=False)
        append_dict(samples, {"x": x, "y": y, "fake_x": fake_x,
                              "fake_y": fake_y, "cycle_x": cycle_x,
                              "cycle_y": cycle_y})
    for key, images in samples.items():
        arr = torch.cat([t.detach().float().cpu() for t in images], dim=0).numpy()
        samples[key] = ((arr + 1) * 127.5).clip(0, 255).astype(np.uint8)

    summary.image_cycle(tag="X_cycle",
                        images=[samples["x"], samples["fake_y"], samples["cycle_x"]],
                        labels=["X", "G(X)", "F(G(X))"], step=epoch,
                        training=False)
    summary.image_cycle(tag="Y_cycle",
                        images=[samples["y"], samples["fake_x"], samples["cycle_y"]],
                        labels=["Y", "F(Y)", "G(F(Y))"], step=epoch,
                        training=False)
