"""Summary helper replicating the reference's two-writer layout and tag
names (/root/reference/cyclegan/utils.py:14-98): train writer at
``output_dir``, test writer at ``output_dir/test``; scalar / image /
figure / image_cycle, with image_cycle emitting one 1x3 panel per sample
under ``{tag}/sample_#NNN``."""

from __future__ import annotations

import io
import typing as t

import numpy as np

from .tb_writer import EventWriter


class Summary:
    def __init__(self, output_dir: str):
        self.dpi = 120
        # reference sets plt.style.use('seaborn-deep') at import
        # (utils.py:19); modern matplotlib renamed the style
        try:
            import matplotlib.pyplot as plt
            for name in ("seaborn-deep", "seaborn-v0_8-deep"):
                if name in plt.style.available:
                    plt.style.use(name)
                    break
        except Exception:
            pass
        import os
        self.writers = [EventWriter(output_dir),
                        EventWriter(os.path.join(output_dir, "test"))]

    def get_writer(self, training: bool):
        return self.writers[0 if training else 1]

    def close(self):
        for w in self.writers:
            w.close()

    def scalar(self, tag, value, step: int = 0, training: bool = False):
        self.get_writer(training).scalar(tag, float(value), step)

    def image(self, tag, values, step: int = 0, training: bool = False):
        """values: uint8 array (N,H,W,C); each sample logged as PNG."""
        import PIL.Image
        arr = np.asarray(values)
        for i in range(arr.shape[0]):
            buf = io.BytesIO()
            PIL.Image.fromarray(arr[i]).save(buf, format="png")
            self.get_writer(training).image_png(
                f"{tag}/{i}" if arr.shape[0] > 1 else tag,
                buf.getvalue(), arr.shape[1], arr.shape[2], step)

    def figure(self, tag, figure, step: int = 0, training: bool = False,
               close: bool = True):
        import matplotlib.pyplot as plt
        buf = io.BytesIO()
        figure.savefig(buf, dpi=self.dpi, format="png", bbox_inches="tight")
        buf.seek(0)
        import PIL.Image
        im = PIL.Image.open(buf)
        w, h = im.size
        self.get_writer(training).image_png(tag, buf.getvalue(), h, w, step)
        if close:
            plt.close(figure)

    def image_cycle(self, tag: str, images: t.List[np.ndarray],
                    labels: t.List[str], step: int = 0,
                    training: bool = False):
        assert len(images) == len(labels) == 3
        import matplotlib.pyplot as plt
        for sample in range(len(images[0])):
            figure, axes = plt.subplots(nrows=1, ncols=3, figsize=(9, 3.25),
                                        dpi=self.dpi)
            for k in range(3):
                axes[k].imshow(images[k][sample, ...], interpolation="none")
                axes[k].set_title(labels[k])
            plt.setp(axes, xticks=[], yticks=[])
            plt.tight_layout()
            figure.subplots_adjust(wspace=0.02, hspace=0.02)
            self.figure(tag=f"{tag}/sample_#{sample:03d}", figure=figure,
                        step=step, training=training, close=True)


def append_dict(dict1: dict, dict2: dict, replace: bool = False):
    """Append items in dict2 to dict1 (reference utils.py:101-109)."""
    for key, value in dict2.items():
        if replace:
            dict1[key] = value
        else:
            dict1.setdefault(key, []).append(value)
