"""PatchGAN discriminator (reference get_discriminator,
/root/reference/cyclegan/model.py:172-213).

conv4x4 s2 'same' 64 (bias) -> LeakyReLU(0.2)
-> conv4x4 s2 'same' 128 no-bias -> IN -> LeakyReLU(0.2)
-> conv4x4 s2 'same' 256 no-bias -> IN -> LeakyReLU(0.2)
-> conv4x4 s1 'same' 512 no-bias -> IN -> LeakyReLU(0.2)
-> conv4x4 s1 'same' 1 (bias)          # (B, 32, 32, 1) patch map @ 256^2

2,765,633 parameters (verified by tests/test_models.py).
"""

from __future__ import annotations

import torch.nn as nn

from .layers import ConvNHWC, InstanceNormNHWC


class Discriminator(nn.Module):
    def __init__(self, in_channels: int = 3, filters: int = 64,
                 num_downsampling: int = 3):
        super().__init__()
        f = filters
        self.stem = ConvNHWC(in_channels, f, 4, 2, padding="same",
                             bias=True, act="lrelu", slope=0.2)
        mids = []
        for i in range(num_downsampling):
            stride = 2 if i < num_downsampling - 1 else 1
            mids += [ConvNHWC(f, f * 2, 4, stride, padding="same"),
                     InstanceNormNHWC(f * 2, act="lrelu", slope=0.2)]
            f *= 2
        self.mids = nn.ModuleList(mids)
        self.head = ConvNHWC(f, 1, 4, 1, padding="same", bias=True)

    def forward(self, x):
        h = self.stem(x)
        for m in self.mids:
            h = m(h)
        return self.head(h)
