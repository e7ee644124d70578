"""ResNet generator (reference get_generator,
/root/reference/cyclegan/model.py:129-169).

Architecture at defaults (filters=64, 2 down, 9 resblocks, 2 up):
reflect-pad(3) -> conv7x7(64) valid no-bias -> IN -> ReLU
-> [conv3x3 s2 'same' no-bias -> IN -> ReLU] x2 (128, 256)
-> ResBlock(256) x9
-> [convT3x3 s2 'same' no-bias -> IN -> ReLU] x2 (128, 64)
-> reflect-pad(3) -> conv7x7(3) valid bias glorot-init -> tanh

11,383,427 parameters (verified by tests/test_models.py).
"""

from __future__ import annotations

import torch.nn as nn

from .layers import ConvNHWC, ConvTransposeNHWC, InstanceNormNHWC, ResBlock


class Generator(nn.Module):
    def __init__(self, in_channels: int = 3, filters: int = 64,
                 num_downsampling_blocks: int = 2, num_residual_blocks: int = 9,
                 num_upsample_blocks: int = 2):
        super().__init__()
        f = filters
        self.stem_conv = ConvNHWC(in_channels, f, 7, 1,
                                  padding=(3, 3, 3, 3), pad_mode="reflect")
        self.stem_norm = InstanceNormNHWC(f, act="relu")

        downs = []
        for _ in range(num_downsampling_blocks):
            downs += [ConvNHWC(f, f * 2, 3, 2, padding="same"),
                      InstanceNormNHWC(f * 2, act="relu")]
            f *= 2
        self.downs = nn.ModuleList(downs)

        self.blocks = nn.ModuleList(ResBlock(f) for _ in range(num_residual_blocks))

        ups = []
        for _ in range(num_upsample_blocks):
            ups += [ConvTransposeNHWC(f, f // 2, 3, 2),
                    InstanceNormNHWC(f // 2, act="relu")]
            f //= 2
        self.ups = nn.ModuleList(ups)

        self.head = ConvNHWC(f, in_channels, 7, 1, padding=(3, 3, 3, 3),
                             pad_mode="reflect", bias=True, act="tanh",
                             init="glorot")

    def forward(self, x):
        h = self.stem_norm(self.stem_conv(x))
        for m in self.downs:
            h = m(h)
        for b in self.blocks:
            h = b(h)
        for m in self.ups:
            h = m(h)
        return self.head(h)
