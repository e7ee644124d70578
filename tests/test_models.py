import torch

from cyclegan_amd.models import Generator, Discriminator
from cyclegan_amd.ops import same_pads


def count_params(m):
    return sum(p.numel() for p in m.parameters())


def test_generator_param_count():
    # reference get_generator: 11,383,427 params
    assert count_params(Generator()) == 11_383_427


def test_discriminator_param_count():
    # reference get_discriminator: 2,765,633 params
    assert count_params(Discriminator()) == 2_765_633


def test_generator_shapes():
    g = Generator(num_residual_blocks=1)
    x = torch.randn(2, 64, 64, 3)
    y = g(x)
    assert y.shape == (2, 64, 64, 3)
    assert y.abs().max() <= 1.0  # tanh output


def test_discriminator_shapes():
    d = Discriminator()
    x = torch.randn(2, 256, 256, 3)
    y = d(x)
    assert y.shape == (2, 32, 32, 1)
    x = torch.randn(1, 64, 64, 3)
    assert d(x).shape == (1, 8, 8, 1)


def test_init_distribution():
    torch.manual_seed(0)
    g = Generator()
    w = g.stem_conv.weight
    assert abs(w.std().item() - 0.02) < 0.005
    assert abs(w.mean().item()) < 0.005
    # head uses glorot_uniform: bounded by limit
    import math
    wh = g.head.weight
    limit = math.sqrt(6.0 / (7 * 7 * 64 + 7 * 7 * 3))
    assert wh.abs().max().item() <= limit + 1e-6
    assert g.head.bias.abs().sum() == 0
    # IN gamma ~ N(0, 0.02), beta zeros
    gm = g.stem_norm.gamma
    assert abs(gm.std().item() - 0.02) < 0.01
    assert g.stem_norm.beta.abs().sum() == 0


def test_same_pads_tf_convention():
    # 3x3 s2 on 256: pad_total=1 -> (0,1)
    assert same_pads(256, 256, 3, 3, 2) == (0, 1, 0, 1)
    # 4x4 s2 on 256: pad_total=2 -> (1,1)
    assert same_pads(256, 256, 4, 4, 2) == (1, 1, 1, 1)
    # 4x4 s1: pad_total=3 -> (1,2)
    assert same_pads(32, 32, 4, 4, 1) == (1, 2, 1, 2)


def test_generator_arbitrary_size():
    g = Generator(num_residual_blocks=1)
    for s in (64, 96):
        assert g(torch.randn(1, s, s, 3)).shape == (1, s, s, 3)
