from .generator import Generator  # noqa: F401
from .discriminator import Discriminator  # noqa: F401
from .layers import ConvNHWC, ConvTransposeNHWC, InstanceNormNHWC, ResBlock  # noqa: F401
