"""Op layer: every compute op of the framework, HIP-kernel-backed on GPU.

Maps the reference's implicit native components
to explicit MI355X ops:

- K1-K6  convolutions         -> ops.conv.conv2d / conv_transpose2d
- K7     InstanceNorm         -> ops.norm.instance_norm (fused act/add)
- K8     ReflectionPad        -> ops.pad.reflection_pad2d (or folded into conv)
- K9-K12 activations/add      -> fused epilogues in conv/norm
- K13    loss elementwise     -> ops.losses.MAE/MSE/MSE_const
- K14    Adam                 -> ops.adam.FusedAdam
"""

from . import backend  # noqa: F401
from .conv import conv2d, conv_transpose2d, same_pads, set_fp8_mode, fp8_mode  # noqa: F401
from .norm import instance_norm  # noqa: F401
from .pad import reflection_pad2d  # noqa: F401
from .losses import MAE, MSE, MSE_const, BCE  # noqa: F401
