"""cyclegan_amd — an MI355X-native CycleGAN training framework.

A from-scratch re-design of the capabilities of bryanlimy/tf2-cyclegan
(reference layout: main.py + cyclegan/{model,utils}.py) for AMD Instinct
MI355X (gfx950, CDNA4):

- PyTorch-ROCm is the framework layer; every hot op (conv fwd/dgrad/wgrad,
  InstanceNorm, reflection pad, activations, losses, Adam) is a hand-written
  HIP kernel for gfx950 exposed through torch.autograd.Function
  (``cyclegan_amd.ops``).
- Tensors are NHWC (channels-last): channels innermost is the natural
  implicit-GEMM layout for MFMA matrix cores.
- Data parallelism is one process per GPU with RCCL over xGMI
  (``cyclegan_amd.parallel``), replicating tf.distribute.MirroredStrategy
  semantics (loss pre-scaled by 1/global_batch, SUM all-reduce) from
  /root/reference/main.py:172-174,249-260.
"""

__version__ = "0.1.0"

from . import ops  # noqa: F401
