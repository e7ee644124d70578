"""Head-conv-only microbench (for focused rocprofv3 runs)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from cyclegan_amd.ops import backend
e = backend.ext()
torch.manual_seed(0)
x = (torch.rand(4, 256, 256, 64, device="cuda", dtype=torch.bfloat16) - 0.5)
w = (torch.rand(8, 7, 7, 64, device="cuda", dtype=torch.bfloat16) - 0.5) * 0.1
for _ in range(20):
    y = e.conv2d_fwd(x, w, None, 1, 3, 3, 3, 3, True, 0, 0.2)
torch.cuda.synchronize()
print("ok", y.shape)
