"""Short real-training run at the headline config; prints the loss
trajectory (convergence evidence; also used for bf16-vs-fp8 numerics
comparison).

    python tools/train_trajectory.py [--steps 300] [--fp8]
"""
import argparse, sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from cyclegan_amd.parallel import DistContext
from cyclegan_amd.trainer import CycleGAN

ap = argparse.ArgumentParser()
ap.add_argument("--steps", type=int, default=300)
ap.add_argument("--batch_size", type=int, default=4)
ap.add_argument("--image_size", type=int, default=256)
ap.add_argument("--fp8", action="store_true")
a = ap.parse_args()
a.output_dir = "/tmp/traj"
a.global_batch_size = a.batch_size
a.num_residual_blocks = 9
a.compute_dtype = torch.bfloat16
torch.manual_seed(1234)
ctx = DistContext(device=torch.device("cuda", 0))
gan = CycleGAN(a, ctx)
g = torch.Generator().manual_seed(99)
data = [((torch.rand(a.batch_size, a.image_size, a.image_size, 3, generator=g) * 2 - 1),
         (torch.rand(a.batch_size, a.image_size, a.image_size, 3, generator=g) * 2 - 1))
        for _ in range(8)]
data = [(x.to(ctx.device, torch.bfloat16), y.to(ctx.device, torch.bfloat16))
        for x, y in data]
for i in range(a.steps):
    r = gan.train_step(*data[i % len(data)])
    if i % 50 == 0 or i == a.steps - 1:
        torch.cuda.synchronize()
        print(f"step {i:4d}  G_total {r['loss_G/total'].item():.4f}  "
              f"G_cycle {r['loss_G/cycle'].item():.4f}  "
              f"X {r['loss_X/loss'].item():.4f}  Y {r['loss_Y/loss'].item():.4f}")
