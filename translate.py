"""Inference CLI: translate images through a trained generator.

    python translate.py --checkpoint runs/checkpoints/checkpoint.pt \
        --direction x2y --input_dir testA/ --output_dir out/

Loads G (x->y) or F (y->x) from a framework checkpoint and writes
translated PNGs. Accepts an image folder or, with --synthetic N, random
inputs (useful without data). Runs the same HIP kernel path as training.
"""

from __future__ import annotations

import argparse
import os

import numpy as np
import torch

from cyclegan_amd.models import Generator
from cyclegan_amd.data.pipeline import folder_images, preprocess_test, synthetic_images


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--checkpoint", required=True)
    ap.add_argument("--direction", default="x2y", choices=["x2y", "y2x"])
    ap.add_argument("--input_dir", default=None)
    ap.add_argument("--synthetic", type=int, default=0)
    ap.add_argument("--output_dir", default="translated")
    ap.add_argument("--image_size", type=int, default=256)
    ap.add_argument("--num_residual_blocks", type=int, default=9)
    ap.add_argument("--batch_size", type=int, default=8)
    args = ap.parse_args()

    device = torch.device("cuda", 0) if torch.cuda.is_available() else torch.device("cpu")
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32

    gen = Generator(num_residual_blocks=args.num_residual_blocks).to(device)
    state = torch.load(args.checkpoint, map_location=device, weights_only=True)
    gen.load_state_dict(state["G" if args.direction == "x2y" else "F"])
    gen.eval()

    if args.input_dir:
        raw = folder_images(args.input_dir)
    elif args.synthetic:
        raw = synthetic_images(args.synthetic, seed=1234,
                               hw=(args.image_size, args.image_size))
    else:
        raise SystemExit("need --input_dir or --synthetic N")

    os.makedirs(args.output_dir, exist_ok=True)
    import PIL.Image
    size = (args.image_size, args.image_size)
    with torch.no_grad():
        for i0 in range(0, len(raw), args.batch_size):
            batch = torch.stack([preprocess_test(im, size)
                                 for im in raw[i0:i0 + args.batch_size]])
            out = gen(batch.to(device, dtype))
            imgs = ((out.float().cpu() + 1) * 127.5).clamp(0, 255).numpy().astype("uint8")
            for j in range(imgs.shape[0]):
                PIL.Image.fromarray(imgs[j]).save(
                    os.path.join(args.output_dir, f"{i0 + j:05d}.png"))
    print(f"wrote {len(raw)} images to {args.output_dir}")


if __name__ == "__main__":
    main()
