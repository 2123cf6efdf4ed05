"""Recipe (a): plain inference — reference README.md:24-33.

Runs the CDNA4 engine when a GPU is present (bf16), eager PyTorch on CPU.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from glom_pytorch_amd import Glom

use_gpu = torch.cuda.is_available()
dev = "cuda" if use_gpu else "cpu"
dtype = torch.bfloat16 if use_gpu else torch.float32

model = Glom(dim=512, levels=6, image_size=224, patch_size=14).to(dev, dtype)

img = torch.randn(1, 3, 224, 224, device=dev, dtype=dtype)
with torch.no_grad():
    levels = model(img, iters=12)       # (1, 256, 6, 512)
print("levels:", tuple(levels.shape), levels.dtype)
