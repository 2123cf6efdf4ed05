"""Recipe (d): stateful multi-frame video — reference README.md:92-112 —
feeding each frame's levels into the next, under hipGraph replay on GPU.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from glom_pytorch_amd import Glom

use_gpu = torch.cuda.is_available()
dev = "cuda" if use_gpu else "cpu"
dtype = torch.bfloat16 if use_gpu else torch.float32
size, patch = (224, 14) if use_gpu else (32, 8)

model = Glom(dim=512 if use_gpu else 64, levels=6 if use_gpu else 3,
             image_size=size, patch_size=patch).to(dev, dtype)
if use_gpu:
    model.enable_graphs()   # capture once per (shape, iters) key

frames = [torch.randn(1, 3, size, size, device=dev, dtype=dtype)
          for _ in range(3)]
with torch.no_grad():
    l1 = model(frames[0], iters=12)
    l2 = model(frames[1], iters=10, levels=l1)   # replayed as one graph
    l3 = model(frames[2], iters=6, levels=l2)
print("final levels:", tuple(l3.shape))
