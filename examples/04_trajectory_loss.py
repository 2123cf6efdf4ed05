"""Recipe (b): attach losses anywhere in the iteration trajectory —
reference README.md:35-54. `return_all=True` yields (T+1, B, N, L, d)
including the initial state; gradients flow through the hand-written
HIP backward on GPU.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F

from glom_pytorch_amd import Glom

use_gpu = torch.cuda.is_available()
dev = "cuda" if use_gpu else "cpu"
dtype = torch.bfloat16 if use_gpu else torch.float32
size, patch = (224, 14) if use_gpu else (32, 8)

model = Glom(dim=512 if use_gpu else 64, levels=6 if use_gpu else 3,
             image_size=size, patch_size=patch).to(dev, dtype)

img = torch.randn(2, 3, size, size, device=dev, dtype=dtype)
# Tip: when your loss only reads trajectory times <= t, pass
# grad_iters=t — later iterations then run forward-only (their gradient
# contribution is exactly zero) and, inside DenoisingTrainer, overlap
# the backward. This loss reads the FINAL state, so the full graph is
# kept here.
all_levels = model(img, iters=2 * model.levels, return_all=True)
print("trajectory:", tuple(all_levels.shape))

# consistency loss between two mid-trajectory top levels
a = all_levels[model.levels, :, :, -1].float()
b = all_levels[-1, :, :, -1].float()
loss = F.mse_loss(a, b)
loss.backward()
print("loss:", loss.item(), "— grads flow:",
      model.bottom_up.net[1].weight.grad is not None)
