"""Recipe (c): denoising self-supervised training — reference
README.md:56-90 — with the optimizer/decoder/clipping the reference
leaves to the user. Launch one process per GPU via torchrun for DP:

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 examples/02_denoising_training.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from glom_pytorch_amd import Glom
from glom_pytorch_amd.parallel.trainer import DenoisingTrainer

use_gpu = torch.cuda.is_available()
distributed = int(os.environ.get("WORLD_SIZE", "1")) > 1
if distributed:
    from glom_pytorch_amd.parallel.failure import init_distributed
    init_distributed()
    torch.cuda.set_device(int(os.environ["LOCAL_RANK"]))

dev = "cuda" if use_gpu else "cpu"
dtype = torch.bfloat16 if use_gpu else torch.float32
size, steps, batch = (224, 20, 32) if use_gpu else (32, 3, 2)
patch = 14 if use_gpu else 8

model = Glom(dim=512 if use_gpu else 64, levels=6 if use_gpu else 3,
             image_size=size, patch_size=patch).to(dev, dtype)
trainer = DenoisingTrainer(model, lr=1e-4, noise_std=0.3,
                           distributed=distributed)

for step in range(steps):
    img = torch.randn(batch, 3, size, size, device=dev, dtype=dtype)
    loss = trainer.step(img, iters=2 * model.levels)
    if step % 5 == 0:
        print(f"step {step}: loss {loss:.4f}")

trainer.save_checkpoint("/tmp/glom_example_ckpt.pt")
print("checkpoint saved; state_dict interchanges with the reference package")
