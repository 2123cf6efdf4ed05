"""Training-stability soak: 200 denoising steps on the native engine; the
loss must fall materially below its initial value and stay finite."""
import sys, os, json, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from glom_pytorch_amd import Glom
from glom_pytorch_amd.parallel.trainer import DenoisingTrainer

torch.manual_seed(0)
m = Glom(dim=512, levels=6, image_size=224, patch_size=14).to("cuda", torch.bfloat16)
tr = DenoisingTrainer(m, lr=1e-4, noise_std=0.3)
losses = []
for step in range(200):
    img = torch.randn(32, 3, 224, 224, device="cuda", dtype=torch.bfloat16)
    losses.append(tr.step(img, iters=12))
first, last = sum(losses[:10]) / 10, sum(losses[-10:]) / 10
print(json.dumps({"first10": first, "last10": last,
                  "min": min(losses), "finite": all(l == l for l in losses)}))
assert last < 0.8 * first, (first, last)
print("SOAK OK")
