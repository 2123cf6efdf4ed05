import sys, os
sys.path.insert(0, os.getcwd())
import torch
from glom_pytorch_amd import Glom
from glom_pytorch_amd.parallel.trainer import DenoisingTrainer

case = sys.argv[1]
torch.manual_seed(0)
m = Glom(dim=64, levels=3, image_size=32, patch_size=8).to("cuda", torch.bfloat16)
kw = dict(noise_std=0.5, decode_step=2)
if case == "nograph":
    tr = DenoisingTrainer(m, graph_step=False, **kw)
elif case == "no_overlap":
    tr = DenoisingTrainer(m, noise_std=0.5, decode_step=3)  # t==iters -> no tail
elif case == "graph":
    tr = DenoisingTrainer(m, **kw)
img = torch.randn(2, 3, 32, 32, device="cuda", dtype=torch.bfloat16)
l1 = tr.step(img, iters=3)
l2 = tr.step(img, iters=3)
torch.cuda.synchronize()
print(case, "OK", l1, l2)
