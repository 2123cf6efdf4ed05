import sys, os, json, torch, math
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from glom_pytorch_amd import Glom
from glom_pytorch_amd.parallel.trainer import DenoisingTrainer

def structured_batch(B, dev, dtype, size=224):
    y, x = torch.meshgrid(torch.linspace(0, 1, size, device=dev),
                          torch.linspace(0, 1, size, device=dev), indexing="ij")
    img = torch.zeros(B, 3, size, size, device=dev)
    for _ in range(4):
        fx, fy = torch.randint(1, 5, (2,), device=dev)
        ph = torch.rand(B, 3, 1, 1, device=dev) * 2 * math.pi
        amp = torch.randn(B, 3, 1, 1, device=dev) * 0.5
        img += amp * torch.sin(2 * math.pi * (fx * x + fy * y) + ph)
    return img.to(dtype)

def run(name, dtype, eager, steps=150):
    torch.manual_seed(0)
    m = Glom(dim=512, levels=6, image_size=224, patch_size=14).to("cuda", dtype)
    if eager: m.force_eager = True
    tr = DenoisingTrainer(m, lr=3e-4, noise_std=0.3)
    losses = []
    for _ in range(steps):
        losses.append(tr.step(structured_batch(16, "cuda", dtype), iters=12))
    print(json.dumps({"run": name, "first10": sum(losses[:10])/10,
                      "l50": losses[49], "l100": losses[99],
                      "last": losses[-1]}))

run("native-bf16", torch.bfloat16, False)
run("eager-bf16", torch.bfloat16, True)
run("eager-fp32", torch.float32, True)
