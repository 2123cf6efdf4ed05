"""Measured numbers for BASELINE configs 4 (stretch) and 5 (stateful video)."""
import sys, os, time, torch, json
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from glom_pytorch_amd import Glom
from glom_pytorch_amd.parallel.trainer import DenoisingTrainer

dev = "cuda"
torch.manual_seed(0)

# --- config 4: dim=1024 L=12 512/16 iters=24, training step ---
m = Glom(dim=1024, levels=12, image_size=512, patch_size=16).to(dev, torch.bfloat16)
tr = DenoisingTrainer(m)
B = 8
img = torch.randn(B, 3, 512, 512, device=dev, dtype=torch.bfloat16)
tr.step(img, iters=24)                      # warmup
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(3): tr.step(img, iters=24)
torch.cuda.synchronize(); dt = (time.perf_counter() - t0) / 3
print(json.dumps({"config": "stretch dim=1024 L=12 512/16 iters=24 train",
                  "batch": B, "ms_per_step": dt*1e3, "images_sec": B/dt,
                  "peak_mem_GB": torch.cuda.max_memory_allocated()/2**30}))
del m, tr, img
torch.cuda.empty_cache(); torch.cuda.reset_peak_memory_stats()

# --- config 5: stateful 3-frame video under hipGraph replay ---
m = Glom(dim=512, levels=6, image_size=224, patch_size=14).to(dev, torch.bfloat16)
frames = [torch.randn(8, 3, 224, 224, device=dev, dtype=torch.bfloat16) for _ in range(3)]
def seq():
    with torch.no_grad():
        l1 = m(frames[0], iters=12)
        l2 = m(frames[1], iters=10, levels=l1)
        l3 = m(frames[2], iters=6, levels=l2)
    return l3
for graphs in (False, True):
    if graphs: m.enable_graphs()
    seq(); seq()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(10): seq()
    torch.cuda.synchronize(); dt = (time.perf_counter() - t0) / 10
    print(json.dumps({"config": "video 3-frame iters=12/10/6 B=8",
                      "hipgraphs": graphs, "ms_per_seq": dt*1e3,
                      "frames_sec": 3*8/dt}))
