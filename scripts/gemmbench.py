import torch, time, sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from glom_pytorch_amd.ops import _load_extension
ext = _load_extension()
torch.manual_seed(0)
B,N,L,d = 64,256,6,512; m4 = 4*d
bf = torch.bfloat16
dev = 'cuda'
tokens = torch.randn(B,N,d,device=dev,dtype=bf)
levels = torch.randn(B,N,L,d,device=dev,dtype=bf)
w1 = torch.randn(L*m4,d,device=dev,dtype=bf)*0.02; b1 = torch.zeros(L*m4,device=dev,dtype=bf)
w2 = torch.randn(L*d,m4,device=dev,dtype=bf)*0.02; b2 = torch.zeros(L*d,device=dev,dtype=bf)

def t(fn, n=20):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n

dt = t(lambda: ext.grouped_ff_fwd(tokens, levels, None, w1, b1, w2, b2, 0))
fl = 2*B*N*d*m4*L*2
print(f"ff fwd (up+down, G={L}): {dt*1e3:.3f} ms  {fl/dt/1e12:.1f} TF")

x = torch.randn(B*N, d, device=dev, dtype=bf)
wg = w1[:m4].contiguous()
dt2 = t(lambda: x @ wg.t())
print(f"rocBLAS up single group (16384x2048x512): {dt2*1e3:.3f} ms {2*B*N*d*m4/dt2/1e12:.1f} TF")
xw = torch.randn(L, B*N, d, device=dev, dtype=bf)
wgt = w1.view(L, m4, d).transpose(1,2).contiguous()
dt3 = t(lambda: torch.bmm(xw, wgt))
print(f"rocBLAS bmm grouped up: {dt3*1e3:.3f} ms {fl/2/dt3/1e12:.1f} TF")
h = torch.randn(B*N, m4, device=dev, dtype=bf)
w2g = w2[:d].contiguous()
dt4 = t(lambda: h @ w2g.t())
print(f"rocBLAS down single group (16384x512x2048): {dt4*1e3:.3f} ms {2*B*N*d*m4/dt4/1e12:.1f} TF")
