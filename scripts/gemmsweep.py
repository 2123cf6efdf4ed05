import torch, sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from glom_pytorch_amd.ops import _load_extension
ext = _load_extension()
NT, NN, TN = 0, 1, 2
NONE, GRAD, PAIR = 0, 1, 2
def tf(M,N,K,P,ms): return 2*M*N*K*P/(ms*1e-3)/1e12
cases = [
    # the model's NT shapes, epilogue ablation
    ("up   K512 N2048 pair", 16384, 2048, 512, 6, NT, PAIR),
    ("up   K512 N2048 none", 16384, 2048, 512, 6, NT, NONE),
    ("dH   K512 N2048 grad", 16384, 2048, 512, 5, NT, GRAD),
    ("down K2048 N512 none", 16384, 512, 2048, 6, NT, NONE),
    ("sq   K2048 N2048 none", 16384, 2048, 2048, 6, NT, NONE),
    ("sq   K512  N512  none", 16384, 512, 512, 6, NT, NONE),
    ("dW1  TN m4xd K16384", 2048, 512, 16384, 6, TN, NONE),
    ("attnAV NN 256x512x256", 256, 512, 256, 384, NN, NONE),
]
for name, M,N,K,P,lay,epi in cases:
    ms = ext.bench_gemm(M,N,K,lay,P,epi,20)
    print(f"{name:26s} {ms*1e3:8.1f} us  {tf(M,N,K,P,ms):6.1f} TF")
ms = ext.bench_gemm(16384,2048,512,NT,6,10,20)
print(f"{'up pair NOGELU dbg':26s} {ms*1e3:8.1f} us  {tf(16384,2048,512,6,ms):6.1f} TF")
ms = ext.bench_gemm(16384,2048,512,NT,6,12,20)
print(f"{'up pair POLY dbg':26s} {ms*1e3:8.1f} us  {tf(16384,2048,512,6,ms):6.1f} TF")
