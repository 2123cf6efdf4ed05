"""Measure gemm_nn_fast at its real call shapes (headline dq, stretch AV)
to decide whether a pipelined rewrite pays."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from glom_pytorch_amd.ops import _load_extension
ext = _load_extension()
LAYOUT_NN, LAYOUT_NT = 2, 0

CASES = [
    # (label, M, N, K, nproblems)
    ("dq headline  (B=64,L=6,N=256,d=512)", 256, 512, 256, 384),
    ("AV stretch   (B=16,L=12,N=1024,d=1024)", 1024, 1024, 1024, 192),
    ("AV stretch B=8", 1024, 1024, 1024, 96),
    ("square 2048 (kernel ceiling probe)", 2048, 2048, 2048, 8),
]
for label, M, N, K, np_ in CASES:
    ms = ext.bench_gemm(M, N, K, LAYOUT_NN, np_, 0, 30)
    tf = 2.0 * M * N * K * np_ / (ms * 1e-3) / 1e12
    print(f"NN {label}: {ms:.3f} ms  {tf:.0f} TF")
for label, M, N, K, np_ in CASES[:2]:
    ms = ext.bench_gemm(M, N, K, LAYOUT_NT, np_, 0, 30)
    tf = 2.0 * M * N * K * np_ / (ms * 1e-3) / 1e12
    print(f"NT {label}: {ms:.3f} ms  {tf:.0f} TF")
