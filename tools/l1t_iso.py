"""Isolated L1t fwd GEMM (B=1M, 376->64) for PMC attribution."""
import os
import sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from dppo_amd.ops import require_hip_ext
ext = require_hip_ext()
B, D, H = 1048576, 376, 64
X = torch.randn(B, D, device="cuda") * 0.5
W1t = torch.randn(D, H, device="cuda") * 0.05
b1 = torch.zeros(H, device="cuda")
h1 = torch.empty(B, H, device="cuda")
for _ in range(12):
    ext.gemm_fwd(X, W1t, b1, 1, 0, h1, h1, h1, 0, 0, 0)
torch.cuda.synchronize()
