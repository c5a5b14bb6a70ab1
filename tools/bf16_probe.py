"""Minimal mm256 launcher for rocprofv3 PMC runs."""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
import torch
from dppo_amd.ops import require_hip_ext

ext = require_hip_ext()
e = torch.empty(0, device="cuda")
eb = torch.empty(0, device="cuda", dtype=torch.bfloat16)
M, N, K = 8192, 8192, 8192
A = (torch.randn(M, K, device="cuda") * 0.3).bfloat16()
B = (torch.randn(N, K, device="cuda") * 0.3).bfloat16()
C = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
for _ in range(int(sys.argv[1]) if len(sys.argv) > 1 else 5):
    ext.bf16_mm256(A, B, C, 0, e, eb, e, 0, eb, 0, e, 0)
torch.cuda.synchronize()
print("done")
