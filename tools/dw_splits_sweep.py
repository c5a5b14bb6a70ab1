"""Sweep DPPO_DW_SPLITS for the dW1 glds kernel at the flagship B.

dW1 (dz[B][64]^T @ states[B][376]) measured 3.4 TB/s with the default
1024-split grid; the narrow variant reaches 5.4 — suspect too few
workgroups walking too-long row ranges.  DPPO_DW_SPLITS is re-read per
call, so one process sweeps all values.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
import torch

from dppo_amd.ops import require_hip_ext

ext = require_hip_ext()
B = 16 * 1024 * 1024
D, H = 376, 64
dz = torch.randn(B, H, device="cuda") * 0.01
X = torch.randn(B, D, device="cuda") * 0.5
grad = torch.zeros(H * D + H, device="cuda")
gb = B * (D + H) * 4e-9

for splits in (512, 1024, 2048, 4096, 8192):
    os.environ["DPPO_DW_SPLITS"] = str(splits)
    for _ in range(2):
        ext.dw_mfma(dz, X, grad, 0, H * D, -1, -1, -1, 0)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(8):
        ext.dw_mfma(dz, X, grad, 0, H * D, -1, -1, -1, 0)
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / 8 * 1e3
    print(f"splits={splits:5d}: {ms:7.3f} ms  {gb / ms * 1e3:6.0f} GB/s-eff")
