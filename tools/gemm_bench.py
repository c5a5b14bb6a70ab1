"""Microbench: the update-path GEMM shapes in isolation (B=1M rows)."""
import sys, time
import torch
sys.path.insert(0, "/root/repo")
from dppo_amd.ops import require_hip_ext

ext = require_hip_ext()
B = 1048576
D, H, A = 376, 64, 17
P = 2 * A
X = torch.randn(B, D, device="cuda") * 0.5
W1 = torch.randn(H, D, device="cuda") * 0.05
b1 = torch.zeros(H, device="cuda")
W2 = torch.randn(H, H, device="cuda") * 0.05
Wh = torch.randn(P + 1, H, device="cuda") * 0.05
bh = torch.zeros(P + 1, device="cuda")
h1 = torch.empty(B, H, device="cuda")
h2 = torch.empty(B, H, device="cuda")
pdf = torch.empty(B, P, device="cuda")
v = torch.empty(B, device="cuda")
gh = torch.randn(B, P + 1, device="cuda") * 0.01
dz2 = torch.empty(B, H, device="cuda")
dummy = torch.zeros(1, device="cuda")

def l1(): ext.gemm_fwd(X, W1, b1, 1, 0, h1, h1, h1, 1)
def l2(): ext.gemm_fwd(h1, W2, b1, 1, 0, h2, h2, h2, 1)
def heads(): ext.gemm_fwd(h2, Wh, bh, 2, 1, pdf, v, pdf, 1)
def dgrad(): ext.gemm_fwd(gh, Wh, dummy, 3, 0, dz2, dz2, h2, 0)
def dw1(): ext.dw_mfma(dz2, X, torch.zeros(H*D+H, device="cuda"), 0, H*D, -1, -1, -1)

for name, fn in [("L1", l1), ("L2", l2), ("heads", heads), ("dgrad", dgrad), ("dw1", dw1)]:
    for _ in range(3): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(10): fn()
    torch.cuda.synchronize()
    print(f"{name}: {(time.perf_counter()-t0)/10*1e6:.0f} us")
