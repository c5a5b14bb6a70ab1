"""Microbench: the update-path GEMM shapes in isolation (B=1M rows)."""
import os
import sys, time
import torch
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
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
# slack mirrors the gh binding: dgrad's pipelined float4 reads overhang
gh = (torch.randn(B * (P + 1) + 4, device="cuda") * 0.01).narrow(0, 0, B * (P + 1)).view(B, P + 1)
dz2 = torch.empty(B, H, device="cuda")
dummy = torch.zeros(1, device="cuda")

ABL = 0
dwbuf = torch.zeros(H * D + H, device="cuda")
W1t = W1.t().contiguous()  # [D][H] layout-0 variant
W2t = W2.t().contiguous()
def l1t(): ext.gemm_fwd(X, W1t, b1, 1, 0, h1, h1, h1, 0, ABL, 0)
def l2t(): ext.gemm_fwd(h1, W2t, b1, 1, 0, h2, h2, h2, 0, ABL, 0)
def l1(): ext.gemm_fwd(X, W1, b1, 1, 0, h1, h1, h1, 1, ABL, 0)
def l2(): ext.gemm_fwd(h1, W2, b1, 1, 0, h2, h2, h2, 1, ABL, 0)
def heads(): ext.gemm_fwd(h2, Wh, bh, 2, 1, pdf, v, pdf, 1, ABL, 0)
def dgrad(): ext.gemm_fwd(gh, Wh, dummy, 3, 0, dz2, dz2, h2, 0, ABL, 0)
def dw1(): ext.dw_mfma(dz2, X, dwbuf, 0, H*D, -1, -1, -1, ABL)

act = torch.randn(B, A, device="cuda")
advb = torch.randn(B, device="cuda")
etrb = torch.randn(B, device="cuda")
oldvb = torch.randn(B, device="cuda")
oldf = pdf + 0.01 * torch.randn_like(pdf)
cde = torch.empty(0, device="cuda")
def ghk(): ext.ppo_loss_gauss_gh(pdf, oldf, v, oldvb, act, advb, etrb, 0.2, 0.01, 0.5, cde)
def lossf(): ext.ppo_loss_gauss_fwd(pdf, oldf, v, oldvb, act, advb, etrb, 0.2, 0.01, 0.5)

# traffic per call (GB), for effective-bandwidth reporting
GB = {"L1": (B*(D+H))*4e-9, "L1t": (B*(D+H))*4e-9, "L2t": (B*2*H)*4e-9, "L2": (B*2*H)*4e-9, "heads": (B*(H+P+1))*4e-9,
      "dgrad": (B*(P+1+2*H))*4e-9, "dw1": (B*(D+H))*4e-9,
      "gh": (B*(2*P+A+3+P+1))*4e-9, "lossf": (B*(2*P+A+3))*4e-9}

def run(label):
    out = {}
    for name, fn in [("L1", l1), ("L1t", l1t), ("L2", l2), ("L2t", l2t),
                     ("heads", heads), ("dgrad", dgrad), ("dw1", dw1),
                     ("gh", ghk), ("lossf", lossf)]:
        for _ in range(3): fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10): fn()
        torch.cuda.synchronize()
        us = (time.perf_counter() - t0) / 10 * 1e6
        out[name] = us
        print(f"{label:18s} {name:6s} {us:7.0f} us   {GB[name]/us*1e6:6.0f} GB/s-eff")
    return out

base = run("full")
if "--ablate" in sys.argv:
    for abl, label in [(1, "no-X-stage"), (2, "no-W-stage"), (3, "no-stage"),
                       (4, "no-mfma"), (8, "no-epilogue"),
                       (4 + 8, "stage-only")]:
        ABL = abl
        run(f"ablate={label}")
    ABL = 0
