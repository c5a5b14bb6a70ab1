"""Perf probe for the bf16 MFMA GEMM (bf16_mm256) at the wide-config
shapes, A/B'd against torch/rocBLAS bf16 matmul on the same random data
(guide §5.4 rule 25: random operands, within-probe interleave)."""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch

from dppo_amd.ops import require_hip_ext


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=20)
    args = p.parse_args()
    ext = require_hip_ext()
    e = torch.empty(0, device="cuda")
    eb = torch.empty(0, device="cuda", dtype=torch.bfloat16)

    shapes = [
        ("fwd 65536x4096x4096", 65536, 4096, 4096),
        ("dW   4096x4096x65536", 4096, 4096, 65536),
        ("sq   8192x8192x8192", 8192, 8192, 8192),
    ]
    for name, M, N, K in shapes:
        A = (torch.randn(M, K, device="cuda") * 0.3).bfloat16()
        B = (torch.randn(N, K, device="cuda") * 0.3).bfloat16()
        C = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
        fl = 2.0 * M * N * K

        t_mm = bench(lambda: ext.bf16_mm256(A, B, C, 0, e, eb, e, 0, eb, 0, e, 0),
                     args.iters)
        Bt = B.float().t().bfloat16().t().contiguous()  # same layout
        out = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
        t_roc = bench(lambda: torch.mm(A, B.t(), out=out), args.iters)
        print(f"{name}: mm256 {t_mm*1e3:8.3f} ms = {fl/t_mm/1e12:7.1f} TF/s"
              f" | rocBLAS {t_roc*1e3:8.3f} ms = {fl/t_roc/1e12:7.1f} TF/s")

        t_tanh = bench(lambda: ext.bf16_mm256(
            A, B, C, 1, torch.zeros(N, device="cuda"), eb, e, 0, eb, 0, e, 0), args.iters)
        print(f"  +tanh-bias epilogue: {t_tanh*1e3:8.3f} ms = "
              f"{fl/t_tanh/1e12:7.1f} TF/s")
        CT = torch.empty(N, M, device="cuda", dtype=torch.bfloat16)
        sums = torch.zeros(N, device="cuda")
        t_dw = bench(lambda: ext.bf16_mm256(
            A, B, C, 1, torch.zeros(N, device="cuda"), eb, e, 0, CT, M,
            sums, 0), args.iters)
        print(f"  +dual-write+colsum:  {t_dw*1e3:8.3f} ms = "
              f"{fl/t_dw/1e12:7.1f} TF/s")
        del CT, sums
        del A, B, C, out, Bt
        torch.cuda.empty_cache()

    # transpose probe
    x = torch.randn(65536, 4096, device="cuda").bfloat16()
    o = torch.empty(4096, 65536, device="cuda", dtype=torch.bfloat16)
    t_tr = bench(lambda: ext.bf16_transpose(x, o, e, 0, 65536, 4096,
                                            4096, 65536), args.iters)
    gb = x.numel() * 2 * 2 / 1e9
    print(f"transpose 65536x4096: {t_tr*1e3:.3f} ms = {gb/t_tr:.0f} GB/s")


if __name__ == "__main__":
    main()
