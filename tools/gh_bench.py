"""Isolated ppo_loss_gauss_gh micro-bench (flagship shape by default).

Run twice to A/B the LDS-tiled kernel against the wave-per-row one —
the dispatch env is latched at first call, so one process per variant:
    python tools/gh_bench.py --tile 1
    python tools/gh_bench.py --tile 0
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
import torch


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--tile", type=int, default=1)
    ap.add_argument("--B", type=int, default=16 * 1024 * 1024)
    ap.add_argument("--A", type=int, default=17)
    ap.add_argument("--iters", type=int, default=20)
    args = ap.parse_args()
    os.environ["DPPO_GH_TILE"] = str(args.tile)
    from dppo_amd.ops import hip_ext

    ext = hip_ext()
    B, A = args.B, args.A
    dev = "cuda"
    g = torch.Generator(device=dev).manual_seed(0)
    pdflat = torch.randn(B, 2 * A, device=dev, generator=g)
    oldflat = pdflat + 0.01 * torch.randn(B, 2 * A, device=dev, generator=g)
    vpred = torch.randn(B, device=dev, generator=g)
    oldv = vpred + 0.01 * torch.randn(B, device=dev, generator=g)
    act = torch.randn(B, A, device=dev, generator=g)
    adv = torch.randn(B, device=dev, generator=g)
    etr = torch.randn(B, device=dev, generator=g)
    clip_dev = torch.empty(0, device=dev)

    def call():
        return ext.ppo_loss_gauss_gh(pdflat, oldflat, vpred, oldv, act, adv,
                                     etr, 0.2, 0.0, 0.5, clip_dev)

    gh = call()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        call()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.iters
    nbytes = B * (2 * A * 4 * 2 + A * 4 + 4 * 4 + gh.shape[1] * 4)
    print(f"tile={args.tile} B={B} A={A}: {dt * 1e3:.3f} ms  "
          f"{nbytes / dt / 1e12:.2f} TB/s")
    print("checksum", gh.double().abs().sum().item())


if __name__ == "__main__":
    main()
