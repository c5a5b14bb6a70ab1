"""Per-phase wall-clock split for a bench.py preset (cuda-sync timers).

Usage: python tools/preset_diag.py <preset> [NUM_ENVS] [rounds]
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from bench import PRESETS  # noqa: E402
from dppo_amd.config import DPPOConfig  # noqa: E402
from dppo_amd.parallel.comm import Comm  # noqa: E402
from dppo_amd.trainer import DPPOEngine  # noqa: E402
from dppo_amd.utils.timers import PhaseTimers  # noqa: E402

preset = sys.argv[1] if len(sys.argv) > 1 else "wide4096"
num_envs = int(sys.argv[2]) if len(sys.argv) > 2 else 0
rounds = int(sys.argv[3]) if len(sys.argv) > 3 else 10

kw = dict(PRESETS[preset])
if num_envs:
    kw["NUM_ENVS"] = num_envs
cfg = DPPOConfig(
    **kw, EPOCH_MAX=10**6, STOP_EPOCH=10**6, LEARNING_RATE=3e-4,
    NUM_WORKERS=1, LOG_FILE_PATH="/tmp/dppo_diag", SEED=1234, DEVICE="cuda",
)
eng = DPPOEngine(cfg, comm=Comm(device="cuda:0"))
for _ in range(3):
    eng.train_round()
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(rounds):
    eng.train_round()
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / rounds
steps = cfg.NUM_ENVS * cfg.MAX_EPOCH_STEPS
print(f"{preset} E={cfg.NUM_ENVS} T={cfg.MAX_EPOCH_STEPS}: "
      f"{dt * 1000:.1f} ms/round  {steps / dt / 1e6:.3f}M env-steps/s")

eng.timers = PhaseTimers(cuda_sync=True)
for _ in range(rounds):
    eng.train_round()
tot = sum(eng.timers.totals.values())
print(f"phase split ({rounds} rounds, sync timers):")
for k, v in sorted(eng.timers.totals.items(), key=lambda kv: -kv[1]):
    print(f"  {k:16s} {v / rounds * 1000:8.2f} ms/round  {100 * v / tot:5.1f}%")
print("mb_graphs:", len(getattr(eng, "_mb_graphs", None) or []),
      "mb_graph_failed:", getattr(eng, "_mb_graph_failed", False),
      "upd_graph:", getattr(eng, "_upd_graph", None) is not None,
      "v3:", eng._can_rollout_v3(), "fuse_rollout:", eng._can_fuse_rollout())
