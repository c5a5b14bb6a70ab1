"""Stability soak: many rounds at the flagship config; reports memory."""
import os
import sys, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from dppo_amd.config import DPPOConfig
from dppo_amd.parallel.comm import Comm
from dppo_amd.trainer import DPPOEngine

cfg = DPPOConfig(GAME="Humanoid-v4", HIDDEN_SIZES=(64, 64), ACTIVATION="tanh",
                 NUM_ENVS=32768, MAX_EPOCH_STEPS=64, EPOCH_MAX=10**6,
                 STOP_EPOCH=10**6, LEARNING_RATE=3e-4, NUM_WORKERS=1,
                 LOG_FILE_PATH="/tmp/soak", DEVICE="cuda")
eng = DPPOEngine(cfg, comm=Comm(device="cuda:0"))
t0 = time.perf_counter()
n = int(sys.argv[1]) if len(sys.argv) > 1 else 500
for i in range(n):
    stats, _ = eng.train_round()
    if i % 100 == 0:
        assert all(v == v for v in stats.values()), f"NaN at round {i}"
        print(f"round {i}: total_loss={stats['total_loss']:.5f} "
              f"mem={torch.cuda.max_memory_allocated()/2**30:.2f} GiB")
torch.cuda.synchronize()
dt = time.perf_counter() - t0
print(f"{n} rounds in {dt:.1f}s -> {cfg.NUM_ENVS*64*n/dt/1e6:.1f}M env-steps/s; "
      f"peak mem {torch.cuda.max_memory_allocated()/2**30:.2f} GiB")
