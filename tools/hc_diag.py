import os
import sys, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from dppo_amd.config import DPPOConfig
from dppo_amd.parallel.comm import Comm
from dppo_amd.trainer import DPPOEngine
base = dict(GAME="HalfCheetah-v4", HIDDEN_SIZES=(64,64), ACTIVATION="tanh",
            NUM_ENVS=64, MAX_EPOCH_STEPS=100, EPOCH_MAX=10**6, STOP_EPOCH=10**6,
            LEARNING_RATE=3e-4, NUM_WORKERS=1, LOG_FILE_PATH="/tmp/l", DEVICE="cuda")
for name, kw in [("default", {}), ("no-graphs", dict(USE_GRAPHS=False)),
                 ("eager-kernels", dict(USE_HIP_KERNELS="never"))]:
    cfg = DPPOConfig(**base, **kw)
    eng = DPPOEngine(cfg, comm=Comm(device="cuda:0"))
    for _ in range(5): eng.train_round()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(20): eng.train_round()
    torch.cuda.synchronize()
    dt=(time.perf_counter()-t0)/20
    print(f"{name:14s} {dt*1000:6.2f} ms/round  {64*100/dt/1e6:.2f}M steps/s")

# per-phase split of the default config (cuda_sync timers)
from dppo_amd.utils.timers import PhaseTimers
cfg = DPPOConfig(**base)
eng = DPPOEngine(cfg, comm=Comm(device="cuda:0"))
for _ in range(5): eng.train_round()
eng.timers = PhaseTimers(cuda_sync=True)
for _ in range(20): eng.train_round()
tot = sum(eng.timers.totals.values())
print("phase split (default config, 20 rounds):")
for k, v in sorted(eng.timers.totals.items(), key=lambda kv: -kv[1]):
    print(f"  {k:16s} {v/20*1000:7.3f} ms/round  {100*v/tot:5.1f}%")
