"""A/B: fused whole-rollout kernel vs v3 per-step GEMM rollout."""
import os, sys, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from dppo_amd.config import DPPOConfig
from dppo_amd.parallel.comm import Comm
from dppo_amd.trainer import DPPOEngine

E = int(sys.argv[1]) if len(sys.argv) > 1 else 65536
for name, env_v in [("fused", "0"), ("v3", "1")]:
    os.environ["DPPO_ROLLOUT_V3"] = env_v
    cfg = DPPOConfig(GAME="Humanoid-v4", HIDDEN_SIZES=(64, 64),
                     ACTIVATION="tanh", NUM_ENVS=E, MAX_EPOCH_STEPS=64,
                     EPOCH_MAX=10**6, STOP_EPOCH=10**6, NUM_WORKERS=1,
                     LOG_FILE_PATH="/tmp/l", DEVICE="cuda")
    eng = DPPOEngine(cfg, comm=Comm(device="cuda:0"))
    for _ in range(3):
        eng.rollout_once()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    n = 10
    for _ in range(n):
        eng.rollout_once()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / n
    print(f"{name:6s} E={E}: {dt*1000:7.2f} ms/rollout "
          f"{E*64/dt/1e6:7.1f}M env-steps/s")
