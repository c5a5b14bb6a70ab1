"""Isolated fused-chunk-kernel benchmark (mlp_train.hip diagnosis)."""
import os
import sys, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from dppo_amd.config import DPPOConfig
from dppo_amd.parallel.comm import Comm
from dppo_amd.trainer import DPPOEngine

B = int(sys.argv[1]) if len(sys.argv) > 1 else 4096
iters = int(sys.argv[2]) if len(sys.argv) > 2 else 50
cfg = DPPOConfig(GAME="Humanoid-v4", HIDDEN_SIZES=(64, 64), ACTIVATION="tanh",
                 NUM_ENVS=256, MAX_EPOCH_STEPS=16, EPOCH_MAX=10**6,
                 STOP_EPOCH=10**6, NUM_WORKERS=1, LOG_FILE_PATH="/tmp/l",
                 DEVICE="cuda")
eng = DPPOEngine(cfg, comm=Comm(device="cuda:0"))
eng.CHUNK_KERNEL_MAX_B = 1 << 30
batch, _ = eng.rollout_once()
n = min(B, batch.states.shape[0])
assert eng._can_chunk_kernel(n)
eng.optimizer.lr_dev.fill_(3e-4)
for _ in range(3):
    eng._chunk_kernel_step(batch, 0, n, 0.2)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(iters):
    eng._chunk_kernel_step(batch, 0, n, 0.2)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / iters
print(f"chunk B={n}: {dt*1e6:.1f} us per fused step pair")
