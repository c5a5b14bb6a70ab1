"""Phase ablation of the rollout kernel (guide §5.4: ablate before
optimizing).  Runtime mask: 1=trunk 2=heads 4=sampling 8=env 16=writes."""
import os
import sys, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from dppo_amd.config import DPPOConfig
from dppo_amd.parallel.comm import Comm
from dppo_amd.trainer import DPPOEngine
from dppo_amd.ops import require_hip_ext

ext = require_hip_ext()
E = int(sys.argv[1]) if len(sys.argv) > 1 else 65536
T = int(sys.argv[2]) if len(sys.argv) > 2 else 64
cfg = DPPOConfig(GAME="Humanoid-v4", HIDDEN_SIZES=(64, 64), ACTIVATION="tanh",
                 NUM_ENVS=E, MAX_EPOCH_STEPS=T, EPOCH_MAX=10**6,
                 STOP_EPOCH=10**6, NUM_WORKERS=1, LOG_FILE_PATH="/tmp/l",
                 DEVICE="cuda")
eng = DPPOEngine(cfg, comm=Comm(device="cuda:0"))
eng._rollout_once_hip()  # allocate persistent buffers
blob, offsets, dims = eng._rollout_weight_blob()
env = eng.env
low, high = -1.0, 1.0

def run(mask):
    ext.rollout_run(blob, offsets, dims, 1, env.blob, env.rank_eff,
                    env.horizons_i32, float(env.NOISE), low, high, 0.2,
                    env.x, env.t, eng.epr, T, 17, 1234,
                    eng._rollout_out, mask)

for name, mask in [("full", 0), ("-trunk", 1), ("-heads", 2), ("-sample", 4),
                   ("-env", 8), ("-writes", 16),
                   ("-env-noise", 32), ("-env-lowain", 64),
                   ("-env-noise-lowain", 96), ("nothing(127)", 127)]:
    for _ in range(2): run(mask)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(6): run(mask)
    torch.cuda.synchronize()
    print(f"{name:42s} {(time.perf_counter()-t0)/6*1000:7.2f} ms")
