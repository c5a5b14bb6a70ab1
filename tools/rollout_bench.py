import os
import torch, time, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from dppo_amd.config import DPPOConfig
from dppo_amd.parallel.comm import Comm
from dppo_amd.trainer import DPPOEngine
cfg = DPPOConfig(GAME="Humanoid-v4", HIDDEN_SIZES=(64,64), ACTIVATION="tanh",
                 NUM_ENVS=4096, MAX_EPOCH_STEPS=64, EPOCH_MAX=10**6, STOP_EPOCH=10**6,
                 NUM_WORKERS=1, LOG_FILE_PATH="/tmp/l", DEVICE="cuda")
eng = DPPOEngine(cfg, comm=Comm(device="cuda:0"))
for _ in range(2): eng._rollout_once_hip()
torch.cuda.synchronize()
t0=time.perf_counter()
for _ in range(5): eng._rollout_once_hip()
torch.cuda.synchronize()
print("rollout ms:", (time.perf_counter()-t0)/5*1000)
