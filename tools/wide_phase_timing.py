import os
import sys, json, os
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
os.environ["DPPO_TIMER_SYNC"] = "1"
import torch
from dppo_amd.config import DPPOConfig
from dppo_amd.parallel.comm import Comm
from dppo_amd.trainer import DPPOEngine

cfg = DPPOConfig(GAME="Wide-4096", HIDDEN_SIZES=(4096,)*4, ACTIVATION="tanh",
                 DTYPE="bfloat16", NUM_ENVS=4096, MAX_EPOCH_STEPS=16,
                 EPOCH_MAX=10**6, STOP_EPOCH=10**6, LEARNING_RATE=3e-4,
                 NUM_WORKERS=1, LOG_FILE_PATH="/tmp/l", SEED=1)
eng = DPPOEngine(cfg, comm=Comm(device="cuda:0"))
for _ in range(2):
    eng.train_round()
eng.timers.reset()
for _ in range(4):
    eng.train_round()
t = eng.timers.summary()
print(json.dumps({k: round(v/4*1000, 2) for k, v in t.items()}))
