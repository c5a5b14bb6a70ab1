import sys, time, torch
sys.path.insert(0, "/root/repo")
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
