"""Learning-curve run: logs per-round stats JSONL (evidence that the
optimized path LEARNS, not just runs fast)."""
import os
import json, sys, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
from dppo_amd.config import DPPOConfig
from dppo_amd.parallel.comm import Comm
from dppo_amd.trainer import DPPOEngine

out = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/curve.jsonl"
n = int(sys.argv[2]) if len(sys.argv) > 2 else 400
E = int(sys.argv[3]) if len(sys.argv) > 3 else 65536
cfg = DPPOConfig(GAME="Humanoid-v4", HIDDEN_SIZES=(64, 64), ACTIVATION="tanh",
                 NUM_ENVS=E, MAX_EPOCH_STEPS=64, EPOCH_MAX=n, STOP_EPOCH=n,
                 LEARNING_RATE=3e-4, NUM_WORKERS=1,
                 LOG_FILE_PATH="/tmp/curve", DEVICE="cuda")
eng = DPPOEngine(cfg, comm=Comm(device="cuda:0"))
t0 = time.perf_counter()
with open(out, "w") as f:
    for i in range(n):
        stats, stop = eng.train_round()
        rec = {"round": i, **{k: float(v) for k, v in stats.items()}}
        f.write(json.dumps(rec) + "\n")
        if i % 50 == 0:
            print(f"r{i}: epr_mean={stats.get('epr_mean', 0):.3f} "
                  f"vloss={stats['valueLoss']:.3f} score={stats.get('score', 0):.3f}")
        if stop:
            break
print(f"{n} rounds in {time.perf_counter()-t0:.0f}s")
