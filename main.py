"""DPPO training entrypoint (reference main.py:11-79, rebuilt MI355X-native).

Single GPU / CPU:
    python main.py --game Humanoid-v4 --rounds 10

One process per GPU over RCCL:
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 main.py --game Humanoid-v4

The reference builds a literal config dict, a shared tf.Session, a
Coordinator, two Events and per-worker deques, then spawns worker threads
and runs the Chief loop (main.py:31-62).  Here each process builds one
Worker (its rank's engine); rank 0 also builds the Chief facade; the
Event/deque fabric is replaced by the synchronous RCCL round protocol.
After training, a short greedy eval loop prints per-episode rewards
(main.py:67-79 — finite by default instead of the reference's infinite
render loop).
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch

from dppo_amd import DPPOConfig
from dppo_amd.chief import Chief
from dppo_amd.checkpoint import save_state, load_state
from dppo_amd.config import game_spaces
from dppo_amd.envs.synthetic import BatchedSyntheticEnv
from dppo_amd.parallel.comm import Comm
from dppo_amd.utils.coordinator import Coordinator
from dppo_amd.worker import Worker


def build_config(args: argparse.Namespace) -> DPPOConfig:
    cfg = DPPOConfig()
    if args.config:
        cfg = DPPOConfig.from_json(args.config)
    overrides = {}
    if args.game:
        overrides["GAME"] = args.game
    if args.num_envs:
        overrides["NUM_ENVS"] = args.num_envs
    if args.epoch_max:
        overrides["EPOCH_MAX"] = args.epoch_max
        overrides["STOP_EPOCH"] = args.epoch_max
    if args.hidden:
        overrides["HIDDEN_SIZES"] = tuple(int(x) for x in args.hidden.split(","))
    if args.activation:
        overrides["ACTIVATION"] = args.activation
    if args.logdir:
        overrides["LOG_FILE_PATH"] = args.logdir
    overrides["NUM_WORKERS"] = int(os.environ.get("WORLD_SIZE", "1"))
    return cfg.replace(**overrides)


def main() -> None:
    p = argparse.ArgumentParser(description="MI355X-native DPPO trainer")
    p.add_argument("--config", type=str, default=None, help="JSON config overlay")
    p.add_argument("--game", type=str, default=None)
    p.add_argument("--num-envs", type=int, default=None)
    p.add_argument("--epoch-max", type=int, default=None)
    p.add_argument("--hidden", type=str, default=None, help="e.g. 64,64")
    p.add_argument("--activation", type=str, default=None, choices=["relu", "tanh"])
    p.add_argument("--rounds", type=int, default=None, help="cap on training rounds")
    p.add_argument("--logdir", type=str, default=None)
    p.add_argument("--save", type=str, default=None, help="checkpoint path to write")
    p.add_argument("--restore", type=str, default=None, help="checkpoint path to load")
    p.add_argument("--eval-episodes", type=int, default=3)
    args = p.parse_args()

    cfg = build_config(args)
    comm = Comm()
    coord = Coordinator()

    worker = Worker(f"Worker_N{comm.rank}", cfg, coord=coord, comm=comm)
    chief = Chief("Chief", cfg, coord=coord, workers=[worker]) if comm.rank == 0 else None
    if args.restore:
        load_state(args.restore, worker.engine)

    start = time.time()
    worker.work(max_rounds=args.rounds)
    comm.barrier()
    if comm.rank == 0:
        print("TRAINING FINISHED.")
        print("Train time elapsed:", time.time() - start, "seconds")
        print("phase timers:", json.dumps(worker.engine.timers.summary(), indent=2))

    if args.save:
        save_state(args.save, worker.engine)

    # greedy eval loop (main.py:67-79), rank 0, finite
    if comm.rank == 0 and args.eval_episodes > 0 and chief is not None:
        obs_space, act_space = game_spaces(cfg.GAME)
        env = BatchedSyntheticEnv(
            obs_space, act_space, num_envs=1, device=str(comm.device),
            seed=cfg.SEED + 99991, horizon=max(cfg.MAX_EPOCH_STEPS // 2, 4),
        )
        s = env.reset()
        epr, done_count = 0.0, 0
        while done_count < args.eval_episodes:
            a = chief.engine.act(s[0]).unsqueeze(0)
            s, r, done, _ = env.step(a)
            epr += float(r[0])
            if bool(done[0]):
                print(f"eval episode reward: {epr:.3f}")
                epr = 0.0
                done_count += 1
    comm.shutdown()


if __name__ == "__main__":
    main()
