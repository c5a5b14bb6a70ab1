"""Multi-process DP tests on a CPU gloo process group (BASELINE.json
config 1: Pendulum-shaped, 2 CPU workers — the distributed-logic harness
that runs without a GPU)."""

import json
import os
import pickle
import sys

import pytest
import torch
import torch.multiprocessing as mp

from dppo_amd.config import DPPOConfig

WORLD = 2


def _worker(rank, world, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    import torch.distributed as dist

    from dppo_amd.parallel.comm import Comm
    from dppo_amd.trainer import DPPOEngine

    cfg = DPPOConfig(
        GAME="Pendulum-v1", NUM_ENVS=4, MAX_EPOCH_STEPS=12, EPOCH_MAX=8,
        STOP_EPOCH=8, LEARNING_RATE=1e-3, NUM_WORKERS=world,
        LOG_FILE_PATH=os.path.join(out_dir, "logs"), DEVICE="cpu",
        BROADCAST_INTERVAL=0,  # rely on determinism; test checks bit-identity
    )
    comm = Comm(backend="gloo", device="cpu")
    eng = DPPOEngine(cfg, comm=comm)

    # ranks start from different seeds BUT the initial broadcast must have
    # aligned parameters (main.py:48-50 analog)
    p0 = eng.flat_pi.flat_param.clone()
    gathered = comm.all_gather_rows(p0)
    init_identical = bool(torch.allclose(gathered[0], gathered[1]))

    stats_list = []
    for _ in range(3):
        stats, stop = eng.train_round()
        stats_list.append(stats)

    pf = eng.flat_pi.flat_param.clone()
    gathered_after = comm.all_gather_rows(pf)
    # replicas must remain BIT-identical under all-reduced grads +
    # identical Adam state (SURVEY.md §2.3 broadcast-elimination claim)
    final_identical = bool(torch.equal(gathered_after[0], gathered_after[1]))

    # l_mul consensus: both ranks used the same best-rank multiplier
    with open(os.path.join(out_dir, f"rank{rank}.pkl"), "wb") as f:
        pickle.dump(
            {
                "init_identical": init_identical,
                "final_identical": final_identical,
                "l_muls": [s["l_mul"] for s in stats_list],
                "best_ranks": [s["best_rank"] for s in stats_list],
                "param_sum": float(pf.sum()),
            },
            f,
        )
    comm.shutdown()


@pytest.mark.timeout(300)
def test_two_rank_gloo_training(tmp_path):
    port = 29741
    mp.spawn(_worker, args=(WORLD, port, str(tmp_path)), nprocs=WORLD, join=True)
    results = []
    for r in range(WORLD):
        with open(tmp_path / f"rank{r}.pkl", "rb") as f:
            results.append(pickle.load(f))
    for r in results:
        assert r["init_identical"], "initial broadcast failed"
        assert r["final_identical"], "replicas diverged"
    assert results[0]["l_muls"] == results[1]["l_muls"], "l_mul consensus broken"
    assert results[0]["best_ranks"] == results[1]["best_ranks"]
    assert results[0]["param_sum"] == results[1]["param_sum"]


def _allreduce_worker(rank, world, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    from dppo_amd.parallel.comm import Comm, FlatBuffers

    comm = Comm(backend="gloo", device="cpu")
    t = torch.full((10,), float(rank + 1))
    comm.allreduce_mean_(t)
    ok = torch.allclose(t, torch.full((10,), 1.5))

    lin = torch.nn.Linear(4, 3)
    fb = FlatBuffers(lin)
    assert fb.numel == 4 * 3 + 3
    # backward accumulates into the flat grad buffer
    loss = lin(torch.randn(5, 4)).pow(2).sum()
    loss.backward()
    grad_ok = bool(fb.flat_grad.abs().sum() > 0)
    with open(os.path.join(out_dir, f"ar{rank}.json"), "w") as f:
        json.dump({"ok": bool(ok), "grad_ok": grad_ok}, f)
    comm.shutdown()


@pytest.mark.timeout(120)
def test_flat_bucket_allreduce(tmp_path):
    mp.spawn(_allreduce_worker, args=(WORLD, 29753, str(tmp_path)), nprocs=WORLD, join=True)
    for r in range(WORLD):
        with open(tmp_path / f"ar{r}.json") as f:
            d = json.load(f)
        assert d["ok"] and d["grad_ok"]


def _invalid_rank_worker(rank, world, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    from dppo_amd.parallel.comm import Comm
    from dppo_amd.trainer import DPPOEngine

    cfg = DPPOConfig(
        GAME="Pendulum-v1", NUM_ENVS=4, MAX_EPOCH_STEPS=12, EPOCH_MAX=8,
        STOP_EPOCH=8, LEARNING_RATE=1e-3, NUM_WORKERS=world,
        LOG_FILE_PATH=os.path.join(out_dir, "logs"), DEVICE="cpu",
        BROADCAST_INTERVAL=0, MAX_ROLLOUT_RETRIES=2,
    )
    comm = Comm(backend="gloo", device="cpu")
    eng = DPPOEngine(cfg, comm=comm)
    if rank == 1:
        # rank 1 never completes an episode -> every batch invalid; the
        # validity flag must keep all collectives aligned (SURVEY.md §7
        # "hard parts": a rank with no completed episode still
        # participates or the ring deadlocks)
        eng.env.horizons.fill_(10**9)
        eng.env.horizons_i32.fill_(10**9)
    stats_list = []
    for _ in range(3):
        stats, _ = eng.train_round()
        stats_list.append(stats)
    pf = eng.flat_pi.flat_param.clone()
    gathered = comm.all_gather_rows(pf)
    with open(os.path.join(out_dir, f"inv{rank}.pkl"), "wb") as f:
        pickle.dump(
            {
                "final_identical": bool(torch.equal(gathered[0], gathered[1])),
                "best_ranks": [s["best_rank"] for s in stats_list],
                "finite": all(
                    all(v == v for v in s.values()) for s in stats_list),
            },
            f,
        )
    comm.shutdown()


@pytest.mark.timeout(300)
def test_invalid_rank_keeps_ring_alive(tmp_path):
    """A rank that never completes an episode must neither deadlock the
    collectives nor win the best-rank sort."""
    mp.spawn(_invalid_rank_worker, args=(WORLD, 29767, str(tmp_path)),
             nprocs=WORLD, join=True)
    results = []
    for r in range(WORLD):
        with open(tmp_path / f"inv{r}.pkl", "rb") as f:
            results.append(pickle.load(f))
    for r in results:
        assert r["final_identical"], "replicas diverged"
        assert r["finite"]
        assert all(b == 0.0 for b in r["best_ranks"]), \
            "the episode-less rank must never win the sort"


def _minibatch_worker(rank, world, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    from dppo_amd.parallel.comm import Comm
    from dppo_amd.trainer import DPPOEngine

    cfg = DPPOConfig(
        GAME="Pendulum-v1", NUM_ENVS=4, MAX_EPOCH_STEPS=12, EPOCH_MAX=8,
        STOP_EPOCH=8, LEARNING_RATE=1e-3, NUM_WORKERS=world,
        LOG_FILE_PATH=os.path.join(out_dir, "logs"), DEVICE="cpu",
        BROADCAST_INTERVAL=0, MINIBATCH_SIZE=16,  # 48 samples -> 3 chunks
    )
    comm = Comm(backend="gloo", device="cpu")
    eng = DPPOEngine(cfg, comm=comm)
    for _ in range(2):
        eng.train_round()
    pf = eng.flat_pi.flat_param.clone()
    gathered = comm.all_gather_rows(pf)
    with open(os.path.join(out_dir, f"mb{rank}.json"), "w") as f:
        json.dump({"identical": bool(torch.equal(gathered[0], gathered[1]))}, f)
    comm.shutdown()


@pytest.mark.timeout(300)
def test_minibatched_distributed_stays_aligned(tmp_path):
    """BASELINE config 4's minibatched update under DP: every rank takes
    the same chunk walk (sequential, no shuffle) with a gradient
    all-reduce + Adam step per chunk — replicas must stay bit-identical."""
    mp.spawn(_minibatch_worker, args=(WORLD, 29781, str(tmp_path)),
             nprocs=WORLD, join=True)
    for r in range(WORLD):
        with open(tmp_path / f"mb{r}.json") as f:
            assert json.load(f)["identical"], "replicas diverged"


def _drift_guard_worker(rank, world, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    from dppo_amd.parallel.comm import Comm
    from dppo_amd.trainer import DPPOEngine

    cfg = DPPOConfig(
        GAME="Pendulum-v1", NUM_ENVS=4, MAX_EPOCH_STEPS=12, EPOCH_MAX=8,
        STOP_EPOCH=8, LEARNING_RATE=1e-3, NUM_WORKERS=world,
        LOG_FILE_PATH=os.path.join(out_dir, "logs"), DEVICE="cpu",
        BROADCAST_INTERVAL=2,  # periodic rank-0 broadcast (drift guard)
    )
    comm = Comm(backend="gloo", device="cpu")
    eng = DPPOEngine(cfg, comm=comm)
    if rank == 1:
        # inject artificial drift; the round-2 broadcast must erase it
        with torch.no_grad():
            eng.flat_pi.flat_param += 0.05
    for _ in range(3):
        eng.train_round()
    gathered = comm.all_gather_rows(eng.flat_pi.flat_param.clone())
    with open(os.path.join(out_dir, f"dg{rank}.json"), "w") as f:
        json.dump({"identical": bool(torch.equal(gathered[0], gathered[1]))}, f)
    comm.shutdown()


def _curation_worker(rank, world, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    from dppo_amd.parallel.comm import Comm
    from dppo_amd.trainer import DPPOEngine

    cfg = DPPOConfig(
        GAME="Pendulum-v1", NUM_ENVS=4, MAX_EPOCH_STEPS=12, EPOCH_MAX=8,
        STOP_EPOCH=8, LEARNING_RATE=1e-3, NUM_WORKERS=world,
        LOG_FILE_PATH=os.path.join(out_dir, "logs"), DEVICE="cpu",
        BROADCAST_INTERVAL=0, BATCH_CURATION=True,
    )
    comm = Comm(backend="gloo", device="cpu")
    eng = DPPOEngine(cfg, comm=comm)
    batch = eng.collect()
    states_all = comm.all_gather_rows(batch.states.reshape(-1).clone())

    results = {}
    # Scenario A (scripted scores, the reference's sort on Chief.py:51):
    # rank 1's batch has the higher max episode reward -> tower 0 gets
    # rank 1's batch, tower 1 gets rank 0's (the top-N one-per-tower
    # assignment of Chief.py:54-63 at N = world = 2).
    g = torch.zeros(world, 11)
    g[:, 10] = 1.0          # both valid
    g[0, 2], g[1, 2] = 1.0, 5.0
    eng._curate_batches(batch, g)
    results["swap_ok"] = bool(torch.equal(
        batch.states.reshape(-1), states_all[1 - rank]))

    # Scenario B: only rank 0's (post-swap) batch is valid -> both ranks
    # train on rank 0's batch (best-batch recycle for the invalid rank).
    states_all2 = comm.all_gather_rows(batch.states.reshape(-1).clone())
    g2 = torch.zeros(world, 11)
    g2[0, 10] = 1.0
    g2[0, 2], g2[1, 2] = 2.0, 99.0  # invalid rank never wins the sort
    eng._curate_batches(batch, g2)
    results["recycle_ok"] = bool(torch.equal(
        batch.states.reshape(-1), states_all2[0]))

    # Scenario C: end-to-end rounds with curation on — replicas must stay
    # bit-identical (collective schedule aligned on every rank).
    for _ in range(2):
        eng.train_round()
    pf = eng.flat_pi.flat_param.clone()
    gathered = comm.all_gather_rows(pf)
    results["identical"] = bool(torch.equal(gathered[0], gathered[1]))
    with open(os.path.join(out_dir, f"cur{rank}.json"), "w") as f:
        json.dump(results, f)
    comm.shutdown()


@pytest.mark.timeout(300)
def test_batch_curation_matches_reference_decisions(tmp_path):
    """BATCH_CURATION=True reproduces the reference Chief's drain-sort-
    assign decisions (Chief.py:33-53) on scripted score sequences, and
    keeps replicas aligned over full rounds."""
    mp.spawn(_curation_worker, args=(WORLD, 29807, str(tmp_path)),
             nprocs=WORLD, join=True)
    for r in range(WORLD):
        with open(tmp_path / f"cur{r}.json") as f:
            d = json.load(f)
        assert d["swap_ok"], "sorted batch->tower assignment wrong"
        assert d["recycle_ok"], "best-batch recycle for invalid rank wrong"
        assert d["identical"], "replicas diverged under curation"


@pytest.mark.timeout(600)
def test_bench_eight_rank_gloo():
    """bench.py's N>1 code path end-to-end (engine per rank, stats
    all-gather, grad all-reduce, MAX-over-ranks timing reduce) on an
    8-rank CPU gloo group with tiny shapes."""
    import subprocess

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    env.pop("LOCAL_RANK", None)
    cmd = [
        sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
        "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
        "--master-port", "29819", os.path.join(repo, "bench.py"),
        "--gpus", "8", "--steps", "2", "--warmup", "1",
        "--preset", "halfcheetah", "--num-envs", "4", "--rollout", "8",
        "--device", "cpu",
    ]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=540,
                         env=env, cwd=repo)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    assert lines, out.stdout[-2000:]
    result = json.loads(lines[-1])
    assert result["n_gpus"] == 8
    assert result["config"]["parallelism"] == "dp8"
    assert result["value"] > 0 and result["ms_per_step"] > 0
    # whole-job aggregate: 8 ranks x E x T x steps env-steps counted
    assert result["config"]["global_batch"] == 8 * 4 * 8


@pytest.mark.timeout(300)
def test_drift_guard_broadcast_heals_divergence(tmp_path):
    """BROADCAST_INTERVAL>0 periodically re-broadcasts rank-0 weights
    (the Chief.py:67-70 analog kept as a drift guard): injected
    divergence must be healed within the interval."""
    mp.spawn(_drift_guard_worker, args=(WORLD, 29793, str(tmp_path)),
             nprocs=WORLD, join=True)
    for r in range(WORLD):
        with open(tmp_path / f"dg{r}.json") as f:
            assert json.load(f)["identical"], "drift guard did not heal"


def _curation4_worker(rank, world, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    from dppo_amd.parallel.comm import Comm
    from dppo_amd.trainer import DPPOEngine

    cfg = DPPOConfig(
        GAME="Pendulum-v1", NUM_ENVS=2, MAX_EPOCH_STEPS=8, EPOCH_MAX=8,
        STOP_EPOCH=8, LEARNING_RATE=1e-3, NUM_WORKERS=world,
        LOG_FILE_PATH=os.path.join(out_dir, "logs4"), DEVICE="cpu",
        BROADCAST_INTERVAL=0, BATCH_CURATION=True,
    )
    comm = Comm(backend="gloo", device="cpu")
    eng = DPPOEngine(cfg, comm=comm)
    batch = eng.collect()
    states_all = comm.all_gather_rows(batch.states.reshape(-1).clone())

    results = {}
    # scripted 4-rank scenario: ranks 1 and 3 valid with EQUAL keys
    # (stable sort keeps rank 1 first), ranks 0 and 2 invalid.
    # Reference drain-sort-assign (Chief.py:33-53): sorted valid order is
    # [1, 3]; towers 0..3 get sources [1, 3, 1, 3] (order[i % V] recycle).
    g = torch.zeros(world, 11)
    g[1, 10] = 1.0
    g[3, 10] = 1.0
    g[:, 2] = torch.tensor([9.0, 4.0, 9.0, 4.0])  # invalid 9s never win
    eng._curate_batches(batch, g)
    expect_src = [1, 3, 1, 3][rank]
    results["assign_ok"] = bool(torch.equal(
        batch.states.reshape(-1), states_all[expect_src]))
    with open(os.path.join(out_dir, f"cur4_{rank}.json"), "w") as f:
        json.dump(results, f)
    comm.shutdown()


@pytest.mark.timeout(300)
def test_batch_curation_four_rank_ties_and_recycle(tmp_path):
    """4 ranks, duplicate sort keys, two invalid ranks: the stable sort
    tie-break and the order[i % V] best-batch recycle both match the
    reference's drain-sort-assign semantics (Chief.py:33-53)."""
    mp.spawn(_curation4_worker, args=(4, 29833, str(tmp_path)),
             nprocs=4, join=True)
    for r in range(4):
        with open(tmp_path / f"cur4_{r}.json") as f:
            d = json.load(f)
        assert d["assign_ok"], f"rank {r} trained on the wrong batch"
