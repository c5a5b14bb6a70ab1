"""Single-process engine tests (CPU)."""

import math

import pytest
import torch

from dppo_amd.config import DPPOConfig
from dppo_amd.trainer import DPPOEngine, STATS_DIM
from dppo_amd.parallel.comm import Comm


def small_cfg(**kw):
    base = dict(
        GAME="Pendulum-v1",
        NUM_ENVS=8,
        MAX_EPOCH_STEPS=16,
        EPOCH_MAX=10,
        STOP_EPOCH=10,
        HIDDEN_SIZES=(16,),
        LEARNING_RATE=1e-3,
        NUM_WORKERS=1,
        LOG_FILE_PATH="/tmp/dppo_test_logs",
        DEVICE="cpu",
    )
    base.update(kw)
    return DPPOConfig(**base)


@pytest.fixture(scope="module")
def engine():
    return DPPOEngine(small_cfg(), comm=Comm(device="cpu"))


def test_round_runs_and_updates(engine):
    p0 = engine.flat_pi.flat_param.clone()
    stats, stop = engine.train_round()
    assert not stop
    assert engine.CUR_EP == 1
    assert all(math.isfinite(v) for v in stats.values())
    assert not torch.allclose(p0, engine.flat_pi.flat_param)  # params moved


def test_oldpi_synced_at_round_start(engine):
    engine.train_round()
    # after a round, oldpi holds the PRE-update params, so they differ now
    assert not torch.allclose(
        engine.flat_old.flat_param, engine.flat_pi.flat_param
    )
    engine.sync_oldpi()
    torch.testing.assert_close(
        engine.flat_old.flat_param, engine.flat_pi.flat_param
    )


def test_anneal_schedules():
    eng = DPPOEngine(small_cfg(EPOCH_MAX=100, STOP_EPOCH=100), comm=Comm(device="cpu"))
    eng.CUR_EP = 0
    assert eng.current_lr_mul() == 1.0
    assert eng.exploration_rate() == pytest.approx(0.4)
    eng.CUR_EP = 50
    assert eng.current_lr_mul() == pytest.approx(0.5)
    assert eng.exploration_rate() == pytest.approx(0.4 + 0.5 * (0.15 - 0.4))
    eng.CUR_EP = 100
    assert eng.current_lr_mul() == 0.0
    assert eng.exploration_rate() == pytest.approx(0.15)
    eng2 = DPPOEngine(
        small_cfg(SCHEDULE="constant", EPOCH_MAX=100, STOP_EPOCH=100),
        comm=Comm(device="cpu"),
    )
    eng2.CUR_EP = 73
    assert eng2.current_lr_mul() == 1.0


def test_stats_row_layout(engine):
    batch = engine.collect()
    losses = engine.eval_losses(batch, batch.cur_lr)
    row = engine.stats_row(batch, losses)
    assert row.shape == (STATS_DIM,)
    assert row[10] == 1.0  # valid
    assert row[8] == float(engine.CUR_EP)
    assert torch.isfinite(row).all()


def test_stop_rule():
    cfg = small_cfg(EPOCH_MAX=3, STOP_EPOCH=3, MAX_EPOCH_STEPS=8, NUM_ENVS=4)
    eng = DPPOEngine(cfg, comm=Comm(device="cpu"))
    stops = []
    for _ in range(5):
        _, stop = eng.train_round()
        stops.append(stop)
        if stop:
            break
    assert stops[-1] is True
    assert eng.CUR_EP == 4  # rounds 0,1,2 run; round with CUR_EP=3 stops


def test_training_improves_on_tiny_task():
    """Sanity: a few rounds of PPO should not diverge and losses stay finite;
    value loss should drop on a stationary reward landscape."""
    cfg = small_cfg(
        EPOCH_MAX=30, STOP_EPOCH=30, LEARNING_RATE=3e-3, NUM_ENVS=16,
        MAX_EPOCH_STEPS=32,
    )
    eng = DPPOEngine(cfg, comm=Comm(device="cpu"))
    first_vl, last_vl = None, None
    for i in range(12):
        stats, _ = eng.train_round()
        if i == 0:
            first_vl = stats["valueLoss"]
        last_vl = stats["valueLoss"]
        assert math.isfinite(stats["total_loss"])
    assert last_vl < first_vl  # critic is learning the synthetic returns


def test_worker_chief_api():
    from dppo_amd.worker import Worker
    from dppo_amd.chief import Chief
    from dppo_amd.utils.coordinator import Coordinator

    cfg = small_cfg()
    coord = Coordinator()
    w = Worker("Worker_N0", cfg, coord=coord, comm=Comm(device="cpu"))
    c = Chief("Chief", cfg, coord=coord, workers=[w])
    assert c.engine is w.engine
    stats = w.work(max_rounds=2)
    assert coord.should_stop()
    assert "total_loss" in stats
    # Chief.act: single state in, action out
    obs_dim = w.engine.obs_space.shape[0]
    a = c.act(torch.randn(obs_dim).numpy())
    assert a.shape == (w.engine.act_space.shape[0],)
    # Worker.act parity: (action, pred_v)
    act, predv = w.act(torch.randn(obs_dim).numpy())
    assert isinstance(predv, float)


def test_minibatched_update():
    """BASELINE config 4 scheme: MINIBATCH_SIZE chunks per update epoch."""
    cfg = small_cfg(MINIBATCH_SIZE=32, NUM_ENVS=8, MAX_EPOCH_STEPS=16)
    eng = DPPOEngine(cfg, comm=Comm(device="cpu"))
    p0 = eng.flat_pi.flat_param.detach().clone()
    stats, _ = eng.train_round()
    assert math.isfinite(stats["total_loss"])
    assert not torch.allclose(p0, eng.flat_pi.flat_param.detach())
    # 16*8=128 samples / 32 per chunk * 4 epochs = 16 optimizer steps
    assert eng.optimizer.state_dict()["state"][0]["step"] == 16


def test_push_guard_retry_returns_invalid_batch():
    """Worker.py:135 push-guard analog: when no episode can complete, the
    engine retries MAX_ROLLOUT_RETRIES times and returns the (invalid)
    batch instead of hanging; the stats row then reports valid=0 and a
    sort key of -inf."""
    import math as _math

    cfg = small_cfg(MAX_ROLLOUT_RETRIES=2)
    eng = DPPOEngine(cfg, comm=Comm(device="cpu"))
    eng.env.horizons.fill_(10**9)
    eng.env.horizons_i32.fill_(10**9)
    batch = eng.collect()
    assert not batch.valid
    row = eng.stats_row(batch, {
        "policyLoss": 0.0, "valueLoss": 0.0,
        "entropyLoss": 0.0, "total_loss": 0.0,
    })
    assert row[10] == 0.0           # valid flag
    assert row[2] == -_math.inf     # epr_max never wins the sort


@pytest.mark.parametrize("game,hidden", [
    ("MultiLever-v0", (32,)), ("BitFlipper-v0", (32,))])
def test_multidiscrete_multibinary_training(game, hidden):
    """Every reference action-space family trains end-to-end
    (make_pdtype dispatch, reference Others/distributions.py:231-243) —
    not just Box/Discrete: MultiDiscrete and MultiBinary get synthetic
    envs and run the full round protocol."""
    cfg = DPPOConfig(
        GAME=game, HIDDEN_SIZES=hidden, ACTIVATION="tanh", NUM_ENVS=8,
        MAX_EPOCH_STEPS=16, EPOCH_MAX=100, STOP_EPOCH=100, NUM_WORKERS=1,
        LOG_FILE_PATH="/tmp/dppo_test_logs", DEVICE="cpu", SEED=3,
    )
    eng = DPPOEngine(cfg, comm=Comm(device="cpu"))
    p0 = eng.flat_pi.flat_param.detach().clone()
    for _ in range(2):
        stats, stop = eng.train_round()
    assert all(v == v for v in stats.values())
    assert not torch.equal(p0, eng.flat_pi.flat_param.detach())
    # recorded actions respect the space
    batch, _ = eng.rollout_once()
    if game == "MultiLever-v0":
        nvec = eng.act_space.nvec
        a = batch.actions.view(-1, len(nvec))
        assert a.dtype == torch.long
        for k, n in enumerate(nvec):
            assert int(a[:, k].min()) >= 0 and int(a[:, k].max()) < int(n)
    else:
        a = batch.actions
        assert set(a.unique().tolist()) <= {0.0, 1.0}


def test_eval_losses_equal_recompute():
    """eval_losses uses the RECORDED rollout outputs instead of re-running
    pi (trainer._losses recorded_pi=True): the reference evaluates its
    losses while pi still equals oldpi (Worker.py:117-118 after the
    sync at Worker.py:42), so the recorded pdflat/v must equal a true
    forward pass bit-for-bit on the eager path."""
    eng = DPPOEngine(small_cfg(), comm=Comm(device="cpu"))
    batch = eng.collect()
    fast = eng.eval_losses(batch, l_mul=1.0)
    with torch.no_grad():
        slow = {k: float(v)
                for k, v in eng._losses(batch, 1.0, recorded_pi=False).items()}
    for k in fast:
        assert abs(fast[k] - slow[k]) < 1e-5, (k, fast[k], slow[k])


@pytest.mark.parametrize("hidden,act", [
    ((16, 16), "relu"),
    ((16, 16, 16), "tanh"),   # 3 hidden layers (engine/kernel-gate max)
    ((24,), "relu"),
])
def test_config_matrix_rounds(hidden, act):
    """Hidden-depth x activation matrix: every combination steps through
    a full round (rollout + GAE + update) and stays finite."""
    eng = DPPOEngine(small_cfg(HIDDEN_SIZES=hidden, ACTIVATION=act),
                     comm=Comm(device="cpu"))
    before = eng.flat_pi.flat_param.clone()
    eng.train_round()
    assert torch.isfinite(eng.flat_pi.flat_param).all()
    assert not torch.equal(before, eng.flat_pi.flat_param)


def test_batch_curation_single_rank_is_identity():
    """BATCH_CURATION=True with world_size 1 must degenerate to the
    rank-local default (the sort over one batch is the identity)."""
    torch.manual_seed(0)
    a = DPPOEngine(small_cfg(SEED=11, BATCH_CURATION=True),
                   comm=Comm(device="cpu"))
    torch.manual_seed(0)
    b = DPPOEngine(small_cfg(SEED=11, BATCH_CURATION=False),
                   comm=Comm(device="cpu"))
    # the eager rollout samples from the GLOBAL torch generator: pin it
    # before each round so the two engines see identical draws
    torch.manual_seed(123)
    a.train_round()
    torch.manual_seed(123)
    b.train_round()
    assert torch.equal(a.flat_pi.flat_param, b.flat_pi.flat_param)


def test_minibatch_with_curation_and_multi_hidden():
    """Interaction case: minibatched epochs + curation flag + 2 hidden
    layers (the config-4 scheme with every optional feature on)."""
    eng = DPPOEngine(small_cfg(MINIBATCH_SIZE=32, NUM_ENVS=8,
                               MAX_EPOCH_STEPS=16, HIDDEN_SIZES=(16, 16),
                               BATCH_CURATION=True),
                     comm=Comm(device="cpu"))
    eng.train_round()
    eng.train_round()
    assert torch.isfinite(eng.flat_pi.flat_param).all()
