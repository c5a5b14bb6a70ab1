"""Fused rollout kernel (rollout.hip) correctness vs eager references.

The kernel's RNG is counter-based (not torch's), so tests check
RNG-independent invariants:
  - recorded pdflats/values == eager pi(recorded states)  (exact MLP math)
  - env transition equation holds exactly at noise=0
  - reward definition, done schedule, episode moments
  - sampled actions are N(mean, std) distributed; epsilon overlay bounds
"""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

from dppo_amd.config import DPPOConfig
from dppo_amd.ops import require_hip_ext
from dppo_amd.parallel.comm import Comm
from dppo_amd.trainer import DPPOEngine


def make_engine(**kw):
    base = dict(
        GAME="Humanoid-v4", HIDDEN_SIZES=(64, 64), ACTIVATION="tanh",
        NUM_ENVS=256, MAX_EPOCH_STEPS=32, EPOCH_MAX=1000, STOP_EPOCH=1000,
        NUM_WORKERS=1, LOG_FILE_PATH="/tmp/dppo_gpu_test_logs", DEVICE="cuda",
    )
    base.update(kw)
    return DPPOEngine(DPPOConfig(**base), comm=Comm(device="cuda:0"))


def run_kernel(eng, T=32, eps=0.0, noise=None, seed=1234):
    ext = require_hip_ext()
    env = eng.env
    if noise is not None:
        env.NOISE = noise
    low = float(eng.act_space.low.flat[0])
    high = float(eng.act_space.high.flat[0])
    with torch.no_grad():
        blob, offsets, dims = eng._rollout_weight_blob()
    return ext.rollout_run(
        blob, offsets, dims,
        1 if eng.cfg.ACTIVATION == "tanh" else 0,
        env.blob, env.rank_eff, env.horizons_i32,
        float(env.NOISE), low, high, float(eps),
        env.x, env.t, eng.epr, T, eng.act_space.shape[0], seed,
        torch.empty(0, device="cuda"), 0,
    )


def test_recorded_policy_outputs_match_eager_forward():
    eng = make_engine()
    T, E = 32, 256
    states, pdflats, actions, values, rewards, dones, boot_v, mom = run_kernel(eng, T)
    s = states.reshape(T * E, -1)
    with torch.no_grad():
        v_ref, flat_ref = eng.pi(s)
    torch.testing.assert_close(pdflats.reshape(T * E, -1), flat_ref,
                               atol=2e-4, rtol=2e-4)
    torch.testing.assert_close(values.reshape(T * E), v_ref, atol=2e-4, rtol=2e-4)
    # bootstrap value on the final env state
    with torch.no_grad():
        vb, _ = eng.pi(eng.env.x)
    torch.testing.assert_close(boot_v, vb, atol=2e-4, rtol=2e-4)


def test_env_transition_exact_at_zero_noise():
    eng = make_engine(NUM_ENVS=128)
    env = eng.env
    T = 16
    states, pdflats, actions, values, rewards, dones, boot_v, mom = run_kernel(
        eng, T, noise=0.0)
    # x_{t+1} = tanh(x_t * d + (x_t @ V) @ U + a_t @ B) wherever not done
    x = states  # [T,E,D]
    a = actions
    for t in range(T - 1):
        pred = torch.tanh(x[t] * env.d + (x[t] @ env.V) @ env.U + a[t] @ env.B)
        live = dones[t] == 0
        torch.testing.assert_close(x[t + 1][live], pred[live],
                                   atol=3e-5, rtol=3e-5)
        # reward is computed on the PRE-reset next state
        r_pred = 1.0 - pred.pow(2).mean(dim=-1)
        torch.testing.assert_close(rewards[t], r_pred, atol=3e-5, rtol=3e-5)


def test_done_schedule_matches_horizons():
    eng = make_engine(NUM_ENVS=64)
    env = eng.env
    T = 40
    *_, rewards, dones, boot_v, mom = run_kernel(eng, T)
    hor = env.horizons_i32.long().cpu()
    d_cpu = dones.cpu()
    for e in range(8):
        h = int(hor[e])
        expect = torch.tensor([1.0 if (t + 1) % h == 0 else 0.0 for t in range(T)])
        torch.testing.assert_close(d_cpu[:, e], expect)


def test_episode_moments_match_recomputation():
    eng = make_engine(NUM_ENVS=128)
    T = 48
    epr0 = eng.epr.clone()
    *_, rewards, dones, boot_v, mom = run_kernel(eng, T)
    # reconstruct on host
    r, d = rewards.cpu(), dones.cpu()
    epr = epr0.cpu().clone()
    eps_list = []
    for t in range(T):
        epr += r[t]
        fin = d[t] > 0
        eps_list += epr[fin].tolist()
        epr[fin] = 0.0
    eps_t = torch.tensor(eps_list)
    assert int(mom[0]) == len(eps_list)
    torch.testing.assert_close(float(mom[1]), float(eps_t.sum()), atol=2e-2, rtol=1e-4)
    torch.testing.assert_close(float(mom[3]), float(eps_t.min()), atol=1e-4, rtol=1e-4)
    torch.testing.assert_close(float(mom[4]), float(eps_t.max()), atol=1e-4, rtol=1e-4)
    # engine's persistent epr matches the reconstruction
    torch.testing.assert_close(eng.epr.cpu(), epr, atol=1e-4, rtol=1e-4)


def test_sampled_actions_are_gaussian():
    eng = make_engine(NUM_ENVS=512)
    T = 32
    states, pdflats, actions, *_ = run_kernel(eng, T, eps=0.0)
    A = eng.act_space.shape[0]
    mean, logstd = pdflats.reshape(-1, 2 * A).chunk(2, dim=-1)
    z = (actions.reshape(-1, A) - mean) / logstd.exp()
    n = z.numel()
    assert abs(float(z.mean())) < 5.0 / math.sqrt(n)
    assert abs(float(z.std()) - 1.0) < 0.01


def test_epsilon_overlay_changes_action_distribution():
    eng = make_engine(NUM_ENVS=256)
    states, pdflats, actions, *_ = run_kernel(eng, 16, eps=1.0)
    a = actions.reshape(-1)
    # eps=1: all actions uniform in [low, high] = [-1, 1]
    assert float(a.min()) >= -1.0 and float(a.max()) <= 1.0
    assert abs(float(a.mean())) < 0.02
    # uniform variance = (b-a)^2/12 = 1/3
    assert abs(float(a.var()) - 1.0 / 3.0) < 0.02


def test_engine_hip_rollout_trains():
    eng = make_engine()
    assert eng._can_fuse_rollout()
    p0 = eng.flat_pi.flat_param.detach().clone()
    for _ in range(3):
        stats, stop = eng.train_round()
    assert all(math.isfinite(v) for v in stats.values())
    assert not torch.equal(p0, eng.flat_pi.flat_param.detach())


def test_rollout_batch_valid_and_stats():
    eng = make_engine()
    batch = eng.collect()
    assert batch.valid
    row = eng.stats_row(batch, eng.eval_losses(batch, batch.cur_lr))
    assert torch.isfinite(row).all()


# ---- v3 per-step GEMM rollout path (trainer._rollout_once_hip_v3) ----

def v3_views(eng):
    c = eng.cfg
    T, E = c.MAX_EPOCH_STEPS, c.NUM_ENVS
    D = eng.obs_space.shape[0]
    A = eng.act_space.shape[0]
    out, o, res = eng._rollout_out, 0, []
    for shape in [(T, E, D), (T, E, 2 * A), (T, E, A), (T, E), (T, E),
                  (T, E), (E,)]:
        n = 1
        for sd in shape:
            n *= sd
        res.append(out.narrow(0, o, n).view(shape))
        o += n
    return res


def test_v3_rollout_invariants(monkeypatch):
    monkeypatch.setenv("DPPO_ROLLOUT_V3", "1")
    eng = make_engine(NUM_ENVS=128, MAX_EPOCH_STEPS=16, USE_GRAPHS=False)
    eng.env.NOISE = 0.0
    assert eng._can_rollout_v3()
    batch, _ = eng.rollout_once()
    assert batch.valid
    env = eng.env
    states, pdflats, actions, values, rewards, dones, boot_v = v3_views(eng)
    T = 16
    # recorded policy outputs match an eager forward of the recorded states
    s = states.reshape(-1, states.shape[-1])
    with torch.no_grad():
        v_ref, flat_ref = eng.pi(s)
    torch.testing.assert_close(pdflats.reshape(s.shape[0], -1), flat_ref,
                               atol=2e-4, rtol=2e-4)
    torch.testing.assert_close(values.reshape(-1), v_ref, atol=2e-4,
                               rtol=2e-4)
    with torch.no_grad():
        vb, _ = eng.pi(env.x)
    torch.testing.assert_close(boot_v, vb, atol=2e-4, rtol=2e-4)
    # exact env transition at noise=0
    for t in range(T - 1):
        pred = torch.tanh(states[t] * env.d +
                          (states[t] @ env.V) @ env.U + actions[t] @ env.B)
        live = dones[t] == 0
        torch.testing.assert_close(states[t + 1][live], pred[live],
                                   atol=3e-5, rtol=3e-5)
        r_pred = 1.0 - pred.pow(2).mean(dim=-1)
        torch.testing.assert_close(rewards[t], r_pred, atol=3e-5, rtol=3e-5)
    # done schedule
    hor = env.horizons_i32.long().cpu()
    d_cpu = dones.cpu()
    for e in range(8):
        h = int(hor[e])
        expect = torch.tensor(
            [1.0 if (t + 1) % h == 0 else 0.0 for t in range(T)])
        torch.testing.assert_close(d_cpu[:, e], expect)


def test_v3_graphed_bitwise_matches_ungraphed(monkeypatch):
    monkeypatch.setenv("DPPO_ROLLOUT_V3", "1")
    a = make_engine(NUM_ENVS=64, MAX_EPOCH_STEPS=12, USE_GRAPHS=False)
    b = make_engine(NUM_ENVS=64, MAX_EPOCH_STEPS=12, USE_GRAPHS=True)
    ba, _ = a.rollout_once()
    bb, _ = b.rollout_once()
    assert b._v3_graph is not None or b._v3_graph_failed
    assert torch.equal(ba.states, bb.states)
    assert torch.equal(ba.actions, bb.actions)
    assert torch.equal(ba.oldflat, bb.oldflat)
    # second round replays the captured graph; engines must stay in lockstep
    sa, _ = a.train_round()
    sb, _ = b.train_round()
    assert torch.equal(a.flat_pi.flat_param, b.flat_pi.flat_param)


def test_v3_vs_fused_same_seed_trajectories(monkeypatch):
    """The fused whole-rollout kernel and the per-step GEMM (v3) engine
    share RNG slot assignments by design (rollout.hip action/noise/reset
    slots == rollout_sample/rollout_env_step slots), so the SAME seed
    must produce the same trajectory through either engine, up to the fp
    reassociation of their different GEMM implementations (VERDICT r01
    weak #6)."""
    kw = dict(NUM_ENVS=64, MAX_EPOCH_STEPS=8, USE_GRAPHS=False, SEED=21)
    monkeypatch.setenv("DPPO_ROLLOUT_V3", "0")
    torch.manual_seed(0)
    a = make_engine(**kw)
    assert a._can_fuse_rollout() and not a._can_rollout_v3()
    monkeypatch.setenv("DPPO_ROLLOUT_V3", "1")
    torch.manual_seed(0)
    b = make_engine(**kw)
    assert b._can_rollout_v3()
    torch.testing.assert_close(a.flat_pi.flat_param, b.flat_pi.flat_param)
    ba, _ = a.rollout_once()
    bb, _ = b.rollout_once()
    # same RNG draws -> same exploration decisions, noise and resets;
    # trajectories agree to GEMM-reassociation tolerance
    torch.testing.assert_close(ba.actions, bb.actions, atol=2e-3, rtol=2e-3)
    torch.testing.assert_close(ba.states, bb.states, atol=2e-3, rtol=2e-3)
    sa = v3_views(a)
    sb = v3_views(b)
    assert torch.equal(sa[5], sb[5])  # done schedules identical
    torch.testing.assert_close(sa[4], sb[4], atol=2e-3, rtol=2e-3)  # rewards


def test_v3_training_rounds(monkeypatch):
    monkeypatch.setenv("DPPO_ROLLOUT_V3", "1")
    eng = make_engine(NUM_ENVS=128, MAX_EPOCH_STEPS=16)
    p0 = eng.flat_pi.flat_param.detach().clone()
    for _ in range(3):
        stats, _ = eng.train_round()
    assert all(math.isfinite(v) for v in stats.values())
    assert not torch.equal(p0, eng.flat_pi.flat_param.detach())


def test_v3_recorded_acts_skip_is_bitwise_neutral(monkeypatch):
    """Update step 1 reusing the v3 rollout's recorded activations must be
    BITWISE identical to recomputing the forward (same kernels, same
    inputs, same parameters)."""
    monkeypatch.setenv("DPPO_ROLLOUT_V3", "1")
    kw = dict(NUM_ENVS=128, MAX_EPOCH_STEPS=16, USE_GRAPHS=False, SEED=31)
    torch.manual_seed(0)
    a = make_engine(**kw)
    torch.manual_seed(0)
    monkeypatch.setenv("DPPO_NO_SKIP1", "1")
    b = make_engine(**kw)
    for _ in range(2):
        monkeypatch.delenv("DPPO_NO_SKIP1")
        sa, _ = a.train_round()
        monkeypatch.setenv("DPPO_NO_SKIP1", "1")
        sb, _ = b.train_round()
    torch.cuda.synchronize()
    assert torch.equal(a.flat_pi.flat_param, b.flat_pi.flat_param)


def test_env_fused_epilogue_matches_default(monkeypatch):
    """The DPPO_ENV_FUSED=1 path (env transition fused into the G-GEMM
    epilogue, gemm_env_step) shares the default path's math and RNG slots
    exactly — same seed must give bitwise-equal trajectories."""
    monkeypatch.setenv("DPPO_ROLLOUT_V3", "1")
    kw = dict(NUM_ENVS=64, MAX_EPOCH_STEPS=12, USE_GRAPHS=False, SEED=41)
    torch.manual_seed(0)
    a = make_engine(**kw)
    torch.manual_seed(0)
    b = make_engine(**kw)
    ba, _ = a.rollout_once()           # default split env path
    monkeypatch.setenv("DPPO_ENV_FUSED", "1")
    bb, _ = b.rollout_once()           # fused epilogue path
    monkeypatch.delenv("DPPO_ENV_FUSED")
    assert torch.equal(ba.states, bb.states)
    assert torch.equal(ba.actions, bb.actions)
    sa, sb = v3_views(a), v3_views(b)
    # Trajectories (states/actions/dones) are bitwise identical — the RNG
    # slots and transition math match exactly.  Rewards alone are compared
    # with a tolerance: the fused path accumulates the per-env reward sum
    # as per-panel partials combined by env_finish2, a reassociation of
    # env_finish's single-kernel sum (last-ulp differences only).
    torch.testing.assert_close(sa[4], sb[4], rtol=1e-6, atol=1e-6)
    assert torch.equal(sa[5], sb[5])  # dones
