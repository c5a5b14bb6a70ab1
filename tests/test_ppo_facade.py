"""PPO facade class (API parity with reference PPO.py:7-65)."""

import torch

from dppo_amd import spaces
from dppo_amd.config import DPPOConfig
from dppo_amd.envs.synthetic import BatchedSyntheticEnv
from dppo_amd.ppo import PPO
from dppo_amd.ops.ppo_loss import PPOLossCoeffs, ppo_losses_ref


def _env(discrete=False):
    obs = spaces.Box(-float("inf"), float("inf"), (6,))
    act = spaces.Discrete(3) if discrete else spaces.Box(-1, 1, (2,))
    return BatchedSyntheticEnv(obs, act, num_envs=4, device="cpu", seed=0,
                               horizon=8)


def _cfg(**kw):
    base = dict(NUM_WORKERS=2, LOG_FILE_PATH="/tmp/dppo_test_logs")
    base.update(kw)
    return DPPOConfig(**base)


def test_construction_and_surfaces():
    ppo = PPO("Worker_N0", _cfg(), _env())
    assert len(ppo.pipara) == len(ppo.oldpipara) == len(list(ppo.pi.parameters()))
    s = torch.randn(5, 6)
    a = ppo.ca(s)
    assert a.shape == (5, 2)
    v = ppo.pipredv(s)
    assert v.shape == (5,)


def test_losses_match_reference_math():
    ppo = PPO("Worker_N0", _cfg(), _env())
    B = 32
    s = torch.randn(B, 6)
    a = torch.randn(B, 2)
    adv, etr = torch.randn(B), torch.randn(B)
    out = ppo.losses(s, a, adv, etr, l_mul=0.5)
    # recompute via the reference eager path
    v, flat = ppo.pi(s)
    oldv, oldflat = ppo.oldpi(s)
    ref = ppo_losses_ref(
        ppo.pdtype.pdfromflat(flat), ppo.pdtype.pdfromflat(oldflat),
        v, oldv, a, adv, etr,
        PPOLossCoeffs(ppo.cfg.CLIP_PARAM * 0.5, ppo.cfg.ENTCOEFF, ppo.cfg.VCOEFF),
    )
    for k in ("policyLoss", "entropyLoss", "valueLoss", "total_loss"):
        torch.testing.assert_close(out[k], ref[k])


def test_sync_pis():
    ppo = PPO("Chief", _cfg(), _env())
    with torch.no_grad():
        for p in ppo.pi.parameters():
            p.add_(torch.randn_like(p))
    ppo.sync_pis()
    for p, op in zip(ppo.pi.parameters(), ppo.oldpi.parameters()):
        torch.testing.assert_close(p, op)


def test_average_gradients():
    """Mean per-variable over towers (PPO.py:55-65)."""
    g1 = [torch.ones(3), torch.full((2, 2), 2.0)]
    g2 = [torch.full((3,), 3.0), torch.full((2, 2), 4.0)]
    avg = PPO._average_gradients([g1, g2])
    torch.testing.assert_close(avg[0], torch.full((3,), 2.0))
    torch.testing.assert_close(avg[1], torch.full((2, 2), 3.0))


def test_gradient_and_train_step():
    env = _env()
    cfg = _cfg(LEARNING_RATE=1e-2)
    workers = [PPO(f"Worker_N{i}", cfg, env) for i in range(2)]
    chief = PPO("Chief", cfg, env, workerLists=workers)
    # align all towers (initial broadcast, main.py:48-50)
    for w in workers:
        with torch.no_grad():
            for cp, wp in zip(chief.pipara, w.pipara):
                wp.copy_(cp)
        w.sync_pis()

    B = 16
    towers = []
    for w in workers:
        s = torch.randn(B, 6)
        a = torch.randn(B, 2)
        towers.append(w.gradient(s, a, torch.randn(B), torch.randn(B), l_mul=1.0))
    p0 = [p.detach().clone() for p in chief.pipara]
    chief.train(towers, l_mul=1.0)
    moved = any(
        not torch.allclose(p0[i], chief.pipara[i].detach())
        for i in range(len(p0))
    )
    assert moved


def test_discrete_spaces_supported():
    ppo = PPO("Worker_N0", _cfg(), _env(discrete=True))
    s = torch.randn(4, 6)
    a = ppo.ca(s)
    assert a.dtype == torch.int64
    out = ppo.losses(s, a, torch.randn(4), torch.randn(4))
    assert torch.isfinite(out["total_loss"])
