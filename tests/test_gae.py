"""GAE tests vs an independent numpy scan (the reference's own loop shape,
Worker.py:84-90, with the intended done-masking — SURVEY.md preamble)."""

import numpy as np
import torch

from dppo_amd.ops.gae import gae_advantages_ref


def numpy_gae(rewards, values, dones, boot, gamma, lam):
    T, E = rewards.shape
    adv = np.zeros((T, E), np.float64)
    lastgaelam = np.zeros(E, np.float64)
    nextv = boot.astype(np.float64)
    for t in reversed(range(T)):
        nonterm = 1.0 - dones[t]
        delta = rewards[t] + gamma * nextv * nonterm - values[t]
        lastgaelam = delta + gamma * lam * nonterm * lastgaelam
        adv[t] = lastgaelam
        nextv = values[t]
    etr = adv + values
    return adv, etr


def test_matches_numpy_scan():
    T, E = 37, 5
    rng = np.random.default_rng(0)
    r = rng.normal(size=(T, E)).astype(np.float32)
    v = rng.normal(size=(T, E)).astype(np.float32)
    d = (rng.random((T, E)) < 0.1).astype(np.float32)
    boot = rng.normal(size=E).astype(np.float32)
    adv_np, etr_np = numpy_gae(r, v, d, boot, 0.99, 0.95)

    adv, etr = gae_advantages_ref(
        torch.from_numpy(r), torch.from_numpy(v), torch.from_numpy(d),
        torch.from_numpy(boot), 0.99, 0.95, whiten=False,
    )
    np.testing.assert_allclose(adv.numpy(), adv_np, rtol=1e-5, atol=1e-5)
    np.testing.assert_allclose(etr.numpy(), etr_np, rtol=1e-5, atol=1e-5)


def test_done_masks_bootstrap():
    """A done at step t must cut both the bootstrap and the recursion."""
    r = torch.tensor([[1.0], [1.0]])
    v = torch.tensor([[0.5], [0.7]])
    d = torch.tensor([[1.0], [0.0]])  # episode ends at t=0
    boot = torch.tensor([10.0])
    adv, etr = gae_advantages_ref(r, v, d, boot, 0.9, 0.8, whiten=False)
    # t=1: delta = 1 + 0.9*10 - 0.7 = 9.3 ; adv1 = 9.3
    # t=0 (done): delta = 1 - 0.5 = 0.5 ; adv0 = 0.5 (no leak from adv1)
    assert torch.allclose(adv[1], torch.tensor([9.3]), atol=1e-6)
    assert torch.allclose(adv[0], torch.tensor([0.5]), atol=1e-6)


def test_etr_uses_prewhitened_adv():
    """etr = adv_raw + v BEFORE whitening (Worker.py:91)."""
    T, E = 16, 3
    r, v = torch.randn(T, E), torch.randn(T, E)
    d = torch.zeros(T, E)
    boot = torch.randn(E)
    adv_w, etr = gae_advantages_ref(r, v, d, boot, 0.99, 0.95, whiten=True)
    adv_raw, etr2 = gae_advantages_ref(r, v, d, boot, 0.99, 0.95, whiten=False)
    torch.testing.assert_close(etr, etr2)
    torch.testing.assert_close(etr, adv_raw + v)
    # whitened advantages are standardized
    assert abs(float(adv_w.mean())) < 1e-5
    assert abs(float(adv_w.std(unbiased=False)) - 1.0) < 1e-4


def test_whiten_guard_no_nan():
    """Constant advantages must not NaN (the reference's unguarded
    divide, Worker.py:92, would)."""
    T, E = 4, 2
    r = torch.ones(T, E)
    v = torch.zeros(T, E)
    d = torch.ones(T, E)  # every step terminal -> adv = const 1
    boot = torch.zeros(E)
    adv, _ = gae_advantages_ref(r, v, d, boot, 0.99, 0.95, whiten=True)
    assert torch.isfinite(adv).all()
