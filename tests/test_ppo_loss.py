"""PPO loss tests: formula identities of reference PPO.py:29-40 and
gradient-routing properties."""

import torch

from dppo_amd.distributions import CategoricalPdType, DiagGaussianPdType
from dppo_amd.ops.ppo_loss import PPOLossCoeffs, ppo_losses_ref


def _gauss_setup(B=64, A=4, seed=0):
    g = torch.Generator().manual_seed(seed)
    flat = torch.randn(B, 2 * A, generator=g, requires_grad=True)
    oldflat = (flat.detach() + 0.1 * torch.randn(B, 2 * A, generator=g))
    v = torch.randn(B, generator=g, requires_grad=True)
    oldv = v.detach() + 0.3 * torch.randn(B, generator=g)
    pdt = DiagGaussianPdType(A)
    actions = pdt.pdfromflat(oldflat).sample()
    adv = torch.randn(B, generator=g)
    etr = torch.randn(B, generator=g)
    return pdt, flat, oldflat, v, oldv, actions, adv, etr


def test_manual_formula_identity():
    """Recompute every term by hand and compare."""
    pdt, flat, oldflat, v, oldv, a, adv, etr = _gauss_setup()
    eps, entc, vc = 0.2, 0.01, 0.5
    pd, oldpd = pdt.pdfromflat(flat), pdt.pdfromflat(oldflat)
    out = ppo_losses_ref(pd, oldpd, v, oldv, a, adv, etr,
                         PPOLossCoeffs(eps, entc, vc))

    ratio = torch.exp(pd.logp(a) - oldpd.logp(a))
    surr1 = ratio * adv
    surr2 = torch.clamp(ratio, 1 - eps, 1 + eps) * adv
    pol = -torch.min(surr1, surr2).mean()
    entl = -entc * pd.entropy().mean()
    vf1 = (v - etr) ** 2
    vclip = oldv + torch.clamp(v - oldv, -eps, eps)
    vl = vc * torch.max(vf1, (vclip - etr) ** 2).mean()
    torch.testing.assert_close(out["policyLoss"], pol)
    torch.testing.assert_close(out["entropyLoss"], entl)
    torch.testing.assert_close(out["valueLoss"], vl)
    torch.testing.assert_close(out["total_loss"], pol + entl + vl)


def test_zero_update_ratio_is_one():
    """pi == oldpi => ratio == 1 => policyLoss == -mean(adv)."""
    pdt, flat, _, v, oldv, a, adv, etr = _gauss_setup()
    pd = pdt.pdfromflat(flat)
    oldpd = pdt.pdfromflat(flat.detach())
    out = ppo_losses_ref(pd, oldpd, v, oldv, a, adv, etr,
                         PPOLossCoeffs(0.2, 0.0, 0.0))
    torch.testing.assert_close(out["policyLoss"], -adv.mean())
    torch.testing.assert_close(out["total_loss"], -adv.mean())


def test_clip_is_active():
    """With a huge policy shift, the clipped surrogate must bound the loss."""
    pdt = DiagGaussianPdType(2)
    B = 32
    old = torch.zeros(B, 4)
    new = torch.cat([torch.full((B, 2), 3.0), torch.zeros(B, 2)], -1)
    a = torch.zeros(B, 2)
    adv = -torch.ones(B)  # negative adv: unclipped ratio->0 term would vanish
    etr = torch.zeros(B)
    v = torch.zeros(B, requires_grad=True)
    out = ppo_losses_ref(pdt.pdfromflat(new), pdt.pdfromflat(old), v, v.detach(),
                         a, adv, etr, PPOLossCoeffs(0.2, 0.0, 0.0))
    # ratio = exp(logp_new - logp_old) = exp(-4.5)<<1-eps; min(surr1,surr2)
    # with adv=-1: surr1=-ratio (≈0), surr2=-(1-eps); min=-(1-eps)=-0.8
    torch.testing.assert_close(out["policyLoss"], torch.tensor(0.8))


def test_value_clip_max():
    """valueLoss takes the elementwise MAX of clipped/unclipped (PPO.py:36-39)."""
    B = 4
    v = torch.tensor([2.0, -2.0, 0.1, 0.0], requires_grad=True)
    oldv = torch.zeros(B)
    etr = torch.zeros(B)
    pdt = DiagGaussianPdType(1)
    flat = torch.zeros(B, 2)
    out = ppo_losses_ref(pdt.pdfromflat(flat), pdt.pdfromflat(flat), v, oldv,
                         torch.zeros(B, 1), torch.zeros(B), etr,
                         PPOLossCoeffs(0.5, 0.0, 1.0))
    # clipped v = clamp to +-0.5 -> vf2 = 0.25 for |v|>0.5; vf1 = v^2 larger
    expect = torch.tensor([4.0, 4.0, 0.01, 0.0]).mean()
    torch.testing.assert_close(out["valueLoss"], expect)


def test_grads_flow_only_to_pi():
    pdt, flat, oldflat, v, oldv, a, adv, etr = _gauss_setup()
    oldflat = oldflat.requires_grad_(True)
    oldv = oldv.requires_grad_(True)
    pd, oldpd = pdt.pdfromflat(flat), pdt.pdfromflat(oldflat.detach())
    out = ppo_losses_ref(pd, oldpd, v, oldv.detach(), a, adv, etr,
                         PPOLossCoeffs(0.2, 0.01, 0.5))
    out["total_loss"].backward()
    assert flat.grad is not None and torch.isfinite(flat.grad).all()
    assert v.grad is not None and torch.isfinite(v.grad).all()
    assert oldflat.grad is None
    assert oldv.grad is None


def test_categorical_policy_loss():
    g = torch.Generator().manual_seed(3)
    B, K = 48, 5
    logits = torch.randn(B, K, generator=g, requires_grad=True)
    oldlogits = logits.detach() + 0.2 * torch.randn(B, K, generator=g)
    pdt = CategoricalPdType(K)
    a = pdt.pdfromflat(oldlogits).sample()
    v = torch.randn(B, generator=g, requires_grad=True)
    oldv = v.detach()
    adv, etr = torch.randn(B, generator=g), torch.randn(B, generator=g)
    out = ppo_losses_ref(pdt.pdfromflat(logits), pdt.pdfromflat(oldlogits),
                         v, oldv, a, adv, etr, PPOLossCoeffs(0.2, 0.01, 0.5))
    out["total_loss"].backward()
    assert torch.isfinite(logits.grad).all()
    for k in ("policyLoss", "entropyLoss", "valueLoss", "total_loss"):
        assert torch.isfinite(out[k])
