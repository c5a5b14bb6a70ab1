"""PPO clipped-surrogate loss (reference PPO.py:29-40, exact formulation).

    ratio        = exp(logp_pi(a) - logp_oldpi(a))
    surr1        = ratio * adv
    surr2        = clip(ratio, 1 - eps, 1 + eps) * adv
    policyLoss   = -mean(min(surr1, surr2))
    entropyLoss  = -ENTCOEFF * mean(entropy(pi))
    vfloss1      = (vpred - etr)^2
    vpredclipped = oldvpred + clip(vpred - oldvpred, -eps, eps)
    vfloss2      = (vpredclipped - etr)^2
    valueLoss    = VCOEFF * mean(max(vfloss1, vfloss2))
    total_loss   = policyLoss + entropyLoss + valueLoss

where eps = CLIP_PARAM * l_mul: the l_mul anneal multiplier scales BOTH
the Adam learning rate and the clip range (reference PPO.py:19-20).

The eager path below is the autograd reference; the fused HIP kernels
(ops/hip/ppo_loss.hip for DiagGaussian, ops/hip/cat_loss.hip for
Categorical — the reference's default family, distributions.py:124-159)
each compute the whole thing — logp for pi and oldpi, ratio, both clips,
entropy, and the three block-reduced means — in one forward kernel and
one analytic backward kernel, and are tested against this path to
tolerance.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict

import torch

from ..distributions import Pd, DiagGaussianPd, CategoricalPd


@dataclass
class PPOLossCoeffs:
    clip_param: float   # CLIP_PARAM * l_mul (already annealed)
    entcoeff: float
    vcoeff: float


def ppo_losses_ref(
    pd: Pd,
    oldpd: Pd,
    vpred: torch.Tensor,      # [B]
    oldvpred: torch.Tensor,   # [B]
    actions: torch.Tensor,    # [B] long or [B, A] float
    adv: torch.Tensor,        # [B] whitened advantages
    etr: torch.Tensor,        # [B] estimated returns
    coeffs: PPOLossCoeffs,
) -> Dict[str, torch.Tensor]:
    """Eager autograd path. Returns the 4 scalar losses of PPO.py:29-40."""
    eps = coeffs.clip_param
    ent = pd.entropy().mean()
    ratio = torch.exp(pd.logp(actions) - oldpd.logp(actions))
    surr1 = ratio * adv
    surr2 = torch.clamp(ratio, 1.0 - eps, 1.0 + eps) * adv
    policy_loss = -torch.min(surr1, surr2).mean()
    entropy_loss = (-coeffs.entcoeff) * ent
    vfloss1 = (vpred - etr) ** 2
    vpredclipped = oldvpred + torch.clamp(vpred - oldvpred, -eps, eps)
    vfloss2 = (vpredclipped - etr) ** 2
    value_loss = coeffs.vcoeff * torch.max(vfloss1, vfloss2).mean()
    total = policy_loss + entropy_loss + value_loss
    return {
        "policyLoss": policy_loss,
        "entropyLoss": entropy_loss,
        "valueLoss": value_loss,
        "total_loss": total,
    }


class _FusedPPOLossGaussian(torch.autograd.Function):
    """Fused DiagGaussian PPO loss: HIP forward + analytic HIP backward.

    Inputs are the raw network outputs (pdflat of pi, detached pdflat of
    oldpi, vpred, detached oldvpred); gradients flow to pdflat_pi and
    vpred only, matching compute_gradients(total_loss, pipara)
    (reference PPO.py:46 — oldpi params receive no gradient).
    """

    @staticmethod
    def forward(ctx, pdflat_pi, pdflat_old, vpred, oldvpred, actions, adv, etr,
                clip_param, entcoeff, vcoeff):
        from . import require_hip_ext

        ext = require_hip_ext()
        out = ext.ppo_loss_gauss_fwd(
            pdflat_pi.contiguous(), pdflat_old.contiguous(),
            vpred.contiguous(), oldvpred.contiguous(),
            actions.contiguous(), adv.contiguous(), etr.contiguous(),
            float(clip_param), float(entcoeff), float(vcoeff),
        )
        # out: losses[4] (policy, entropy, value, total)
        ctx.save_for_backward(pdflat_pi, pdflat_old, vpred, oldvpred,
                              actions, adv, etr)
        ctx.coeffs = (float(clip_param), float(entcoeff), float(vcoeff))
        return out

    @staticmethod
    def backward(ctx, grad_losses):
        from . import require_hip_ext

        ext = require_hip_ext()
        pdflat_pi, pdflat_old, vpred, oldvpred, actions, adv, etr = ctx.saved_tensors
        clip_param, entcoeff, vcoeff = ctx.coeffs
        # Upstream gradient on total_loss (index 3); the three component
        # losses are observational outputs.
        g_total = grad_losses[3]
        g_pdflat, g_v = ext.ppo_loss_gauss_bwd(
            pdflat_pi, pdflat_old, vpred, oldvpred, actions, adv, etr,
            clip_param, entcoeff, vcoeff, g_total,
        )
        return (g_pdflat, None, g_v) + (None,) * 7


class _FusedPPOLossCategorical(torch.autograd.Function):
    """Fused Categorical PPO loss: HIP forward + analytic HIP backward
    (ops/hip/cat_loss.hip — logsumexp CE logp, shifted-logit entropy per
    reference distributions.py:131-153).  Gradients flow to the pi logits
    and vpred only, like the Gaussian variant."""

    @staticmethod
    def forward(ctx, logits_pi, logits_old, vpred, oldvpred, actions, adv,
                etr, clip_param, entcoeff, vcoeff):
        from . import require_hip_ext

        ext = require_hip_ext()
        out = ext.ppo_loss_cat_fwd(
            logits_pi.contiguous(), logits_old.contiguous(),
            vpred.contiguous(), oldvpred.contiguous(),
            actions.contiguous(), adv.contiguous(), etr.contiguous(),
            float(clip_param), float(entcoeff), float(vcoeff),
        )
        ctx.save_for_backward(logits_pi, logits_old, vpred, oldvpred,
                              actions, adv, etr)
        ctx.coeffs = (float(clip_param), float(entcoeff), float(vcoeff))
        return out

    @staticmethod
    def backward(ctx, grad_losses):
        from . import require_hip_ext

        ext = require_hip_ext()
        (logits_pi, logits_old, vpred, oldvpred, actions, adv,
         etr) = ctx.saved_tensors
        clip_param, entcoeff, vcoeff = ctx.coeffs
        g_total = grad_losses[3]
        g_logits, g_v = ext.ppo_loss_cat_bwd(
            logits_pi, logits_old, vpred, oldvpred, actions, adv, etr,
            clip_param, entcoeff, vcoeff, g_total,
        )
        return (g_logits, None, g_v) + (None,) * 7


def _loss_dict(losses: torch.Tensor) -> Dict[str, torch.Tensor]:
    return {
        "policyLoss": losses[0],
        "entropyLoss": losses[1],
        "valueLoss": losses[2],
        "total_loss": losses[3],
    }


def ppo_losses(
    pd: Pd,
    oldpd: Pd,
    vpred: torch.Tensor,
    oldvpred: torch.Tensor,
    actions: torch.Tensor,
    adv: torch.Tensor,
    etr: torch.Tensor,
    coeffs: PPOLossCoeffs,
    policy: str = "auto",
) -> Dict[str, torch.Tensor]:
    """PPO losses with fused-HIP dispatch for DiagGaussian and Categorical
    policies (the other families run the eager reference)."""
    from . import use_hip

    if (
        isinstance(pd, DiagGaussianPd)
        and isinstance(oldpd, DiagGaussianPd)
        and use_hip(vpred, policy)
    ):
        return _loss_dict(_FusedPPOLossGaussian.apply(
            pd.flatparam(), oldpd.flatparam().detach(),
            vpred, oldvpred.detach(), actions, adv, etr,
            coeffs.clip_param, coeffs.entcoeff, coeffs.vcoeff,
        ))
    if (
        isinstance(pd, CategoricalPd)
        and isinstance(oldpd, CategoricalPd)
        and pd.flatparam().shape[-1] <= 64  # one wave covers the columns
        and use_hip(vpred, policy)
    ):
        return _loss_dict(_FusedPPOLossCategorical.apply(
            pd.flatparam(), oldpd.flatparam().detach(),
            vpred, oldvpred.detach(), actions.long(), adv, etr,
            coeffs.clip_param, coeffs.entcoeff, coeffs.vcoeff,
        ))
    return ppo_losses_ref(pd, oldpd, vpred, oldvpred, actions, adv, etr, coeffs)
