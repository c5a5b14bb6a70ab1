"""Generalized Advantage Estimation.

Semantics of the reference's host-side scan (reference Worker.py:82-92),
vectorized over E parallel envs:

    delta_t = r_t + gamma * V_{t+1} * nonterm_t - V_t
    adv_t   = delta_t + gamma * lam * nonterm_t * adv_{t+1}   (reverse scan)
    etr     = adv + V          (computed BEFORE whitening, Worker.py:91)
    adv     = (adv - mean) / std                              (Worker.py:92)

with V_T = bootstrap value of the state after the last step, masked by the
last step's done flag.  nonterm_t = 1 - done_t where done_t means "the
episode ended AT step t" (state t+1 is a reset state) — the textbook GAE
done-masking.  NOTE the reference indexes its appended done array as
done[t+1] (Worker.py:87), an off-by-one that masks one step late, and it
bootstraps with V(s_{T-1}) instead of V(s_T) (Worker.py:83 uses the loop's
stale pred_v); both are implementation bugs the survey directs us NOT to
replicate (SURVEY.md preamble) — this module implements the intended
estimator.  The whitening divide is guarded with eps (the reference's
unguarded std (Worker.py:92) NaNs on constant advantages).

The HIP fast path (ops/hip/gae_scan.hip) runs the same recurrence as a
per-env segmented reverse scan on-device with a fused mean/var reduction.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch


def gae_advantages_ref(
    rewards: torch.Tensor,      # [T, E] float32
    values: torch.Tensor,       # [T, E] float32  V(s_t)
    dones: torch.Tensor,        # [T, E] float/bool  episode ended at step t
    bootstrap_value: torch.Tensor,  # [E] V(s_T)
    gamma: float,
    lam: float,
    whiten: bool = True,
    eps: float = 1e-8,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Pure-PyTorch reference. Returns (adv, etr), both [T, E]."""
    T = rewards.shape[0]
    dones = dones.to(rewards.dtype)
    adv = torch.empty_like(rewards)
    lastgaelam = torch.zeros_like(bootstrap_value)
    nextvalue = bootstrap_value
    for t in range(T - 1, -1, -1):
        nonterm = 1.0 - dones[t]
        delta = rewards[t] + gamma * nextvalue * nonterm - values[t]
        lastgaelam = delta + gamma * lam * nonterm * lastgaelam
        adv[t] = lastgaelam
        nextvalue = values[t]
    etr = adv + values
    if whiten:
        adv = (adv - adv.mean()) / (adv.std(unbiased=False) + eps)
    return adv, etr


def gae_advantages(
    rewards: torch.Tensor,
    values: torch.Tensor,
    dones: torch.Tensor,
    bootstrap_value: torch.Tensor,
    gamma: float,
    lam: float,
    whiten: bool = True,
    eps: float = 1e-8,
    policy: str = "auto",
    adv_out: Optional[torch.Tensor] = None,
    etr_out: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """GAE with HIP dispatch on CUDA tensors.  adv_out/etr_out, when
    given, receive the results in place (stable addresses for hipGraph
    capture of the downstream update)."""
    from . import use_hip, hip_ext

    if use_hip(rewards, policy):
        ext = hip_ext()
        rewards = rewards.contiguous()
        values = values.contiguous()
        dones = dones.to(rewards.dtype).contiguous()
        bootstrap_value = bootstrap_value.contiguous()
        empty = torch.empty(0, device=rewards.device)
        adv, etr = ext.gae_scan(
            rewards, values, dones, bootstrap_value,
            float(gamma), float(lam), bool(whiten), float(eps),
            adv_out if adv_out is not None else empty,
            etr_out if etr_out is not None else empty,
        )
        return adv, etr
    return gae_advantages_ref(
        rewards, values, dones, bootstrap_value, gamma, lam, whiten, eps
    )
