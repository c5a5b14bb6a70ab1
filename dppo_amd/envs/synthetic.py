"""Batched synthetic environments, resident on the training device.

The reference steps one real gym env per worker thread with one
sess.run per env step (reference Worker.py:10,49-50 — launch overhead
dominates).  The MI355X rebuild keeps E envs per rank as
device tensors and steps all of them with batched ops (or the fused HIP
rollout kernel), so one launch serves the whole batch and rollouts never
leave the GPU.

There is no network access for real simulators, so dynamics are synthetic
(BASELINE.json: "synthetic Humanoid-shaped observations/actions"):
deterministic-per-seed diagonal + low-rank linear dynamics with a tanh
squash, dense bounded rewards, and staggered fixed per-env horizons so
done flags and resets exercise the GAE masking.  Every op is
capture-safe (no host control flow) for hipGraph capture.

    x' = tanh(x*d + (x @ V) @ U + a_in + sigma*noise)
    r  = 1 - mean(x'^2)          in [0, 1]
    done at per-env horizon; auto-reset on done.
"""

from __future__ import annotations

import math


import torch

from .. import spaces
from ..config import DPPOConfig, game_spaces


class BatchedSyntheticEnv:
    """E parallel synthetic envs on one device.

    API mirrors a vectorized gym env: reset() -> obs[E, obs_dim];
    step(actions) -> (obs, reward[E], done[E], info) with auto-reset.
    """

    RANK = 16          # low-rank dynamics width (keeps env cost << policy cost)
    NOISE = 0.05

    def __init__(
        self,
        obs_space: spaces.Box,
        act_space,
        num_envs: int,
        device: str = "cpu",
        seed: int = 0,
        horizon: int = 64,
        dtype: torch.dtype = torch.float32,
    ):
        self.observation_space = obs_space
        self.action_space = act_space
        self.num_envs = num_envs
        self.device = torch.device(device)
        self.horizon = horizon
        self.dtype = dtype
        obs_dim = obs_space.shape[0]
        self.obs_dim = obs_dim

        gen = torch.Generator(device="cpu").manual_seed(seed)

        def randn(*shape):
            return torch.randn(*shape, generator=gen, dtype=torch.float32).to(
                device=self.device, dtype=dtype
            )

        r = min(self.RANK, obs_dim)
        # Spectral-radius-controlled dynamics: decay diag + small low-rank mix.
        self.d = (0.9 + 0.05 * torch.rand(obs_dim, generator=gen)).to(self.device, dtype)
        self.V = randn(obs_dim, r) * (0.3 / math.sqrt(obs_dim))
        self.U = randn(r, obs_dim) * (0.3 / math.sqrt(r))
        if isinstance(act_space, spaces.Discrete):
            self.B = randn(act_space.n, obs_dim) * 0.3
            self._kind = "discrete"
        elif isinstance(act_space, spaces.Box):
            self.B = randn(act_space.shape[0], obs_dim) * (
                0.3 / math.sqrt(act_space.shape[0])
            )
            self._kind = "box"
        elif isinstance(act_space, spaces.MultiDiscrete):
            # one embedding row per (component, choice): a_in sums the
            # selected rows, like the Discrete path per component
            nvec = [int(n) for n in act_space.nvec]
            self.B = randn(sum(nvec), obs_dim) * (0.3 / math.sqrt(len(nvec)))
            off = torch.tensor(
                [0] + list(torch.tensor(nvec).cumsum(0)[:-1]),
                dtype=torch.long)
            self._md_offsets = off.to(self.device)
            self._kind = "multidiscrete"
        elif isinstance(act_space, spaces.MultiBinary):
            self.B = randn(act_space.n, obs_dim) * (
                0.3 / math.sqrt(act_space.n))
            self._kind = "multibinary"
        else:
            raise NotImplementedError(f"synthetic env for {act_space!r}")
        self._discrete = self._kind == "discrete"

        # Staggered fixed horizons in [horizon//2, 3*horizon//2) so resets
        # spread across steps and every rollout sees some done flags.
        e = torch.arange(num_envs)
        span = max(horizon, 2)
        self.horizons = (horizon // 2 + (e * 2654435761 % span)).to(self.device)
        self.horizons = torch.clamp(self.horizons, min=2)
        self.horizons_i32 = self.horizons.to(torch.int32).contiguous()

        # Transposed parameter layouts for the fused HIP rollout kernel
        # (per-lane contiguous streams in the env phase; rollout.hip).
        self.Vt = self.V.t().contiguous()   # [r, D]
        self.Ut = self.U.t().contiguous()   # [D, r]
        self.Bt = self.B.t().contiguous()   # [D, A] (or [D, n] discrete)
        # single packed blob (d | Vt | U | B) so the kernel takes one
        # pointer.  Vt rows feed the per-lane xv dot products; U[r][D] and
        # B[A][D] are read row-wise with lane==d, i.e. fully coalesced.
        self.blob = torch.cat(
            [self.d.reshape(-1), self.Vt.reshape(-1), self.U.reshape(-1),
             self.B.reshape(-1)]
        ).contiguous()
        self.rank_eff = self.Vt.shape[0]

        self._noise_gen = torch.Generator(device=self.device.type).manual_seed(seed + 1)
        self.x = torch.zeros(num_envs, obs_dim, device=self.device, dtype=dtype)
        self.t = torch.zeros(num_envs, device=self.device, dtype=torch.int32)

    def seed_state(self) -> torch.Tensor:
        return 0.1 * torch.randn(
            self.num_envs, self.obs_dim,
            generator=self._noise_gen, device=self.device, dtype=self.dtype,
        )

    def reset(self) -> torch.Tensor:
        self.x = self.seed_state()
        self.t.zero_()
        return self.x

    def step(self, actions: torch.Tensor):
        if self._kind == "discrete":
            a_in = self.B.index_select(0, actions.long().reshape(-1))
        elif self._kind == "multidiscrete":
            idx = (actions.long() + self._md_offsets).reshape(-1)  # [E*K]
            a_in = self.B.index_select(0, idx).view(
                self.num_envs, -1, self.obs_dim).sum(dim=1)
        else:  # box / multibinary: linear action input
            a_in = actions.to(self.dtype) @ self.B
        noise = self.NOISE * torch.randn(
            self.num_envs, self.obs_dim,
            generator=self._noise_gen, device=self.device, dtype=self.dtype,
        )
        x = torch.tanh(self.x * self.d + (self.x @ self.V) @ self.U + a_in + noise)
        reward = 1.0 - x.pow(2).mean(dim=-1)
        self.t += 1
        done = self.t >= self.horizons
        # auto-reset (capture-safe): done envs restart from a fresh seed state
        fresh = self.seed_state()
        donef = done.unsqueeze(-1).to(self.dtype)
        self.x = x * (1.0 - donef) + fresh * donef
        self.t = torch.where(done, torch.zeros_like(self.t), self.t)
        return self.x, reward.to(self.dtype), done, {}


def make_env(cfg: DPPOConfig, device: str, seed: int) -> BatchedSyntheticEnv:
    obs_space, act_space = game_spaces(cfg.GAME)
    return BatchedSyntheticEnv(
        obs_space,
        act_space,
        num_envs=cfg.NUM_ENVS,
        device=device,
        seed=seed,
        horizon=max(cfg.MAX_EPOCH_STEPS // 2, 4),
        dtype=cfg.torch_dtype() if cfg.DTYPE == "float32" else torch.float32,
    )
