"""PPO — API-parity facade (reference PPO.py:7-65).

The reference's PPO class builds the pi/oldpi towers, the loss graph, the
per-tower gradients and the chief's averaged-gradient train op inside one
tf.Graph.  In the MI355X rebuild those live in eager PyTorch (and HIP
kernels on the training hot path, dppo_amd.trainer); this facade offers
the same construction signature and the reference's tensor surface as
methods, so code written against the reference's PPO maps 1:1:

  reference tensor        here
  ----------------        ----------------------------------
  ca                      .ca(s)            sampled action (PPO.py:23)
  pipredv                 .pipredv(s)       value head
  policyLoss/valueLoss/
  entropyLoss/total_loss  .losses(...)      dict, exact PPO.py:29-40 math
  gradient                .gradient(...)    per-tower grads (PPO.py:46)
  sync_pis                .sync_pis()       oldpi <- pi (PPO.py:47)
  train                   .train(...)       averaged-grad Adam step
                                            (chief branch, PPO.py:48-53)
  l_mul                   l_mul argument    joint lr+clip anneal (PPO.py:18-20)
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Sequence

import torch

from .config import DPPOConfig
from .models.mlp import PolicyValueMLP
from .ops import ppo_losses, PPOLossCoeffs


class PPO:
    def __init__(
        self,
        scope: str,
        parameter_dict: Any,
        env: Any,
        workerLists: Optional[Sequence["PPO"]] = None,
        device: str = "cpu",
    ):
        if isinstance(parameter_dict, DPPOConfig):
            cfg = parameter_dict
        else:
            cfg = DPPOConfig.from_dict(dict(parameter_dict))
        self.cfg = cfg
        self.scope = scope
        self.device = torch.device(device)
        self.workerLists = list(workerLists) if workerLists else None

        obs_space = env.observation_space
        act_space = env.action_space

        def build() -> PolicyValueMLP:
            return PolicyValueMLP(
                obs_dim=obs_space.shape[0],
                action_space=act_space,
                hidden_sizes=cfg.HIDDEN_SIZES,
                activation=cfg.ACTIVATION,
                init_std=cfg.INIT_STD,
            ).to(self.device)

        # pi + oldpi towers (PPO.py:21-22); scope-qualified param access
        self.pi = build()
        self.oldpi = build()
        self.sync_pis()
        self.pipara: List[torch.nn.Parameter] = list(self.pi.parameters())
        self.oldpipara: List[torch.nn.Parameter] = list(self.oldpi.parameters())
        self.pdtype = self.pi.pdtype

        # the chief owns the only live optimizer (PPO.py:20; lr scaled by
        # l_mul at apply time)
        self.optimizer = torch.optim.Adam(self.pipara, lr=cfg.LEARNING_RATE)

    # -- acting --------------------------------------------------------
    @torch.no_grad()
    def ca(self, s: torch.Tensor) -> torch.Tensor:
        """Sampled action op (PPO.py:23)."""
        _, flat = self.pi(self._t(s))
        return self.pdtype.pdfromflat(flat).sample()

    @torch.no_grad()
    def pipredv(self, s: torch.Tensor) -> torch.Tensor:
        v, _ = self.pi(self._t(s))
        return v

    # -- losses / gradients (worker branch, PPO.py:25-47) --------------
    def losses(
        self,
        s: torch.Tensor,
        a: torch.Tensor,
        adv: torch.Tensor,
        etr: torch.Tensor,
        l_mul: float = 1.0,
    ) -> Dict[str, torch.Tensor]:
        v, flat = self.pi(self._t(s))
        with torch.no_grad():
            oldv, oldflat = self.oldpi(self._t(s))
        coeffs = PPOLossCoeffs(
            clip_param=self.cfg.CLIP_PARAM * l_mul,  # PPO.py:19
            entcoeff=self.cfg.ENTCOEFF,
            vcoeff=self.cfg.VCOEFF,
        )
        pd = self.pdtype.pdfromflat(flat)
        oldpd = self.pdtype.pdfromflat(oldflat)
        return ppo_losses(pd, oldpd, v, oldv, self._t(a), self._t(adv),
                          self._t(etr), coeffs)

    def gradient(self, s, a, adv, etr, l_mul: float = 1.0) -> List[torch.Tensor]:
        """Per-tower gradients of total_loss wrt pipara (PPO.py:46)."""
        out = self.losses(s, a, adv, etr, l_mul)
        grads = torch.autograd.grad(out["total_loss"], self.pipara)
        return [g.detach() for g in grads]

    @torch.no_grad()
    def sync_pis(self) -> None:
        """oldpi <- pi assigns (PPO.py:47)."""
        for p, oldp in zip(self.pi.parameters(), self.oldpi.parameters()):
            oldp.copy_(p)

    # -- chief branch (PPO.py:48-65) -----------------------------------
    @staticmethod
    def _average_gradients(
        tower_grads: Sequence[Sequence[torch.Tensor]],
    ) -> List[torch.Tensor]:
        """Mean per-variable over towers (PPO.py:55-65: expand_dims ->
        concat -> reduce_mean, here a stacked mean)."""
        return [
            torch.stack(list(gs), dim=0).mean(dim=0)
            for gs in zip(*tower_grads)
        ]

    def train(
        self,
        tower_grads: Sequence[Sequence[torch.Tensor]],
        l_mul: float = 1.0,
    ) -> None:
        """apply_gradients of the tower-averaged grads with lr =
        LEARNING_RATE * l_mul (PPO.py:20,53)."""
        avg = self._average_gradients(tower_grads)
        for g in self.optimizer.param_groups:
            g["lr"] = self.cfg.LEARNING_RATE * l_mul
        for p, gr in zip(self.pipara, avg):
            p.grad = gr.clone()
        self.optimizer.step()
        self.optimizer.zero_grad(set_to_none=True)

    # ------------------------------------------------------------------
    def _t(self, x) -> torch.Tensor:
        t = torch.as_tensor(x)
        if t.is_floating_point():
            t = t.to(self.device, torch.float32)
        else:
            t = t.to(self.device)
        return t
