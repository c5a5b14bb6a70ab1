"""Fused flat-bucket Adam (K10 in SURVEY.md §2.4).

One kernel launch updates the whole parameter set (param, grad, and both
moments are single flat fp32 buffers).  Update rule matches
torch.optim.Adam exactly (bias-corrected moments, eps after the sqrt) so
a run can switch implementations without drift and replicas stay
bit-identical across ranks.  state_dict() speaks torch.optim.Adam's
format for checkpoint compatibility (checkpoint.py).
"""

from __future__ import annotations

from typing import Dict, List

import torch


class FusedFlatAdam:
    def __init__(
        self,
        flat_param: torch.Tensor,
        flat_grad: torch.Tensor,
        lr: float,
        betas=(0.9, 0.999),
        eps: float = 1e-8,
    ):
        from . import require_hip_ext

        self._ext = require_hip_ext()
        self.flat_param = flat_param
        self.flat_grad = flat_grad
        self.exp_avg = torch.zeros_like(flat_param.data)
        self.exp_avg_sq = torch.zeros_like(flat_param.data)
        self.betas = tuple(betas)
        self.eps = eps
        self.param_groups: List[Dict] = [
            {"lr": lr, "betas": self.betas, "eps": eps, "params": [flat_param]}
        ]
        # device-resident step/lr/bias-correction state: kernel args are
        # frozen under hipGraph replay, so the annealed lr and the growing
        # step count must come from device memory (adam_step_dev).
        dev = flat_param.device
        self.step_dev = torch.zeros(1, device=dev, dtype=torch.int32)
        self.lr_dev = torch.tensor([lr], device=dev, dtype=torch.float32)
        self.coef = torch.zeros(3, device=dev, dtype=torch.float32)

    @property
    def step_count(self) -> int:
        return int(self.step_dev.item())

    @step_count.setter
    def step_count(self, v: int) -> None:
        self.step_dev.fill_(int(v))

    @torch.no_grad()
    def step(self) -> None:
        self.lr_dev.fill_(float(self.param_groups[0]["lr"]))
        self.step_captured()

    @torch.no_grad()
    def step_captured(self) -> None:
        """Capture-safe step: no host->device scalar traffic; the caller
        refreshes lr_dev outside the captured region."""
        g = self.param_groups[0]
        self._ext.adam_step_dev(
            self.flat_param.data,
            self.flat_grad,
            self.exp_avg,
            self.exp_avg_sq,
            self.step_dev,
            self.lr_dev,
            self.coef,
            float(g["betas"][0]),
            float(g["betas"][1]),
            float(g["eps"]),
        )

    def zero_grad(self, set_to_none: bool = False) -> None:
        self.flat_grad.zero_()

    # -- torch.optim.Adam-compatible (de)serialization ------------------
    def state_dict(self) -> Dict:
        return {
            "state": {
                0: {
                    "step": torch.tensor(float(self.step_count)),
                    "exp_avg": self.exp_avg,
                    "exp_avg_sq": self.exp_avg_sq,
                }
            },
            "param_groups": [
                {k: v for k, v in self.param_groups[0].items() if k != "params"}
                | {"params": [0]}
            ],
        }

    def load_state_dict(self, sd: Dict) -> None:
        st = sd["state"].get(0, {})
        if "step" in st:
            self.step_count = int(float(st["step"]))
        if "exp_avg" in st:
            self.exp_avg.copy_(st["exp_avg"].to(self.exp_avg.device))
        if "exp_avg_sq" in st:
            self.exp_avg_sq.copy_(st["exp_avg_sq"].to(self.exp_avg_sq.device))
        pg = sd.get("param_groups")
        if pg:
            for k in ("lr", "betas", "eps"):
                if k in pg[0]:
                    self.param_groups[0][k] = pg[0][k]
