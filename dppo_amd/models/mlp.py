"""Policy + value MLP (the reference's Model.FC, rebuilt).

The reference network (reference Model.py:7-18) is a single shared hidden
dense layer (width 16, relu, normc init std 0.01) with two heads: a scalar
value head `predv` and a distribution-parameter head `logits` of
pdtype.param_shape().  The reference inserts a spurious middle dim of 1
via expand_dims (Model.py:11) which forces `action[0][0]` indexing at call
sites — dropped here.

The rebuild generalizes widths/activation (HIDDEN_SIZES/ACTIVATION config)
because the MI355X benchmark configs (BASELINE.json: Humanoid-shaped,
Wide-4096) need deeper/wider trunks; the defaults reproduce the reference.
"""

from __future__ import annotations

from typing import Sequence, Tuple

import torch
import torch.nn as nn

from ..distributions import make_pdtype, PdType


def normc_init_(weight: torch.Tensor, std: float = 1.0) -> torch.Tensor:
    """Column-normalized Gaussian init (reference Others/tf_util.py:286-291).

    TF dense kernels are [in, out] with each output column normalized over
    the input axis; torch nn.Linear weights are [out, in], so each ROW is
    normalized over in_features.
    """
    with torch.no_grad():
        out = torch.randn_like(weight)
        out *= std / out.pow(2).sum(dim=1, keepdim=True).sqrt()
        weight.copy_(out)
    return weight


_ACTIVATIONS = {"relu": torch.relu, "tanh": torch.tanh}


class PolicyValueMLP(nn.Module):
    """Shared-trunk MLP: obs -> hidden stack -> (value, pd-params).

    forward(s[B, obs_dim]) -> (predv[B], pdflat[B, param_dim]).
    """

    def __init__(
        self,
        obs_dim: int,
        action_space,
        hidden_sizes: Sequence[int] = (16,),
        activation: str = "relu",
        init_std: float = 0.01,
    ):
        super().__init__()
        self.obs_dim = obs_dim
        self.pdtype: PdType = make_pdtype(action_space)
        self.hidden_sizes = tuple(hidden_sizes)
        self.activation = activation
        self._act = _ACTIVATIONS[activation]

        dims = [obs_dim, *self.hidden_sizes]
        self.hidden = nn.ModuleList(
            nn.Linear(dims[i], dims[i + 1]) for i in range(len(dims) - 1)
        )
        last = dims[-1]
        self.vf = nn.Linear(last, 1)                       # predv head (Model.py:13)
        self.pi = nn.Linear(last, self.pdtype.param_shape()[0])  # logits head (Model.py:14)

        for m in [*self.hidden, self.vf, self.pi]:
            normc_init_(m.weight, init_std)
            nn.init.zeros_(m.bias)

    def forward(self, s: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        h = s
        for layer in self.hidden:
            h = self._act(layer(h))
        return self.vf(h).squeeze(-1), self.pi(h)

    def pd(self, s: torch.Tensor):
        """Distribution over actions at states s (plus values)."""
        v, flat = self.forward(s)
        return v, self.pdtype.pdfromflat(flat)


class Model:
    """API-parity shim for the reference's Model class (Model.py:6-18).

    `FC(scope, obs_space, action_space, ...)` returns
    (net, pdtype, list(parameters)) — the rebuild's analog of the
    reference's (predv, pd, para) triple: the net computes predv and the
    pd params for any batch of states, and `para` is the trainable set
    under the scope.
    """

    def FC(
        self,
        scope: str,
        obs_space,
        action_space,
        hidden_sizes: Sequence[int] = (16,),
        activation: str = "relu",
        init_std: float = 0.01,
    ):
        net = PolicyValueMLP(
            obs_dim=obs_space.shape[0],
            action_space=action_space,
            hidden_sizes=hidden_sizes,
            activation=activation,
            init_std=init_std,
        )
        net.scope = scope
        return net, net.pdtype, list(net.parameters())
