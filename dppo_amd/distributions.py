"""Probability distributions for policy outputs (PyTorch).

Capability-parity rebuild of the reference's vendored Baselines module
(reference Others/distributions.py): the Pd / PdType class families with
the same math —

- CategoricalPd: numerically stable entropy/KL via the shifted-logit
  (logsumexp) form (reference distributions.py:139-159), Gumbel-max
  sampling (:154-156), and a neglogp written as softmax cross-entropy
  against a one-hot target so it stays second-order differentiable
  (:131-138).
- DiagGaussianPd: closed-form logp/KL/entropy (:195-203) and
  reparameterized sampling mean + std * N(0,1) (:204-205).
- MultiCategoricalPd (:161-182) and BernoulliPd (:210-229) for parity.
- make_pdtype dispatch: Box -> DiagGaussian, Discrete -> Categorical,
  MultiDiscrete -> MultiCategorical, MultiBinary -> Bernoulli (:231-243).

Everything here runs on CPU or GPU tensors; the HIP fused kernels in
dppo_amd.ops reproduce the DiagGaussian/Categorical logp+entropy math on
the training hot path and are tested against this module.
"""

from __future__ import annotations

import math
from typing import List

import torch

from . import spaces

_LOG_2PI = math.log(2.0 * math.pi)


class Pd:
    """A particular probability distribution (reference distributions.py:8-26)."""

    def flatparam(self) -> torch.Tensor:
        raise NotImplementedError

    def mode(self) -> torch.Tensor:
        raise NotImplementedError

    def neglogp(self, x: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    def kl(self, other: "Pd") -> torch.Tensor:
        raise NotImplementedError

    def entropy(self) -> torch.Tensor:
        raise NotImplementedError

    def sample(self) -> torch.Tensor:
        raise NotImplementedError

    def logp(self, x: torch.Tensor) -> torch.Tensor:
        return -self.neglogp(x)


class PdType:
    """Parametrized family of distributions (reference distributions.py:28-46)."""

    def pdclass(self):
        raise NotImplementedError

    def pdfromflat(self, flat: torch.Tensor) -> Pd:
        return self.pdclass()(flat)

    def param_shape(self) -> List[int]:
        raise NotImplementedError

    def sample_shape(self) -> List[int]:
        raise NotImplementedError

    def sample_dtype(self) -> torch.dtype:
        raise NotImplementedError

    def __eq__(self, other):
        return type(self) == type(other) and self.__dict__ == other.__dict__


# ---------------------------------------------------------------------------
# Categorical
# ---------------------------------------------------------------------------


class CategoricalPd(Pd):
    """Categorical over the last dim of `logits` (reference distributions.py:124-159)."""

    def __init__(self, logits: torch.Tensor):
        self.logits = logits

    def flatparam(self):
        return self.logits

    def mode(self):
        return torch.argmax(self.logits, dim=-1)

    def neglogp(self, x: torch.Tensor) -> torch.Tensor:
        # Softmax cross-entropy against a one-hot target, written out so it
        # is differentiable to second order (reference distributions.py:131-138
        # keeps this property on purpose). logsumexp is the stable CE form.
        x = x.long()
        if x.dim() == self.logits.dim():  # tolerate a trailing [.,1] action dim
            x = x.squeeze(-1)
        z = torch.logsumexp(self.logits, dim=-1)
        picked = torch.gather(self.logits, -1, x.unsqueeze(-1)).squeeze(-1)
        return z - picked

    def kl(self, other: "CategoricalPd") -> torch.Tensor:
        # Shifted-logit form, stable for large logits (reference :139-147).
        a0 = self.logits - self.logits.max(dim=-1, keepdim=True).values
        a1 = other.logits - other.logits.max(dim=-1, keepdim=True).values
        ea0, ea1 = torch.exp(a0), torch.exp(a1)
        z0 = ea0.sum(dim=-1, keepdim=True)
        z1 = ea1.sum(dim=-1, keepdim=True)
        p0 = ea0 / z0
        return (p0 * (a0 - torch.log(z0) - a1 + torch.log(z1))).sum(dim=-1)

    def entropy(self) -> torch.Tensor:
        a0 = self.logits - self.logits.max(dim=-1, keepdim=True).values
        ea0 = torch.exp(a0)
        z0 = ea0.sum(dim=-1, keepdim=True)
        p0 = ea0 / z0
        return (p0 * (torch.log(z0) - a0)).sum(dim=-1)

    def sample(self) -> torch.Tensor:
        # Gumbel-max (reference distributions.py:154-156): argmax(logits - log(-log U)).
        u = torch.rand_like(self.logits)
        return torch.argmax(self.logits - torch.log(-torch.log(u)), dim=-1)


class CategoricalPdType(PdType):
    def __init__(self, ncat: int):
        self.ncat = ncat

    def pdclass(self):
        return CategoricalPd

    def param_shape(self):
        return [self.ncat]

    def sample_shape(self):
        return []

    def sample_dtype(self):
        return torch.int64


# ---------------------------------------------------------------------------
# MultiCategorical
# ---------------------------------------------------------------------------


class MultiCategoricalPd(Pd):
    """Independent categoricals (reference distributions.py:161-182)."""

    def __init__(self, nvec, flat: torch.Tensor):
        self.flat = flat
        self.categoricals = [
            CategoricalPd(part)
            for part in torch.split(flat, list(nvec), dim=-1)
        ]

    def flatparam(self):
        return self.flat

    def mode(self):
        return torch.stack([p.mode() for p in self.categoricals], dim=-1)

    def neglogp(self, x: torch.Tensor) -> torch.Tensor:
        return sum(
            p.neglogp(x[..., i]) for i, p in enumerate(self.categoricals)
        )

    def kl(self, other: "MultiCategoricalPd") -> torch.Tensor:
        return sum(p.kl(q) for p, q in zip(self.categoricals, other.categoricals))

    def entropy(self) -> torch.Tensor:
        return sum(p.entropy() for p in self.categoricals)

    def sample(self) -> torch.Tensor:
        return torch.stack([p.sample() for p in self.categoricals], dim=-1)


class MultiCategoricalPdType(PdType):
    def __init__(self, nvec):
        self.nvec = tuple(int(n) for n in nvec)

    def pdclass(self):
        return MultiCategoricalPd

    def pdfromflat(self, flat):
        return MultiCategoricalPd(self.nvec, flat)

    def param_shape(self):
        return [int(sum(self.nvec))]

    def sample_shape(self):
        return [len(self.nvec)]

    def sample_dtype(self):
        return torch.int64


# ---------------------------------------------------------------------------
# DiagGaussian
# ---------------------------------------------------------------------------


class DiagGaussianPd(Pd):
    """Diagonal Gaussian; flat = concat(mean, logstd) on the last dim
    (reference distributions.py:184-208)."""

    def __init__(self, flat: torch.Tensor):
        self.flat = flat
        self.mean, self.logstd = torch.chunk(flat, 2, dim=-1)
        self.std = torch.exp(self.logstd)

    def flatparam(self):
        return self.flat

    def mode(self):
        return self.mean

    def neglogp(self, x: torch.Tensor) -> torch.Tensor:
        # 0.5*sum(((x-mu)/std)^2) + 0.5*log(2*pi)*d + sum(logstd)
        # (reference distributions.py:195-198)
        d = self.mean.shape[-1]
        return (
            0.5 * (((x - self.mean) / self.std) ** 2).sum(dim=-1)
            + 0.5 * _LOG_2PI * d
            + self.logstd.sum(dim=-1)
        )

    def kl(self, other: "DiagGaussianPd") -> torch.Tensor:
        # (reference distributions.py:199-201)
        return (
            other.logstd
            - self.logstd
            + (self.std**2 + (self.mean - other.mean) ** 2)
            / (2.0 * other.std**2)
            - 0.5
        ).sum(dim=-1)

    def entropy(self) -> torch.Tensor:
        # sum(logstd + 0.5*log(2*pi*e)) (reference distributions.py:202-203)
        return (self.logstd + 0.5 * (_LOG_2PI + 1.0)).sum(dim=-1)

    def sample(self) -> torch.Tensor:
        # Reparameterized: mean + std * N(0,1) (reference distributions.py:204-205)
        return self.mean + self.std * torch.randn_like(self.mean)


class DiagGaussianPdType(PdType):
    def __init__(self, size: int):
        self.size = size

    def pdclass(self):
        return DiagGaussianPd

    def param_shape(self):
        return [2 * self.size]

    def sample_shape(self):
        return [self.size]

    def sample_dtype(self):
        return torch.float32


# ---------------------------------------------------------------------------
# Bernoulli
# ---------------------------------------------------------------------------


class BernoulliPd(Pd):
    """Independent Bernoullis from logits (reference distributions.py:210-229)."""

    def __init__(self, logits: torch.Tensor):
        self.logits = logits
        self.ps = torch.sigmoid(logits)

    def flatparam(self):
        return self.logits

    def mode(self):
        return torch.round(self.ps)

    def neglogp(self, x: torch.Tensor) -> torch.Tensor:
        return torch.nn.functional.binary_cross_entropy_with_logits(
            self.logits, x.to(self.logits.dtype), reduction="none"
        ).sum(dim=-1)

    def kl(self, other: "BernoulliPd") -> torch.Tensor:
        # E_p[log p - log q], written with BCE-with-logits for stability
        f = torch.nn.functional.binary_cross_entropy_with_logits
        return (
            f(other.logits, self.ps, reduction="none").sum(dim=-1)
            - f(self.logits, self.ps, reduction="none").sum(dim=-1)
        )

    def entropy(self) -> torch.Tensor:
        return torch.nn.functional.binary_cross_entropy_with_logits(
            self.logits, self.ps, reduction="none"
        ).sum(dim=-1)

    def sample(self) -> torch.Tensor:
        return (torch.rand_like(self.ps) < self.ps).to(torch.float32)


class BernoulliPdType(PdType):
    def __init__(self, size: int):
        self.size = size

    def pdclass(self):
        return BernoulliPd

    def param_shape(self):
        return [self.size]

    def sample_shape(self):
        return [self.size]

    def sample_dtype(self):
        return torch.float32


# ---------------------------------------------------------------------------
# Dispatch
# ---------------------------------------------------------------------------


def validate_probtype(pdtype: PdType, pdparam, n: int = 100_000,
                      atol_scale: float = 3.0) -> None:
    """Monte-Carlo self-check of a distribution family (the reference
    embeds this in its distributions module, distributions.py:269-295):
    (a) E[-logp(x)] == entropy and (b) KL(p||q) == -H(p) - E_p[log q],
    both within `atol_scale` standard errors.  Raises AssertionError on
    violation.  Also used by tests/test_distributions.py."""
    import math as _math

    pdparam = torch.as_tensor(pdparam, dtype=torch.float32)
    M = pdparam.unsqueeze(0).repeat(n, 1)
    pd = pdtype.pdfromflat(M)
    x = pd.sample()
    calc_logp = pd.logp(x)
    ent = pd.entropy().mean().item()
    mean_neglogp = -calc_logp.mean().item()
    stderr = calc_logp.std().item() / _math.sqrt(n)
    assert abs(ent - mean_neglogp) < atol_scale * stderr, (
        ent, mean_neglogp, stderr)

    pdparam2 = pdparam + torch.randn_like(pdparam) * 0.1
    q = pdtype.pdfromflat(pdparam2.unsqueeze(0).repeat(n, 1))
    kl_analytic = pd.kl(q).mean().item()
    logq = q.logp(x)
    kl_mc = (-ent - logq.mean()).item()
    stderr2 = logq.std().item() / _math.sqrt(n)
    assert abs(kl_analytic - kl_mc) < atol_scale * stderr2, (kl_analytic, kl_mc)


def test_probtypes() -> None:
    """Run the self-check over every family (reference distributions.py:252-266)."""
    validate_probtype(DiagGaussianPdType(3), [-0.2, 0.3, 0.4, -0.5, 0.1, -0.5])
    validate_probtype(CategoricalPdType(4), [-0.2, 0.3, 0.5, 0.1])
    validate_probtype(MultiCategoricalPdType([3, 2]), [-0.1, 0.4, 0.2, 0.3, -0.2])
    validate_probtype(BernoulliPdType(3), [-0.4, 0.2, 0.6])


def make_pdtype(ac_space) -> PdType:
    """Space -> PdType dispatch (reference distributions.py:231-243)."""
    if isinstance(ac_space, spaces.Box):
        assert len(ac_space.shape) == 1, "Box action space must be rank 1"
        return DiagGaussianPdType(ac_space.shape[0])
    if isinstance(ac_space, spaces.Discrete):
        return CategoricalPdType(ac_space.n)
    if isinstance(ac_space, spaces.MultiDiscrete):
        return MultiCategoricalPdType(ac_space.nvec.tolist())
    if isinstance(ac_space, spaces.MultiBinary):
        return BernoulliPdType(ac_space.n)
    raise NotImplementedError(f"no pdtype for space {ac_space!r}")
