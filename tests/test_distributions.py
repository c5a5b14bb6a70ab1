"""Distribution tests.

Port of the reference's embedded statistical self-check
(reference Others/distributions.py:252-295: validate_probtype) plus
cross-checks against torch.distributions closed forms.
"""

import math

import numpy as np
import pytest
import torch

from dppo_amd import spaces
from dppo_amd.distributions import (
    BernoulliPdType,
    CategoricalPdType,
    DiagGaussianPdType,
    MultiCategoricalPdType,
    make_pdtype,
)

from dppo_amd.distributions import validate_probtype


def test_validate_diag_gaussian():
    validate_probtype(DiagGaussianPdType(3), [-0.2, 0.3, 0.4, -0.5, 0.1, -0.5])


def test_validate_categorical():
    validate_probtype(CategoricalPdType(4), [-0.2, 0.3, 0.5, 0.1])


def test_validate_multicategorical():
    validate_probtype(MultiCategoricalPdType([3, 2]), [-0.1, 0.4, 0.2, 0.3, -0.2])


def test_validate_bernoulli():
    validate_probtype(BernoulliPdType(3), [-0.4, 0.2, 0.6])


def test_categorical_vs_torch():
    logits = torch.randn(64, 7)
    pd = CategoricalPdType(7).pdfromflat(logits)
    ref = torch.distributions.Categorical(logits=logits)
    a = pd.sample()
    torch.testing.assert_close(pd.logp(a), ref.log_prob(a), atol=1e-5, rtol=1e-5)
    torch.testing.assert_close(pd.entropy(), ref.entropy(), atol=1e-5, rtol=1e-5)
    logits2 = torch.randn(64, 7)
    q = CategoricalPdType(7).pdfromflat(logits2)
    qref = torch.distributions.Categorical(logits=logits2)
    torch.testing.assert_close(
        pd.kl(q), torch.distributions.kl_divergence(ref, qref), atol=1e-5, rtol=1e-5
    )


def test_diag_gaussian_vs_torch():
    mean = torch.randn(64, 5)
    logstd = torch.randn(64, 5) * 0.3
    flat = torch.cat([mean, logstd], dim=-1)
    pd = DiagGaussianPdType(5).pdfromflat(flat)
    ref = torch.distributions.Independent(
        torch.distributions.Normal(mean, logstd.exp()), 1
    )
    x = pd.sample()
    torch.testing.assert_close(pd.logp(x), ref.log_prob(x), atol=1e-5, rtol=1e-5)
    torch.testing.assert_close(pd.entropy(), ref.entropy(), atol=1e-5, rtol=1e-5)
    mean2, logstd2 = torch.randn(64, 5), torch.randn(64, 5) * 0.2
    q = DiagGaussianPdType(5).pdfromflat(torch.cat([mean2, logstd2], -1))
    qref = torch.distributions.Independent(
        torch.distributions.Normal(mean2, logstd2.exp()), 1
    )
    torch.testing.assert_close(
        pd.kl(q), torch.distributions.kl_divergence(ref, qref), atol=1e-5, rtol=1e-5
    )


def test_categorical_neglogp_second_order_differentiable():
    """The reference writes categorical neglogp as softmax CE on one-hot
    specifically to keep it twice-differentiable (distributions.py:131-138)."""
    logits = torch.randn(8, 5, requires_grad=True)
    pd = CategoricalPdType(5).pdfromflat(logits)
    a = torch.randint(5, (8,))
    loss = pd.neglogp(a).sum()
    (g,) = torch.autograd.grad(loss, logits, create_graph=True)
    (g2,) = torch.autograd.grad(g.pow(2).sum(), logits)
    assert torch.isfinite(g2).all()


def test_categorical_sample_distribution():
    """Gumbel-max sampling reproduces softmax probabilities."""
    logits = torch.tensor([0.0, 1.0, 2.0])
    pd = CategoricalPdType(3).pdfromflat(logits.repeat(200_000, 1))
    counts = torch.bincount(pd.sample(), minlength=3).float() / 200_000
    probs = torch.softmax(logits, 0)
    assert torch.allclose(counts, probs, atol=0.01)


def test_neglogp_accepts_trailing_dim():
    """Call sites in the reference feed actions shaped [B,1] (Worker.py:104)."""
    logits = torch.randn(16, 4)
    pd = CategoricalPdType(4).pdfromflat(logits)
    a = torch.randint(4, (16,))
    torch.testing.assert_close(pd.neglogp(a), pd.neglogp(a.unsqueeze(-1)))


def test_make_pdtype_dispatch():
    assert isinstance(make_pdtype(spaces.Box(-1, 1, (4,))), DiagGaussianPdType)
    assert isinstance(make_pdtype(spaces.Discrete(3)), CategoricalPdType)
    assert isinstance(make_pdtype(spaces.MultiDiscrete([2, 3])), MultiCategoricalPdType)
    assert isinstance(make_pdtype(spaces.MultiBinary(5)), BernoulliPdType)
    assert make_pdtype(spaces.Box(-1, 1, (4,))).param_shape() == [8]


def test_mode_and_shapes():
    pd = DiagGaussianPdType(3).pdfromflat(torch.randn(10, 6))
    assert pd.mode().shape == (10, 3)
    assert pd.sample().shape == (10, 3)
    assert pd.entropy().shape == (10,)
    cpd = CategoricalPdType(5).pdfromflat(torch.randn(10, 5))
    assert cpd.sample().shape == (10,)
    assert cpd.sample().dtype == torch.int64
