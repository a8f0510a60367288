"""Distribution tests (reference test model: pytorch/rl
test/test_distributions.py): sampling bounds, log-prob consistency,
masking semantics, deterministic modes."""
import math

import pytest
import torch

from rl_amd.modules.distributions import (
    Delta,
    IndependentNormal,
    MaskedCategorical,
    MaskedOneHotCategorical,
    OneHotCategorical,
    Ordinal,
    TanhDelta,
    TanhNormal,
    TruncatedNormal,
)


class TestTanhNormal:
    def test_bounds_and_logprob_finite(self):
        torch.manual_seed(0)
        loc = torch.randn(64, 4)
        scale = torch.rand(64, 4) + 0.1
        d = TanhNormal(loc, scale, low=-2.0, high=2.0)
        x = d.sample()
        assert (x >= -2.0).all() and (x <= 2.0).all()
        lp = d.log_prob(x)
        assert lp.shape == (64,)
        assert torch.isfinite(lp).all()

    def test_mode_is_squashed_mean(self):
        loc = torch.tensor([[0.5, -0.3]])
        d = TanhNormal(loc, torch.ones_like(loc))
        assert torch.allclose(d.mode, torch.tanh(loc))

    def test_rsample_grads(self):
        loc = torch.randn(8, 2, requires_grad=True)
        d = TanhNormal(loc, torch.ones(8, 2) * 0.5)
        x = d.rsample()
        x.sum().backward()
        assert loc.grad is not None


class TestTruncatedNormal:
    def test_support(self):
        torch.manual_seed(0)
        d = TruncatedNormal(torch.zeros(128, 2), torch.ones(128, 2), low=-1.0, high=1.0)
        x = d.sample()
        assert (x >= -1).all() and (x <= 1).all()
        assert torch.isfinite(d.log_prob(x)).all()


class TestDeltaFamily:
    def test_delta_logprob(self):
        v = torch.randn(5, 3)
        d = Delta(v)
        assert (d.sample() == v).all()
        assert (d.log_prob(v) == 0).all() or torch.isfinite(d.log_prob(v)).all()

    def test_tanh_delta_squashes(self):
        v = torch.randn(5, 3) * 3
        d = TanhDelta(v, low=-1.0, high=1.0)
        x = d.sample()
        assert (x.abs() <= 1).all()


class TestCategoricalFamily:
    def test_one_hot_sample_and_logprob(self):
        torch.manual_seed(0)
        logits = torch.randn(16, 5)
        d = OneHotCategorical(logits=logits)
        x = d.sample()
        assert x.shape == (16, 5) and (x.sum(-1) == 1).all()
        lp = d.log_prob(x)
        ref = torch.distributions.Categorical(logits=logits).log_prob(x.argmax(-1))
        assert torch.allclose(lp, ref, atol=1e-5)

    def test_masked_categorical_respects_mask(self):
        torch.manual_seed(0)
        logits = torch.randn(32, 4)
        mask = torch.tensor([True, False, True, False]).expand(32, 4)
        d = MaskedCategorical(logits=logits, mask=mask)
        x = d.sample((64,))
        assert ((x == 0) | (x == 2)).all()
        assert torch.isfinite(d.log_prob(x)).all()

    def test_masked_one_hot(self):
        torch.manual_seed(0)
        logits = torch.randn(8, 3)
        mask = torch.tensor([True, True, False]).expand(8, 3)
        d = MaskedOneHotCategorical(logits=logits, mask=mask)
        x = d.sample()
        assert (x[..., 2] == 0).all()

    def test_ordinal(self):
        torch.manual_seed(0)
        scores = torch.randn(16, 6)
        d = Ordinal(scores)
        x = d.sample()
        assert x.max() < 6 and x.min() >= 0
        assert torch.isfinite(d.log_prob(x)).all()


class TestIndependentNormal:
    def test_event_dim_reduction(self):
        d = IndependentNormal(torch.zeros(7, 3), torch.ones(7, 3))
        x = d.sample()
        assert d.log_prob(x).shape == (7,)
        ref = torch.distributions.Normal(0.0, 1.0).log_prob(x).sum(-1)
        assert torch.allclose(d.log_prob(x), ref, atol=1e-5)
