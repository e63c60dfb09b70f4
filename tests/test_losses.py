"""Loss tests vs hand-computed values and reference semantics."""

import math

import pytest
import torch
from torch.distributions import Normal
from torch.distributions.independent import Independent

from npf import CNPFLoss, ELBOLossLNPF, NLLLossLNPF, SUMOLossLNPF
from npf.utils.helpers import LightTailPareto, MultivariateNormalDiag


def _toy_pred(Z=1, B=2, T=3, Y=1, seed=0):
    g = torch.Generator().manual_seed(seed)
    loc = torch.randn(Z, B, T, Y, generator=g)
    scale = torch.rand(Z, B, T, Y, generator=g) + 0.5
    y = torch.randn(B, T, Y, generator=g)
    return MultivariateNormalDiag(loc, scale), y


def test_cnpf_loss_hand_computed():
    p_y, y = _toy_pred()
    loss = CNPFLoss(reduction=None)
    loss.train()
    got = loss((p_y, None, None, None), y)
    # by hand: -sum_t log N(y; loc, scale)
    lp = Normal(p_y.base_dist.loc, p_y.base_dist.scale).log_prob(y)
    expected = -lp.sum(dim=(2, 3)).squeeze(0)
    assert torch.allclose(got, expected, atol=1e-6)


def test_cnpf_reductions():
    p_y, y = _toy_pred()
    for red, f in [("mean", torch.mean), ("sum", torch.sum)]:
        loss = CNPFLoss(reduction=red)
        loss.train()
        got = loss((p_y, None, None, None), y)
        unred = CNPFLoss(reduction=None)
        unred.train()
        assert torch.allclose(got, f(unred((p_y, None, None, None), y)))


def test_elbo_hand_computed():
    g = torch.Generator().manual_seed(1)
    Z, B, T = 4, 2, 3
    p_y, y = _toy_pred(Z=Z, B=B, T=T)
    q_zCc = MultivariateNormalDiag(
        torch.randn(B, 1, 8, generator=g), torch.rand(B, 1, 8, generator=g) + 0.5
    )
    q_zCct = MultivariateNormalDiag(
        torch.randn(B, 1, 8, generator=g), torch.rand(B, 1, 8, generator=g) + 0.5
    )
    loss = ELBOLossLNPF(reduction=None)
    loss.train()
    got = loss((p_y, None, q_zCc, q_zCct), y)
    lp = Normal(p_y.base_dist.loc, p_y.base_dist.scale).log_prob(y).sum((2, 3))
    kl = torch.distributions.kl_divergence(q_zCct, q_zCc).sum(1)
    expected = -(lp.mean(0) - kl)
    assert torch.allclose(got, expected, atol=1e-5)


def test_npml_logsumexp_and_importance_weights():
    g = torch.Generator().manual_seed(2)
    Z, B = 5, 3
    p_y, y = _toy_pred(Z=Z, B=B)
    loss = NLLLossLNPF(reduction=None)
    loss.train()
    got = loss((p_y, None, None, None), y)
    lp = Normal(p_y.base_dist.loc, p_y.base_dist.scale).log_prob(y).sum((2, 3))
    expected = -(torch.logsumexp(lp, 0) - math.log(Z))
    assert torch.allclose(got, expected, atol=1e-5)

    # with q_zCct: importance weights enter
    z = torch.randn(Z, B, 1, 8, generator=g)
    q_zCc = MultivariateNormalDiag(
        torch.randn(B, 1, 8, generator=g), torch.rand(B, 1, 8, generator=g) + 0.5
    )
    q_zCct = MultivariateNormalDiag(
        torch.randn(B, 1, 8, generator=g), torch.rand(B, 1, 8, generator=g) + 0.5
    )
    got_iw = loss((p_y, z, q_zCc, q_zCct), y)
    w = lp + q_zCc.log_prob(z).sum(-1) - q_zCct.log_prob(z).sum(-1)
    expected_iw = -(torch.logsumexp(w, 0) - math.log(Z))
    assert torch.allclose(got_iw, expected_iw, atol=1e-5)


def test_eval_forces_npml():
    """In eval mode every loss reports NPML with q_zCct dropped
    (reference losses.py:62-69)."""
    p_y, y = _toy_pred(Z=6)
    g = torch.Generator().manual_seed(3)
    z = torch.randn(6, 2, 1, 8, generator=g)
    q_zCc = MultivariateNormalDiag(
        torch.randn(2, 1, 8, generator=g), torch.rand(2, 1, 8, generator=g) + 0.5
    )
    q_zCct = MultivariateNormalDiag(
        torch.randn(2, 1, 8, generator=g), torch.rand(2, 1, 8, generator=g) + 0.5
    )
    elbo = ELBOLossLNPF(reduction=None)
    npml = NLLLossLNPF(reduction=None)
    elbo.eval()
    npml.eval()
    a = elbo((p_y, z, q_zCc, q_zCct), y)
    b = npml((p_y, z, q_zCc, None), y)
    assert torch.allclose(a, b, atol=1e-6)


def test_sumo_runs_and_is_finite():
    dist_k = LightTailPareto(a=5).freeze(85)
    Z = int(dist_k.rvs())
    p_y, y = _toy_pred(Z=max(Z, 6))
    loss = SUMOLossLNPF()
    loss.train()
    got = loss.get_loss(p_y, None, None, None, y)
    assert torch.isfinite(got).all()


def test_logcumsumexp_matches_reference_loop():
    from npf.utils.helpers import logcumsumexp

    g = torch.Generator().manual_seed(4)
    x = torch.randn(7, 3, generator=g)
    # reference O(n^2) formulation (npf/utils/helpers.py:20-33)
    ref = torch.cat(
        [torch.logsumexp(x[:i], dim=0, keepdim=True) for i in range(1, 8)], dim=0
    )
    assert torch.allclose(logcumsumexp(x, 0), ref, atol=1e-6)
