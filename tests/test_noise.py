"""Graph-safe latent sampling (static noise pool, npf/ops/noise.py)."""

import pytest
import torch

from npf.ops import noise


@pytest.fixture(autouse=True)
def _clean_pool():
    noise.clear_noise_pool()
    noise.enable_noise_pool(False)
    yield
    noise.clear_noise_pool()
    noise.enable_noise_pool(False)


def test_pool_returns_same_buffer_and_refreshes():
    b1 = noise.pool_noise((4, 8), "cpu")
    b2 = noise.pool_noise((4, 8), "cpu")
    assert b1 is b2  # same address: a capture requirement
    old = b1.clone()
    noise.refresh_noise_()
    assert not torch.equal(b1, old)
    assert noise.pool_noise((4, 8), "cpu") is b1


def test_rsample_pool_matches_distribution_and_grads():
    from npf.neuralproc.base import LatentNeuralProcessFamily
    from npf.utils.helpers import MultivariateNormalDiag

    torch.manual_seed(0)
    loc = torch.randn(16, 32, requires_grad=True)
    scale = torch.rand(16, 32).add(0.5).requires_grad_()
    dist = MultivariateNormalDiag(loc, scale)

    noise.enable_noise_pool(True)
    zs = []
    for _ in range(200):
        noise.refresh_noise_()
        zs.append(LatentNeuralProcessFamily._rsample(dist, 4).detach())
    z = torch.stack(zs)  # [200, 4, 16, 32]
    # matches N(loc, scale) moments
    err_m = (z.mean((0, 1)) - loc.detach()).abs().max()
    err_s = (z.std((0, 1)) - scale.detach()).abs().max()
    assert float(err_m) < 0.2, float(err_m)
    assert float(err_s) < 0.2, float(err_s)
    # reparameterized: grads flow to loc and scale (sample AFTER the last
    # refresh — the saved eps must not be refreshed before backward, which
    # is exactly the replay discipline the graph path follows)
    noise.refresh_noise_()
    z_live = LatentNeuralProcessFamily._rsample(dist, 4)
    z_live.sum().backward()
    assert loc.grad is not None and scale.grad is not None
    assert torch.all(loc.grad == 4)  # d(sum)/d(loc) = n_z per element


def test_latent_model_forward_backward_with_pool():
    from npf import ELBOLossLNPF
    from npf.zoo import lnp_1d

    torch.manual_seed(0)
    m = lnp_1d()
    m.train()
    crit = ELBOLossLNPF()
    crit.train()
    noise.enable_noise_pool(True)
    Xc = torch.rand(2, 5, 1) * 2 - 1
    Yc = torch.randn(2, 5, 1)
    Xt = torch.rand(2, 16, 1) * 2 - 1
    Yt = torch.randn(2, 16, 1)
    noise.refresh_noise_()
    loss = crit(m(Xc, Yc, Xt, Yt), Yt)
    loss.backward()
    assert torch.isfinite(loss)
    # a second step with refreshed noise gives a different draw
    z1 = m._rsample(
        __import__("npf.utils.helpers", fromlist=["MultivariateNormalDiag"])
        .MultivariateNormalDiag(torch.zeros(2, 3), torch.ones(2, 3)),
        2,
    ).clone()
    noise.refresh_noise_()
    z2 = m._rsample(
        __import__("npf.utils.helpers", fromlist=["MultivariateNormalDiag"])
        .MultivariateNormalDiag(torch.zeros(2, 3), torch.ones(2, 3)),
        2,
    )
    assert not torch.equal(z1, z2)


def test_disabled_pool_uses_plain_rsample():
    from npf.neuralproc.base import LatentNeuralProcessFamily
    from npf.utils.helpers import MultivariateNormalDiag

    dist = MultivariateNormalDiag(torch.zeros(2, 3), torch.ones(2, 3))
    z = LatentNeuralProcessFamily._rsample(dist, 5)
    assert z.shape == (5, 2, 3)
    assert len(noise._POOLS) == 0
