"""Training objectives for the Neural Process Family.

Parity with /root/reference/npf/losses.py (BaseLossNPF :27-109, CNPFLoss
:112-123, ELBOLossLNPF :126-150, NLLLossLNPF :153-203, SUMOLossLNPF
:207-276).  Evaluation ALWAYS uses NPML with q_zCct=None (reference
losses.py:62-69) — this defines the reported test log-likelihood.

MI355X-first: `sum_log_prob` on a diagonal Gaussian runs through the fused
HIP log-prob+reduce kernel (npf.ops.gaussian_nll_sum); logcumsumexp for SUMO
uses the native scan instead of the reference's O(n^2) loop.
"""

import abc
import math

import torch
import torch.nn as nn
from torch.distributions import Normal
from torch.distributions.independent import Independent
from torch.distributions.kl import kl_divergence

from npf.ops import functional as ops
from npf.utils.helpers import LightTailPareto, logcumsumexp, sum_from_nth_dim

__all__ = ["CNPFLoss", "ELBOLossLNPF", "SUMOLossLNPF", "NLLLossLNPF"]


def sum_log_prob(prob, sample):
    """log prob summed over all but the (z_samples, batch) dims.

    Dispatches diagonal Gaussians (Independent(Normal, 1)) to the fused
    kernel; anything else uses the generic distribution API.
    """
    if (
        isinstance(prob, Independent)
        and isinstance(prob.base_dist, Normal)
        and prob.reinterpreted_batch_ndims == 1
        and prob.base_dist.loc.dim() >= 2
    ):
        loc = prob.base_dist.loc
        scale = prob.base_dist.scale
        if sample.dim() == loc.dim():
            return ops.gaussian_nll_sum(loc, scale, sample)
    log_p = prob.log_prob(sample)
    return sum_from_nth_dim(log_p, 2)


class BaseLossNPF(nn.Module, abc.ABC):
    """Shared NPF loss driver (reference losses.py:27-109).

    reduction : {None, "mean", "sum"} over the batch.
    is_force_mle_eval : force NPML (q_zCct=None) at eval time.
    """

    def __init__(self, reduction="mean", is_force_mle_eval=True):
        super().__init__()
        self.reduction = reduction
        self.is_force_mle_eval = is_force_mle_eval

    def forward(self, pred_outputs, Y_trgt):
        """pred_outputs = NeuralProcessFamily forward tuple; returns
        [batch_size] if reduction is None else a scalar."""
        p_yCc, z_samples, q_zCc, q_zCct = pred_outputs

        if self.training:
            loss = self.get_loss(p_yCc, z_samples, q_zCc, q_zCct, Y_trgt)
        else:
            if self.is_force_mle_eval:
                q_zCct = None
            loss = NLLLossLNPF.get_loss(self, p_yCc, z_samples, q_zCc, q_zCct, Y_trgt)

        if self.reduction is None:
            return loss
        if self.reduction == "mean":
            return loss.mean(0)
        if self.reduction == "sum":
            return loss.sum(0)
        raise ValueError(f"Unknown {self.reduction}")

    @abc.abstractmethod
    def get_loss(self, p_yCc, z_samples, q_zCc, q_zCct, Y_trgt):
        """Return per-task loss [batch_size]."""


class CNPFLoss(BaseLossNPF):
    """Exact NLL for the conditional sub-family (reference losses.py:112-123)."""

    def get_loss(self, p_yCc, _, q_zCc, ___, Y_trgt):
        assert q_zCc is None
        sum_log_p_yCz = sum_log_prob(p_yCc, Y_trgt)  # [1, B]
        return -sum_log_p_yCz.squeeze(0)


class ELBOLossLNPF(BaseLossNPF):
    """NPVI: approximate conditional ELBO (reference losses.py:126-150)."""

    def get_loss(self, p_yCc, _, q_zCc, q_zCct, Y_trgt):
        # E_{q(z|C,T)}[ sum_t log p(y^t|z) ]
        E_z_sum_log_p_yCz = sum_log_prob(p_yCc, Y_trgt).mean(0)
        # sum_l KL[ q(z^l|C,T) || q(z^l|C) ]
        if (
            isinstance(q_zCct, Independent)
            and isinstance(q_zCct.base_dist, Normal)
            and isinstance(q_zCc, Independent)
            and isinstance(q_zCc.base_dist, Normal)
            # the fused kernel indexes by q_zCct's layout: all four tensors
            # must have identical (non-broadcast) shapes
            and q_zCct.base_dist.loc.shape == q_zCc.base_dist.loc.shape
            and q_zCct.base_dist.scale.shape == q_zCct.base_dist.loc.shape
            and q_zCc.base_dist.scale.shape == q_zCc.base_dist.loc.shape
        ):
            # fused HIP kernel: analytic KL + latent-set reduce in one pass
            E_z_kl = ops.gaussian_kl_sum(
                q_zCct.base_dist.loc, q_zCct.base_dist.scale,
                q_zCc.base_dist.loc, q_zCc.base_dist.scale,
            )
        else:
            kl_z = kl_divergence(q_zCct, q_zCc)
            E_z_kl = sum_from_nth_dim(kl_z, 1)
        return -(E_z_sum_log_p_yCz - E_z_kl)


class NLLLossLNPF(BaseLossNPF):
    """NPML: MC estimate of the marginal log likelihood, with importance
    weights when sampling from q(z|C,T) (reference losses.py:153-203)."""

    def get_loss(self, p_yCc, z_samples, q_zCc, q_zCct, Y_trgt):
        n_z_samples = p_yCc.batch_shape[0]

        if (
            q_zCct is None
            and isinstance(p_yCc, Independent)
            and isinstance(p_yCc.base_dist, Normal)
            and p_yCc.reinterpreted_batch_ndims == 1
            and Y_trgt.dim() == p_yCc.base_dist.loc.dim() - 1
        ):
            # no importance weights: the whole objective (per-z target-summed
            # log-lik + logmeanexp over z) is one fused kernel pair
            return -ops.gaussian_nll_logmeanexp(
                p_yCc.base_dist.loc, p_yCc.base_dist.scale, Y_trgt
            )

        # [Z, B]
        sum_log_w_k = sum_log_prob(p_yCc, Y_trgt)
        if q_zCct is not None:
            # importance sampling: + log q(z|C) - log q(z|C,T)
            sum_log_w_k = (
                sum_log_w_k
                + sum_log_prob(q_zCc, z_samples)
                - sum_log_prob(q_zCct, z_samples)
            )

        # log mean_z exp(.) = logsumexp_z - log Z
        log_E_z_sum_p_yCz = torch.logsumexp(sum_log_w_k, 0) - math.log(n_z_samples)
        return -log_E_z_sum_p_yCz


class SUMOLossLNPF(BaseLossNPF):
    """SUMO: unbiased log-marginal estimator via Russian-roulette telescoping
    over a random number of z samples (Luo et al. 2020; reference
    losses.py:207-276)."""

    def __init__(self, p_n_z_samples=LightTailPareto(a=5).freeze(85), **kwargs):
        super().__init__()
        self.p_n_z_samples = p_n_z_samples

    def get_loss(self, p_yCc, z_samples, q_zCc, q_zCct, Y_trgt):
        n_z_samples = p_yCc.batch_shape[0]

        sum_log_w_k = sum_log_prob(p_yCc, Y_trgt)
        if q_zCct is not None:
            sum_log_w_k = (
                sum_log_w_k
                + sum_log_prob(q_zCc, z_samples)
                - sum_log_prob(q_zCct, z_samples)
            )

        ks = (torch.arange(n_z_samples) + 1).unsqueeze(-1)
        log_ks = ks.float().log().to(sum_log_w_k.device)

        # cumulative IWAE estimates [Z, B]
        cum_iwae = logcumsumexp(sum_log_w_k, 0) - log_ks

        # inverse of the reverse-cdf P(K >= k)
        inv_weights = torch.from_numpy(1 - self.p_n_z_samples.cdf(ks - 1)).to(
            sum_log_w_k.device
        )

        m = self.p_n_z_samples.support()[0]
        sumo = cum_iwae[m - 1] + (
            inv_weights[m:] * (cum_iwae[m:] - cum_iwae[m - 1 : -1])
        ).sum(0)
        return -sumo
