"""Set convolution (continuous conv over off-grid sets) and its RBFs.

Parity with /root/reference/npf/architectures/setcnn.py (SetConv :194-268,
ExpRBF :86-142, UnsharedExpRBF :14-83, MlpRBF :145-191).

MI355X-first: the ExpRBF SetConv path (the ConvCNP/ConvLNP hot loop, SURVEY.md
§2.3 rows 1-2) is a single fused HIP kernel (`npf.ops.setconv_gaussian`) that
LDS-stages the key/value tiles and never materializes the [B, Q, K] pairwise
tensor on HBM; the final channel resize stays a GEMM.
"""

import math

import torch
import torch.nn as nn
from torch.nn import functional as F

from npf.ops import functional as ops
from npf.utils.helpers import mask_and_apply
from npf.utils.initialization import weights_init

from .mlp import MLP

__all__ = ["SetConv", "MlpRBF", "ExpRBF", "UnsharedExpRBF"]


def _max_dist_param(max_dist, max_dist_weight, p):
    """Initial pre-softplus length-scale so that a point at `max_dist` gets
    weight `max_dist_weight`: sigma = max_dist / (-log w)^(1/p), inverted
    through softplus (reference setcnn.py:114-124)."""
    sigma = max_dist / ((-math.log(max_dist_weight)) ** (1 / p))
    return math.log(math.exp(sigma) - 1)  # inverse softplus


class ExpRBF(nn.Module):
    """Exponential (p=2: Gaussian) RBF with softmax-normalized weights and a
    raw exp-sum density channel (reference setcnn.py:86-142)."""

    def __init__(self, x_dim, max_dist=1 / 256, max_dist_weight=0.9, p=2, **kwargs):
        super().__init__()
        self.max_dist = max_dist
        self.max_dist_weight = max_dist_weight
        self.p = p
        self.length_scale_param = nn.Parameter(torch.tensor([0.0]))
        self.reset_parameters()

    def reset_parameters(self):
        weights_init(self)
        self.length_scale_param = nn.Parameter(
            torch.tensor([_max_dist_param(self.max_dist, self.max_dist_weight, self.p)])
        )

    def sigma(self):
        """Positive length-scale; the 1e-5 floor is part of the numerics."""
        return 1e-5 + F.softplus(self.length_scale_param)

    def forward(self, diff):
        """diff [B, Q, K, d] -> (weights [B,Q,K,1], density [B,Q,1])."""
        dist = torch.norm(diff, p=self.p, dim=-1, keepdim=True)
        inp = -(dist / self.sigma()).pow(self.p)
        # softmax over keys = numerically-stable density normalization
        out = torch.softmax(inp, dim=-2)
        density = torch.exp(inp).sum(dim=-2)
        return out, density


class UnsharedExpRBF(nn.Module):
    """ExpRBF variant with separate length-scales for weight and density
    (reference setcnn.py:14-83)."""

    def __init__(self, x_dim, max_dist=1 / 256, max_dist_weight=0.99, p=2, **kwargs):
        super().__init__()
        self.max_dist = max_dist
        self.max_dist_weight = max_dist_weight
        self.p = p
        self.length_scale_param = nn.Parameter(torch.tensor([0.0] * 2))
        self.reset_parameters()

    def reset_parameters(self):
        weights_init(self)
        self.length_scale_param = nn.Parameter(
            torch.tensor(
                [_max_dist_param(self.max_dist, self.max_dist_weight, self.p)] * 2
            )
        )

    def forward(self, diff):
        dist = torch.norm(diff, p=self.p, dim=-1, keepdim=True)
        sigma = 1e-5 + F.softplus(self.length_scale_param)
        out = torch.exp(-(dist / sigma).pow(self.p))
        # channel 1 drives the density, channel 0 the (density-normalized) weight
        density = out[..., 1:].sum(dim=-2)
        weight = out[..., 0:1] / (density.unsqueeze(2) + 1e-8)
        return weight, density


class MlpRBF(nn.Module):
    """Learned kernel with window-sparse application (reference setcnn.py:145-191)."""

    def __init__(self, x_dim, is_abs_dist=True, window_size=0.25, **kwargs):
        super().__init__()
        self.is_abs_dist = is_abs_dist
        self.window_size = window_size
        self.mlp = MLP(x_dim, 1, n_hidden_layers=3, hidden_size=16)
        self.reset_parameters()

    def reset_parameters(self):
        weights_init(self)

    def forward(self, diff):
        abs_diff = diff.abs()
        mask = abs_diff < self.window_size
        if self.is_abs_dist:
            diff = abs_diff
        # sparse apply: MLP only evaluated inside the window
        weight = mask_and_apply(
            diff, mask, lambda x: self.mlp(x.unsqueeze(1)).abs().squeeze()
        )
        weight = weight * mask.float()
        density = weight.sum(dim=-2, keepdim=True)
        out = weight / (density + 1e-5)
        return out, density.squeeze(-1)


class SetConv(nn.Module):
    """Continuous set convolution {key,value},{query} -> {target}
    (reference setcnn.py:194-268).

    Output per query: density-normalized RBF-weighted value sum, concatenated
    with the density channel, then a linear channel resize.  With the default
    `ExpRBF` (p=2) the whole reduction runs as one fused HIP kernel on GPU.
    """

    def __init__(self, x_dim, in_channels, out_channels, RadialBasisFunc=ExpRBF, **kwargs):
        super().__init__()
        assert x_dim == 1, "Currently only supports single spatial dimension `x_dim==1`"
        self.radial_basis_func = RadialBasisFunc(x_dim, **kwargs)
        self.resizer = nn.Linear(in_channels + 1, out_channels)
        self.reset_parameters()

    def reset_parameters(self):
        weights_init(self)

    def _is_fused(self):
        return isinstance(self.radial_basis_func, ExpRBF) and self.radial_basis_func.p == 2

    def forward(self, keys, queries, values):
        """keys [B,K,x], queries [B,Q,x], values [B,K,C] -> [B,Q,out]."""
        if self._is_fused():
            targets = ops.setconv_gaussian(
                keys, queries, values, self.radial_basis_func.sigma()
            )
        else:
            diff = keys.unsqueeze(1) - queries.unsqueeze(2)  # [B,Q,K,x]
            weight, density = self.radial_basis_func(diff)
            targets = (weight * values.unsqueeze(1)).sum(dim=2)
            targets = torch.cat([targets, density], dim=-1)
        return self.resizer(targets)
