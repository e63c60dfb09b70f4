"""Shared tensor / distribution utilities.

Capability parity with /root/reference/npf/utils/helpers.py (cited per item).
Implementations are MI355X-first: `logcumsumexp` uses the native single-pass
scan (the reference used an O(n^2) python loop, helpers.py:20-33).
"""

import operator
from functools import reduce

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F
from scipy.stats import rv_discrete
from torch.distributions import Normal
from torch.distributions.independent import Independent

from .initialization import weights_init

__all__ = [
    "sum_from_nth_dim",
    "logcumsumexp",
    "LightTailPareto",
    "isin_range",
    "channels_to_2nd_dim",
    "channels_to_last_dim",
    "mask_and_apply",
    "indep_shuffle_",
    "ratio_to_int",
    "prod",
    "rescale_range",
    "MultivariateNormalDiag",
    "clamp",
    "ProbabilityConverter",
    "dist_to_device",
    "make_abs_conv",
    "make_padded_conv",
    "make_depth_sep_conv",
    "CircularPad2d",
]


def sum_from_nth_dim(t, dim):
    """Sum every dimension from `dim` onward (reference helpers.py:15-17)."""
    return t.reshape(*t.shape[:dim], -1).sum(-1)


def logcumsumexp(x, dim):
    """Numerically stable log-cumsum-exp along `dim`.

    The reference worked around a missing torch op with an O(n^2) loop
    (helpers.py:20-33); torch has had a fused single-pass implementation for
    years, which is what runs on the GPU here.
    """
    return torch.logcumsumexp(x, dim)


class LightTailPareto(rv_discrete):
    """Light-tailed Pareto over sample counts for SUMO (reference helpers.py:36-52)."""

    def _cdf(self, k, alpha):
        m = self.a  # minimum number of samples (support lower bound)
        # reference uses P(K >= k): cdf(k) = 1 - P(K >= k + 1)
        k = np.clip(k + 1 - m, a_min=1, a_max=None)
        alpha = alpha - m
        return 1 - np.where(k < alpha, 1 / k, (1 / alpha) * 0.9 ** (k - alpha))


def isin_range(x, valid_range):
    """True iff every element of `x` lies in [lo, hi] (reference helpers.py:55-57)."""
    return bool(((x >= valid_range[0]) & (x <= valid_range[1])).all())


def channels_to_2nd_dim(X):
    """[B, *spatial, C] -> [B, C, *spatial] (reference helpers.py:60-65)."""
    return X.permute(0, X.dim() - 1, *range(1, X.dim() - 1))


def channels_to_last_dim(X):
    """[B, C, *spatial] -> [B, *spatial, C] (reference helpers.py:68-73)."""
    return X.permute(0, *range(2, X.dim()), 1)


def mask_and_apply(x, mask, f):
    """Apply `f` only on the masked entries of `x` (reference helpers.py:76-79)."""
    return x.masked_scatter(mask, f(x.masked_select(mask)))


def indep_shuffle_(a, axis=-1):
    """In-place independent shuffle of each 1-D slice along `axis`.

    Matches reference helpers.py:82-96 semantics (numpy in-place), implemented
    via a vectorized argsort-of-random-keys permutation instead of a Python
    loop over slices.
    """
    b = a.swapaxes(axis, -1)
    perm = np.argsort(np.random.random(b.shape), axis=-1)
    b[...] = np.take_along_axis(b, perm, axis=-1)


def ratio_to_int(percentage, max_val):
    """Interpret a value < 1 as a ratio of `max_val` (reference helpers.py:99-108)."""
    if 1 <= percentage <= max_val:
        out = percentage
    elif 0 <= percentage < 1:
        out = percentage * max_val
    else:
        raise ValueError(f"percentage={percentage} outside of [0,{max_val}].")
    return int(out)


def prod(iterable):
    """Product of an iterable (reference helpers.py:111-113)."""
    return reduce(operator.mul, iterable, 1)


def rescale_range(X, old_range, new_range):
    """Linear rescale from `old_range` to `new_range` (reference helpers.py:116-122)."""
    old_min, old_max = old_range
    new_min, new_max = new_range
    return (X - old_min) * (new_max - new_min) / (old_max - old_min) + new_min


def MultivariateNormalDiag(loc, scale_diag):
    """Diagonal-covariance Gaussian over the last dim (reference helpers.py:125-129)."""
    if loc.dim() < 1:
        raise ValueError("loc must be at least one-dimensional.")
    return Independent(Normal(loc, scale_diag), 1)


def clamp(
    x,
    minimum=-float("Inf"),
    maximum=float("Inf"),
    is_leaky=False,
    negative_slope=0.01,
    hard_min=None,
    hard_max=None,
):
    """(Leaky) clamp with optional hard bounds (reference helpers.py:132-164)."""
    lower = (
        minimum + negative_slope * (x - minimum)
        if is_leaky
        else torch.zeros_like(x) + minimum
    )
    upper = (
        maximum + negative_slope * (x - maximum)
        if is_leaky
        else torch.zeros_like(x) + maximum
    )
    out = torch.max(lower, torch.min(x, upper))
    if hard_min is not None or hard_max is not None:
        hard_min = -float("Inf") if hard_min is None else hard_min
        hard_max = float("Inf") if hard_max is None else hard_max
        out = clamp(x, minimum=hard_min, maximum=hard_max, is_leaky=False)
    return out


class ProbabilityConverter(nn.Module):
    """Map reals to probabilities elementwise (reference helpers.py:167-305)."""

    def __init__(
        self,
        min_p=0.0,
        activation="sigmoid",
        is_train_temperature=False,
        is_train_bias=False,
        trainable_dim=1,
        initial_temperature=1.0,
        initial_probability=0.5,
        initial_x=0,
        bias_transformer=nn.Identity(),
        temperature_transformer=nn.Identity(),
    ):
        super().__init__()
        self.min_p = min_p
        self.activation = activation
        self.is_train_temperature = is_train_temperature
        self.is_train_bias = is_train_bias
        self.trainable_dim = trainable_dim
        self.initial_temperature = initial_temperature
        self.initial_probability = initial_probability
        self.initial_x = initial_x
        self.bias_transformer = bias_transformer
        self.temperature_transformer = temperature_transformer
        self.reset_parameters()

    def reset_parameters(self):
        self.temperature = torch.tensor([self.initial_temperature] * self.trainable_dim)
        if self.is_train_temperature:
            self.temperature = nn.Parameter(self.temperature)
        initial_bias = self._probability_to_bias(
            self.initial_probability, initial_x=self.initial_x
        )
        self.bias = torch.tensor([initial_bias] * self.trainable_dim)
        if self.is_train_bias:
            self.bias = nn.Parameter(self.bias)

    def forward(self, x):
        self.temperature.to(x.device)
        self.bias.to(x.device)
        temperature = self.temperature_transformer(self.temperature)
        bias = self.bias_transformer(self.bias)

        if self.activation == "sigmoid":
            full_p = torch.sigmoid((x + bias) * temperature)
        elif self.activation in ("hard-sigmoid", "leaky-hard-sigmoid"):
            y = 0.2 * ((x + bias) * temperature) + 0.5
            if self.activation == "leaky-hard-sigmoid":
                full_p = clamp(
                    y,
                    minimum=0.1,
                    maximum=0.9,
                    is_leaky=True,
                    negative_slope=0.01,
                    hard_min=0,
                    hard_max=0,
                )
            else:
                full_p = clamp(y, minimum=0.0, maximum=1.0, is_leaky=False)
        else:
            raise ValueError(f"Unknown activation : {self.activation}")

        return rescale_range(full_p, (0, 1), (self.min_p, 1 - self.min_p))

    def _probability_to_bias(self, p, initial_x=0):
        assert p > self.min_p and p < 1 - self.min_p
        range_p = 1 - self.min_p * 2
        p = torch.tensor((p - self.min_p) / range_p, dtype=torch.float)
        if self.activation == "sigmoid":
            return -(torch.log((1 - p) / p) / self.initial_temperature + initial_x)
        elif self.activation in ("hard-sigmoid", "leaky-hard-sigmoid"):
            return ((p - 0.5) / 0.2) / self.initial_temperature - initial_x
        raise ValueError(f"Unknown activation : {self.activation}")


def dist_to_device(dist, device):
    """Move a wrapped Independent(Normal) distribution to a device
    (reference helpers.py:308-313, including its loc/scale quirk)."""
    if dist is None:
        return
    dist.base_dist.loc = dist.base_dist.loc.to(device)
    dist.base_dist.scale = dist.base_dist.scale.to(device)


def make_abs_conv(Conv):
    """Wrap a conv class so its effective weight is |W| (reference helpers.py:316-331).

    Used by GridConvCNP's density encoder so the depthwise kernel is a valid
    (non-negative) smoothing filter.
    """

    class AbsConv(Conv):
        def forward(self, input):
            return self._conv_forward(input, self.weight.abs(), self.bias)

    return AbsConv


def make_padded_conv(Conv, Padder):
    """Wrap a conv class to apply an arbitrary `Padder` first
    (reference helpers.py:334-351)."""

    class PaddedConv(Conv):
        def __init__(self, *args, Padder=Padder, padding=0, **kwargs):
            native_padding = 0
            if Padder is None:
                Padder = nn.Identity
                native_padding = padding
            super().__init__(*args, padding=native_padding, **kwargs)
            self.padder = Padder(padding)

        def forward(self, X):
            return super().forward(self.padder(X))

    return PaddedConv


def make_depth_sep_conv(Conv):
    """Depthwise-separable wrapper around a conv class (reference helpers.py:354-403).

    The `depthwise` / `pointwise` attribute names are part of the checkpoint
    format (e.g. `...conv1.depthwise.weight`).
    """

    class DepthSepConv(nn.Module):
        def __init__(
            self,
            in_channels,
            out_channels,
            kernel_size,
            confidence=False,
            bias=True,
            **kwargs,
        ):
            super().__init__()
            self.depthwise = Conv(
                in_channels,
                in_channels,
                kernel_size,
                groups=in_channels,
                bias=bias,
                **kwargs,
            )
            self.pointwise = Conv(in_channels, out_channels, 1, bias=bias)
            self.reset_parameters()

        def forward(self, x):
            return self.pointwise(self.depthwise(x))

        def reset_parameters(self):
            weights_init(self)

    return DepthSepConv


class CircularPad2d(nn.Module):
    """2-D circular padding module (reference helpers.py:406-414)."""

    def __init__(self, padding):
        super().__init__()
        self.padding = padding

    def forward(self, x):
        return F.pad(x, (self.padding,) * 4, mode="circular")


class BackwardPDB(torch.autograd.Function):
    """Autograd identity that drops into pdb when its gradient is non-finite
    (reference helpers.py:417-436 — debugging hook for exploding losses)."""

    @staticmethod
    def forward(ctx, x, name="debugger"):
        ctx.name = name
        return x

    @staticmethod
    def backward(ctx, grad_output):
        if not torch.isfinite(grad_output).all():
            import pdb

            pdb.set_trace()  # noqa: T100  (intentional: debug hook)
        return grad_output, None


def backward_pdb(x, name="debugger"):
    """Insert a BackwardPDB probe on `x`."""
    return BackwardPDB.apply(x, name)
