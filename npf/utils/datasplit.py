"""Context/target splitters and grid maskers — the meta-learning episode
generator.

Parity with /root/reference/npf/utils/datasplit.py (indexers :30-145,
CntxtTrgtGetter :148-255, maskers :259-333, GridCntxtTrgtGetter :336-452,
SuperresolutionCntxtTrgtGetter :455-490).
"""

import functools
import random

import numpy as np
import torch
import torch.nn.functional as F
from scipy.stats import betabinom

from .helpers import (
    channels_to_2nd_dim,
    channels_to_last_dim,
    indep_shuffle_,
    prod,
    ratio_to_int,
)

__all__ = [
    "get_all_indcs",
    "GetRangeIndcs",
    "GetRandomIndcs",
    "CntxtTrgtGetter",
    "RandomMasker",
    "half_masker",
    "no_masker",
    "GridCntxtTrgtGetter",
]


# --------------------------------------------------------------------------- #
# index selectors
# --------------------------------------------------------------------------- #


def get_all_indcs(batch_size, n_possible_points):
    """All indices, shared across the batch (reference :30-34)."""
    return torch.arange(n_possible_points).expand(batch_size, n_possible_points)


class GetRangeIndcs:
    """Indices in a fixed range (reference :37-45)."""

    def __init__(self, arange):
        self.arange = arange

    def __call__(self, batch_size, n_possible_points):
        indcs = torch.arange(*self.arange)
        return indcs.expand(batch_size, len(indcs))


class GetIndcsMerger:
    """Concatenate the outputs of several indexers (reference :48-57)."""

    def __init__(self, indexers):
        self.indexers = indexers

    def __call__(self, batch_size, n_possible_points):
        return torch.cat(
            [ix(batch_size, n_possible_points) for ix in self.indexers], dim=1
        )


class GetRandomIndcs:
    """Random index subsets, uniform or beta-binomial count
    (reference :60-145).

    a/b: min/max count (ratios if < 1), or alpha/beta when
    `is_beta_binomial`.  `is_batch_share` reuses one permutation for the
    whole batch; `proba_uniform` mixes in fully-uniform counts.
    """

    def __init__(
        self,
        a=0.1,
        b=0.5,
        is_batch_share=False,
        range_indcs=None,
        is_ensure_one=False,
        is_beta_binomial=False,
        proba_uniform=0,
    ):
        self.a = a
        self.b = b
        self.is_batch_share = is_batch_share
        self.range_indcs = range_indcs
        self.is_ensure_one = is_ensure_one
        self.is_beta_binomial = is_beta_binomial
        self.proba_uniform = proba_uniform

    def __call__(self, batch_size, n_possible_points):
        if self.range_indcs is not None:
            n_possible_points = self.range_indcs[1] - self.range_indcs[0]

        if np.random.uniform(size=1) < self.proba_uniform:
            n_indcs = random.randint(0, n_possible_points)
        elif self.is_beta_binomial:
            n_indcs = betabinom(n_possible_points, self.a, self.b).rvs()
        else:
            a = ratio_to_int(self.a, n_possible_points)
            b = ratio_to_int(self.b, n_possible_points)
            n_indcs = random.randint(a, b)

        if self.is_ensure_one and n_indcs < 1:
            n_indcs = 1

        if self.is_batch_share:
            indcs = torch.randperm(n_possible_points)[:n_indcs]
            indcs = indcs.unsqueeze(0).expand(batch_size, n_indcs)
        else:
            indcs = (
                np.arange(n_possible_points)
                .reshape(1, n_possible_points)
                .repeat(batch_size, axis=0)
            )
            indep_shuffle_(indcs, -1)
            indcs = torch.from_numpy(indcs[:, :n_indcs])

        if self.range_indcs is not None:
            indcs = indcs + self.range_indcs[0]
        return indcs


# --------------------------------------------------------------------------- #
# set (off-grid) splitter
# --------------------------------------------------------------------------- #


class CntxtTrgtGetter:
    """Split (X, y) into context and target sets by indices
    (reference :148-255)."""

    def __init__(
        self,
        contexts_getter=GetRandomIndcs(),
        targets_getter=get_all_indcs,
        is_add_cntxts_to_trgts=False,
    ):
        self.contexts_getter = contexts_getter
        self.targets_getter = targets_getter
        self.is_add_cntxts_to_trgts = is_add_cntxts_to_trgts

    def __call__(
        self, X, y=None, context_indcs=None, target_indcs=None, is_return_indcs=False
    ):
        batch_size, num_points = self.getter_inputs(X)

        if context_indcs is None:
            context_indcs = self.contexts_getter(batch_size, num_points)
        if target_indcs is None:
            target_indcs = self.targets_getter(batch_size, num_points)

        if self.is_add_cntxts_to_trgts:
            target_indcs = self.add_cntxts_to_trgts(
                num_points, target_indcs, context_indcs
            )

        X_pre_cntxt = self.preprocess_context(X)

        if is_return_indcs:
            return context_indcs, X_pre_cntxt, target_indcs, X

        X_cntxt, Y_cntxt = self.select(X_pre_cntxt, y, context_indcs)
        X_trgt, Y_trgt = self.select(X, y, target_indcs)
        return X_cntxt, Y_cntxt, X_trgt, Y_trgt

    def preprocess_context(self, X):
        return X

    def add_cntxts_to_trgts(self, num_points, target_indcs, context_indcs):
        """Append contexts to targets (may duplicate); cap at num_points."""
        target_indcs = torch.cat([target_indcs, context_indcs], dim=-1)
        return target_indcs[:, :num_points]

    def getter_inputs(self, X):
        batch_size, num_points, _ = X.shape
        return batch_size, num_points

    def select(self, X, y, indcs):
        """Gather-select the indexed points from X and y."""
        batch_size, num_points, x_dim = X.shape
        y_dim = y.size(-1)
        indcs = indcs.to(X.device)
        gx = indcs.unsqueeze(-1).expand(batch_size, -1, x_dim)
        gy = indcs.unsqueeze(-1).expand(batch_size, -1, y_dim)
        return (
            torch.gather(X, 1, gx).contiguous(),
            torch.gather(y, 1, gy).contiguous(),
        )


# --------------------------------------------------------------------------- #
# grid maskers
# --------------------------------------------------------------------------- #


class RandomMasker(GetRandomIndcs):
    """Random boolean mask over a grid (reference :259-278)."""

    def __call__(self, batch_size, mask_shape, **kwargs):
        n_possible_points = prod(mask_shape)
        nnz_indcs = super().__call__(batch_size, n_possible_points, **kwargs)

        if self.is_batch_share:
            mask = torch.zeros(n_possible_points).bool()
            mask = mask.unsqueeze(0).expand(batch_size, n_possible_points)
        else:
            mask = torch.zeros((batch_size, n_possible_points)).bool()

        mask.scatter_(1, nnz_indcs, True)
        return mask.view(batch_size, *mask_shape, 1).contiguous()


class ResolutionMasker:
    """Regular subsampling mask (resolution / `factor`) (reference :281-298)."""

    def __init__(self, factor):
        self.factor = factor

    def __call__(self, batch_size, mask_shape):
        mask = torch.zeros(mask_shape).bool()
        mask[self.factor // 2 :: self.factor, self.factor // 2 :: self.factor] = True
        return mask.unsqueeze(-1).expand(batch_size, *mask_shape, 1)


def and_masks(*masks):
    return functools.reduce(lambda a, b: a & b, masks)


def or_masks(*masks):
    return functools.reduce(lambda a, b: a | b, masks)


def not_masks(mask, not_mask):
    return and_masks(mask, ~not_mask)


def half_masker(batch_size, mask_shape, dim=0):
    """Mask the first half of `dim` (reference :319-326)."""
    mask = torch.zeros(mask_shape).bool()
    slcs = [slice(None)] * len(mask_shape)
    slcs[dim] = slice(0, mask_shape[dim] // 2)
    mask[slcs] = True
    return mask.unsqueeze(-1).expand(batch_size, *mask_shape, 1)


def no_masker(batch_size, mask_shape):
    """All-ones mask (reference :329-333)."""
    return torch.ones(1).bool().expand(batch_size, *mask_shape, 1)


# --------------------------------------------------------------------------- #
# grid (image) splitter
# --------------------------------------------------------------------------- #


class GridCntxtTrgtGetter(CntxtTrgtGetter):
    """Split grids (images) into context/target point sets
    (reference :336-452).  Mask nonzeros become [-1,1]-normalized coordinates
    scaled by `upscale_factor` (used for zero-shot scale extrapolation)."""

    def __init__(
        self,
        context_masker=RandomMasker(),
        target_masker=no_masker,
        upscale_factor=1,
        **kwargs,
    ):
        self.upscale_factor = upscale_factor
        super().__init__(
            contexts_getter=context_masker, targets_getter=target_masker, **kwargs
        )

    def __call__(
        self, X, y=None, context_mask=None, target_mask=None, is_return_masks=False,
        **kwargs,
    ):
        """X: [batch_size, y_dim, *grid_shape] (channels-first grid input)."""
        return super().__call__(
            channels_to_last_dim(X),
            context_indcs=context_mask,
            target_indcs=target_mask,
            is_return_indcs=is_return_masks,
            **kwargs,
        )

    def add_cntxts_to_trgts(self, grid_shape, target_mask, context_mask):
        return or_masks(target_mask, context_mask)

    def getter_inputs(self, X):
        batch_size, *grid_shape, y_dim = X.shape
        return batch_size, grid_shape

    def select(self, X, y, mask, extrapolation=1):
        """Mask-select grid values; coordinates = normalized nonzero indices."""
        batch_size, *grid_shape, y_dim = X.shape
        n_grid_dim = len(grid_shape)
        mask = mask.to(X.device)

        nonzero_idcs = mask.nonzero()
        # assumes the same count of nonzeros per batch element
        n_cntxt = mask[0].nonzero().size(0)

        X_masked = nonzero_idcs[:, 1:-1].view(batch_size, n_cntxt, n_grid_dim).float()
        for i, size in enumerate(grid_shape):
            X_masked[:, :, i] = X_masked[:, :, i] * (2 / (size - 1)) - 1
        X_masked = X_masked * self.upscale_factor

        mask = mask.expand(batch_size, *grid_shape, y_dim)
        Y_masked = X[mask].view(batch_size, n_cntxt, y_dim)
        return X_masked.contiguous(), Y_masked.contiguous()


class SuperresolutionCntxtTrgtGetter(GridCntxtTrgtGetter):
    """Context = downsampled(+nearest-upsampled) image, target = full image
    (reference :455-490)."""

    def __init__(self, resolution_factor=1 / 4, downsample_mode="area", **kwargs):
        self.resolution_factor = resolution_factor
        self.downsample_mode = downsample_mode
        super().__init__(
            context_masker=ResolutionMasker(factor=int(1 / self.resolution_factor)),
            target_masker=no_masker,
            **kwargs,
        )

    def preprocess_context(self, X):
        X = channels_to_2nd_dim(X)
        X_down = F.interpolate(
            X, scale_factor=self.resolution_factor, mode=self.downsample_mode
        )
        X_lowres = F.interpolate(
            X_down, scale_factor=int(1 / self.resolution_factor), mode="nearest"
        )
        return channels_to_last_dim(X_lowres)
