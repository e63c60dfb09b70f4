"""Context/target episode sampler — turns raw functions (or images) into
meta-learning tasks.

Capability parity with the reference splitters
(/root/reference/npf/utils/datasplit.py: index getters :30-145,
CntxtTrgtGetter :148-255, maskers :259-333, GridCntxtTrgtGetter :336-452,
SuperresolutionCntxtTrgtGetter :455-490), re-designed MI355X-first:

- **torch-native and device-capable**: every getter draws its indices with
  torch RNG and accepts a `device`, so episode construction can run on the
  GPU against a resident meta-batch (no numpy round-trip, no per-batch H2D of
  gathered sets).  `CntxtTrgtGetter` forwards `X.device` to its getters.
- The per-row random subsets come from one `argsort(rand(B, N))` — a single
  batched kernel on GPU — instead of per-row CPU shuffles.
- Count distributions are preserved: one count per call shared by the batch
  (uniform-integer, beta-binomial, or a `proba_uniform` mixture), because
  rectangular [B, n] episodes are part of the model ABI.
- `StratifiedCountIndcs` (new) cycles counts deterministically over [a, b]
  for low-variance evaluation: same marginal as the uniform draw once every
  count is visited equally often, but the count noise — the dominant noise
  term of mean-LL estimates — is removed.
"""

import functools

import torch
import torch.nn.functional as F

from .helpers import prod, ratio_to_int

__all__ = [
    "get_all_indcs",
    "GetRangeIndcs",
    "GetRandomIndcs",
    "StratifiedCountIndcs",
    "CntxtTrgtGetter",
    "RandomMasker",
    "half_masker",
    "no_masker",
    "GridCntxtTrgtGetter",
]


# --------------------------------------------------------------------------- #
# index getters: (batch_size, n_possible_points[, device]) -> [B, n] long
# --------------------------------------------------------------------------- #


def get_all_indcs(batch_size, n_possible_points, device=None):
    """Every index, shared across the batch."""
    return (
        torch.arange(n_possible_points, device=device)
        .unsqueeze(0)
        .expand(batch_size, -1)
    )


class GetRangeIndcs:
    """Indices of a fixed [start, stop) range, shared across the batch."""

    def __init__(self, arange):
        self.arange = arange

    def __call__(self, batch_size, n_possible_points, device=None):
        idx = torch.arange(*self.arange, device=device)
        return idx.unsqueeze(0).expand(batch_size, -1)


class GetIndcsMerger:
    """Concatenation of several index getters along the point dim."""

    def __init__(self, indexers):
        self.indexers = indexers

    def __call__(self, batch_size, n_possible_points, device=None):
        parts = [ix(batch_size, n_possible_points, device=device) for ix in self.indexers]
        return torch.cat(parts, dim=1)


def _rand_subsets(batch_size, n_possible_points, n, device=None, shared=False):
    """[B, n] random index subsets (without replacement).

    `shared=True` reuses one subset for the whole batch.  Implemented as an
    argsort over uniforms: one batched kernel, runs on any device.
    """
    rows = 1 if shared else batch_size
    keys = torch.rand(rows, n_possible_points, device=device)
    subsets = keys.argsort(dim=-1)[:, :n]
    if shared:
        subsets = subsets.expand(batch_size, n)
    return subsets


class GetRandomIndcs:
    """Random index subsets with a random per-call count.

    Count distribution (one draw per call, shared by the batch so episodes
    stay rectangular):

    - default: uniform integer in [a, b] (ratios of N when < 1);
    - `is_beta_binomial`: count ~ BetaBinomial(N; a, b), sampled as
      p ~ Beta(a, b) then count ~ Binomial(N, p) — the standard mixture
      identity, so the marginal matches scipy's `betabinom(N, a, b)`;
    - with prob `proba_uniform`, fully uniform in [0, N] regardless of a/b.

    `is_batch_share` additionally shares *which* indices across the batch;
    `range_indcs` restricts to an index window; `is_ensure_one` floors the
    count at 1.
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

    def sample_count(self, n_possible_points):
        if self.proba_uniform > 0 and float(torch.rand(())) < self.proba_uniform:
            n = int(torch.randint(0, n_possible_points + 1, ()))
        elif self.is_beta_binomial:
            p = torch.distributions.Beta(self.a, self.b).sample()
            n = int(torch.distributions.Binomial(n_possible_points, p).sample())
        else:
            lo = ratio_to_int(self.a, n_possible_points)
            hi = ratio_to_int(self.b, n_possible_points)
            n = int(torch.randint(lo, hi + 1, ()))
        if self.is_ensure_one:
            n = max(n, 1)
        return n

    def __call__(self, batch_size, n_possible_points, device=None):
        window = n_possible_points
        if self.range_indcs is not None:
            window = self.range_indcs[1] - self.range_indcs[0]

        n = self.sample_count(window)
        subsets = _rand_subsets(
            batch_size, window, n, device=device, shared=self.is_batch_share
        )
        if self.range_indcs is not None:
            subsets = subsets + self.range_indcs[0]
        return subsets


class StratifiedCountIndcs(GetRandomIndcs):
    """Deterministic count schedule for low-variance evaluation.

    Cycles the context count through a, a+1, ..., b, a, ... across calls
    (indices themselves stay random).  Over any whole number of cycles the
    count marginal equals `GetRandomIndcs(a, b)`'s uniform draw, but the
    count-sampling noise — which dominates the variance of mean-LL
    estimates (per-task LL swings hundreds of nats between 0 and max
    contexts) — is stratified away.
    """

    def __init__(self, a=0, b=50, **kwargs):
        super().__init__(a=a, b=b, **kwargs)
        self._call_idx = 0

    def sample_count(self, n_possible_points):
        lo = ratio_to_int(self.a, n_possible_points)
        hi = ratio_to_int(self.b, n_possible_points)
        n = lo + self._call_idx % (hi - lo + 1)
        self._call_idx += 1
        if self.is_ensure_one:
            n = max(n, 1)
        return n

    def reset(self):
        self._call_idx = 0


# --------------------------------------------------------------------------- #
# set (off-grid) splitter
# --------------------------------------------------------------------------- #


def _call_getter(getter, batch_size, num_points, device):
    """Invoke an index getter / masker, passing `device` when it takes one
    (user-supplied callables may have the bare 2-arg signature)."""
    try:
        return getter(batch_size, num_points, device=device)
    except TypeError:
        return getter(batch_size, num_points)


class CntxtTrgtGetter:
    """Split (X, y) point sets into a context and a target set.

    X: [B, N, x_dim], y: [B, N, y_dim].  Returns
    (X_cntxt, Y_cntxt, X_trgt, Y_trgt); episodes are rectangular ([B, n, .])
    and live on X's device.
    """

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
        device = X.device

        if context_indcs is None:
            context_indcs = _call_getter(
                self.contexts_getter, batch_size, num_points, device
            )
        if target_indcs is None:
            target_indcs = _call_getter(
                self.targets_getter, batch_size, num_points, device
            )

        if self.is_add_cntxts_to_trgts:
            target_indcs = self.add_cntxts_to_trgts(
                num_points, target_indcs, context_indcs
            )

        X_pre_cntxt = self.preprocess_context(X)

        if is_return_indcs:
            # raw selections (indices + source tensors): used by callers that
            # need the mask/index view of the episode, e.g. viz overlays
            return context_indcs, X_pre_cntxt, target_indcs, X

        X_cntxt, Y_cntxt = self.select(X_pre_cntxt, y, context_indcs)
        X_trgt, Y_trgt = self.select(X, y, target_indcs)
        return X_cntxt, Y_cntxt, X_trgt, Y_trgt

    def preprocess_context(self, X):
        return X

    def add_cntxts_to_trgts(self, num_points, target_indcs, context_indcs):
        """Append the context indices to the targets (capped at N points)."""
        joint = torch.cat([target_indcs, context_indcs], dim=-1)
        return joint[:, :num_points]

    def getter_inputs(self, X):
        return X.shape[0], X.shape[1]

    def select(self, X, y, indcs):
        """Batched gather of the indexed points from X and y."""
        indcs = indcs.to(X.device)
        sel_x = torch.take_along_dim(X, indcs.unsqueeze(-1), dim=1)
        sel_y = torch.take_along_dim(y, indcs.unsqueeze(-1), dim=1)
        return sel_x.contiguous(), sel_y.contiguous()


# --------------------------------------------------------------------------- #
# grid maskers: (batch_size, mask_shape) -> [B, *grid, 1] bool
# --------------------------------------------------------------------------- #


class RandomMasker(GetRandomIndcs):
    """Random boolean grid mask with a GetRandomIndcs-distributed count."""

    def __call__(self, batch_size, mask_shape, device=None, **kwargs):
        n_pts = prod(mask_shape)
        nnz = super().__call__(batch_size, n_pts, device=device, **kwargs)
        rows = 1 if self.is_batch_share else batch_size
        flat = torch.zeros(rows, n_pts, dtype=torch.bool, device=device)
        flat.scatter_(1, nnz[:rows], True)
        mask = flat.unsqueeze(-1).view(rows, *mask_shape, 1)
        if self.is_batch_share:
            mask = mask.expand(batch_size, *mask_shape, 1)
        return mask.contiguous()


class ResolutionMasker:
    """Every `factor`-th grid point (phase-centered), shared across batch."""

    def __init__(self, factor):
        self.factor = factor

    def __call__(self, batch_size, mask_shape, device=None):
        mask = torch.zeros(*mask_shape, dtype=torch.bool, device=device)
        sl = [slice(self.factor // 2, None, self.factor)] * len(mask_shape)
        mask[tuple(sl)] = True
        return mask.unsqueeze(-1).expand(batch_size, *mask_shape, 1)


def and_masks(*masks):
    return functools.reduce(lambda a, b: a & b, masks)


def or_masks(*masks):
    return functools.reduce(lambda a, b: a | b, masks)


def not_masks(mask, not_mask):
    return and_masks(mask, ~not_mask)


def half_masker(batch_size, mask_shape, dim=0, device=None):
    """First half of dimension `dim`."""
    mask = torch.zeros(*mask_shape, dtype=torch.bool, device=device)
    sl = [slice(None)] * len(mask_shape)
    sl[dim] = slice(0, mask_shape[dim] // 2)
    mask[tuple(sl)] = True
    return mask.unsqueeze(-1).expand(batch_size, *mask_shape, 1)


def no_masker(batch_size, mask_shape, device=None):
    """Everything."""
    return torch.ones(1, dtype=torch.bool, device=device).expand(
        batch_size, *mask_shape, 1
    )


# --------------------------------------------------------------------------- #
# grid (image) splitter
# --------------------------------------------------------------------------- #


class GridCntxtTrgtGetter(CntxtTrgtGetter):
    """Episode sampler for gridded functions (images).

    X: [B, y_dim, *grid] channels-first.  Masked grid positions become
    point sets: coordinates are the grid positions normalized to [-1, 1]
    per axis and scaled by `upscale_factor` (zero-shot scale extrapolation),
    values are the channel vectors at those positions.
    """

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
        # channels-last view of the grid: [B, *grid, y_dim]
        Xl = X.movedim(1, -1)
        return super().__call__(
            Xl,
            context_indcs=context_mask,
            target_indcs=target_mask,
            is_return_indcs=is_return_masks,
            **kwargs,
        )

    def add_cntxts_to_trgts(self, grid_shape, target_mask, context_mask):
        return or_masks(target_mask, context_mask)

    def getter_inputs(self, X):
        return X.shape[0], list(X.shape[1:-1])

    def _grid_coords(self, grid_shape, device):
        """[prod(grid), n_dims] coordinates, each axis linspaced over [-1,1]
        (row-major order, matching a flattened mask)."""
        axes = [
            torch.linspace(-1.0, 1.0, s, device=device) * self.upscale_factor
            for s in grid_shape
        ]
        mesh = torch.meshgrid(*axes, indexing="ij")
        return torch.stack([m.reshape(-1) for m in mesh], dim=-1)

    def select(self, X, y, mask, extrapolation=1):
        """Mask-select grid values and their normalized coordinates.

        `mask`: [B or 1, *grid, 1] bool with the same count per batch row
        (maskers guarantee it), so the output stays rectangular.
        """
        batch_size = X.shape[0]
        grid_shape = list(X.shape[1:-1])
        y_dim = X.shape[-1]
        mask = mask.to(X.device)

        flat = mask.reshape(mask.shape[0], -1)  # [B or 1, P] row-major
        if flat.shape[0] == 1:
            flat = flat.expand(batch_size, -1)
        n_sel = int(flat[0].sum())

        coords = self._grid_coords(grid_shape, X.device)  # [P, D]
        # row-major masked gather; equal counts per row make the view valid
        X_sel = (
            coords.unsqueeze(0)
            .expand(batch_size, -1, -1)[flat]
            .view(batch_size, n_sel, coords.shape[-1])
        )
        Y_sel = X.reshape(batch_size, -1, y_dim)[flat].view(batch_size, n_sel, y_dim)
        return X_sel.contiguous(), Y_sel.contiguous()


class SuperresolutionCntxtTrgtGetter(GridCntxtTrgtGetter):
    """Context = the image seen through a lower resolution (area-downsampled
    then nearest-upsampled), target = the full-resolution image."""

    def __init__(self, resolution_factor=1 / 4, downsample_mode="area", **kwargs):
        self.resolution_factor = resolution_factor
        self.downsample_mode = downsample_mode
        super().__init__(
            context_masker=ResolutionMasker(factor=int(1 / self.resolution_factor)),
            target_masker=no_masker,
            **kwargs,
        )

    def preprocess_context(self, X):
        # X here is channels-last: go channels-first for interpolate
        Xc = X.movedim(-1, 1)
        down = F.interpolate(
            Xc, scale_factor=self.resolution_factor, mode=self.downsample_mode
        )
        lowres = F.interpolate(
            down, scale_factor=int(1 / self.resolution_factor), mode="nearest"
        )
        return lowres.movedim(1, -1)
