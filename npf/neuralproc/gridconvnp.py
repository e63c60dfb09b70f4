"""On-the-grid convolutional neural processes (image case).

Parity with /root/reference/npf/neuralproc/gridconvnp.py (GridConvCNP :28-178,
GridConvLNP :181-289).  X is a boolean mask over the grid, not coordinates;
context/target/induced points all share the grid, so `trgt_dependent` is the
identity.  The masked abs-depthwise-conv density encoder (reference :136-162)
is SURVEY.md §2.3 "Grid density encoder".
"""

import logging
from functools import partial

import torch
import torch.nn as nn

from npf.architectures import CNN, ResConvBlock
from npf.utils.helpers import (
    channels_to_2nd_dim,
    channels_to_last_dim,
    make_abs_conv,
)

from .base import LatentNeuralProcessFamily, NeuralProcessFamily
from .convnp import ConvCNP, ConvLNP
from .helpers import collapse_z_samples_batch

__all__ = ["GridConvCNP", "GridConvLNP"]

logger = logging.getLogger(__name__)


class GridConvCNP(NeuralProcessFamily):
    """ConvCNP special case with context/target/induced on one shared grid.

    x_dim must be 1 or y_dim (masks multiply Y with broadcasting,
    reference gridconvnp.py:103-105).
    """

    _valid_paths = ["deterministic"]

    def __init__(
        self,
        x_dim,
        y_dim,
        # depthwise |W| conv: output interpretable as a density
        Conv=lambda y_dim: make_abs_conv(nn.Conv2d)(
            y_dim, y_dim, groups=y_dim, kernel_size=11, padding=11 // 2, bias=False
        ),
        CNN=partial(
            CNN,
            ConvBlock=ResConvBlock,
            Conv=nn.Conv2d,
            n_blocks=3,
            Normalization=nn.Identity,
            is_chan_last=True,
            kernel_size=11,
        ),
        **kwargs,
    ):
        assert (
            x_dim == 1 or x_dim == y_dim
        ), "Ensure that feature masks can be multiplied with Y"

        if "Decoder" in kwargs and kwargs["Decoder"] != nn.Identity:
            logger.warning(
                "`Decoder` was given to `GridConvCNP`; use "
                "`discard_ith_arg(Decoder, i=0)` to stay translation equivariant."
            )

        kwargs["encoded_path"] = kwargs.get("encoded_path", "deterministic")
        super().__init__(x_dim, y_dim, x_transf_dim=None, XEncoder=nn.Identity, **kwargs)

        self.CNN = CNN
        self.conv = Conv(y_dim)
        # 2x channels: normalized signal ++ density/confidence
        self.resizer = nn.Linear(self.y_dim * 2, self.r_dim)
        self.induced_to_induced = CNN(self.r_dim)
        self.reset_parameters()

    dflt_Modules = ConvCNP.dflt_Modules

    def _fused_density_ok(self, X):
        from npf.ops import has_extension

        conv = self.conv
        return (
            X.is_cuda
            and has_extension()
            and isinstance(conv, nn.Conv2d)
            and conv.bias is None
            and conv.groups == conv.in_channels
            and conv.padding_mode == "zeros"
            and conv.dilation == (1, 1)
            and "AbsConv" in type(conv).__name__
        )

    def cntxt_to_induced(self, mask_cntxt, X):
        """Masked abs-depthwise conv pair -> normalized signal + density
        (reference gridconvnp.py:136-162)."""
        # channels to 2nd dim for the convolution
        X = channels_to_2nd_dim(X)
        mask_cntxt = channels_to_2nd_dim(mask_cntxt).float()

        if self._fused_density_ok(X):
            # ONE HIP kernel (csrc/npf_hip/griddensity.hip) for the masked
            # abs-conv pair + divide + concat
            from npf.ops import grid_density

            out = grid_density(X, mask_cntxt, self.conv.weight)
        else:
            signal = self.conv(X * mask_cntxt)
            density = self.conv(mask_cntxt.expand_as(X))
            out = signal / torch.clamp(density, min=1e-5)
            out = torch.cat([out, density], dim=1)

        out = channels_to_last_dim(out)
        return self.resizer(out)  # [B, *grid, r_dim]

    def encode_globally(self, mask_cntxt, X):
        R_induced = self.cntxt_to_induced(mask_cntxt, X)
        return self.induced_to_induced(R_induced)

    def trgt_dependent_representation(self, _, __, R_induced, ___):
        # grid targets == induced points: identity, Z dim = 1
        return R_induced.unsqueeze(0)

    def set_extrapolation(self, min_max):
        raise NotImplementedError("GridConvCNP cannot be used for extrapolation.")


class GridConvLNP(LatentNeuralProcessFamily, GridConvCNP):
    """On-the-grid ConvLNP (reference gridconvnp.py:181-289)."""

    _valid_paths = ["latent", "both"]

    def __init__(
        self, x_dim, y_dim, CNNPostZ=None, encoded_path="latent", is_global=False,
        **kwargs,
    ):
        super().__init__(x_dim, y_dim, encoded_path=encoded_path, **kwargs)
        self.is_global = is_global
        if CNNPostZ is None:
            CNNPostZ = self.CNN
        self.induced_to_induced_post_sampling = CNNPostZ(self.r_dim)
        self.reset_parameters()

    dflt_Modules = ConvLNP.dflt_Modules
    add_global_latent = ConvLNP.add_global_latent
    rep_to_lat_input = ConvLNP.rep_to_lat_input

    def trgt_dependent_representation(self, X_cntxt, z_samples, R_induced, X_trgt):
        batch_size, *grid_shape, _ = X_trgt.shape
        n_z_samples = z_samples.size(0)

        if self.encoded_path == "latent":
            z_samples = collapse_z_samples_batch(z_samples)

            # NOTE: unlike ConvLNP, the global latent is added BEFORE the
            # post-sampling CNN here (reference gridconvnp.py:253-256)
            if self.is_global:
                z_samples = self.add_global_latent(z_samples)

            if self.z_dim != self.r_dim:
                z_samples = self.reshaper_z(z_samples)

            R_trgt = self.induced_to_induced_post_sampling(z_samples)

        else:  # "both"
            z_samples = z_samples.view(
                n_z_samples, batch_size, *([1] * len(grid_shape)), self.r_dim
            ).expand(n_z_samples, batch_size, *grid_shape, self.r_dim)
            R_induced = self.merge_r_z(R_induced, z_samples)
            R_induced = collapse_z_samples_batch(R_induced)
            R_trgt = self.induced_to_induced_post_sampling(R_induced)

        return R_trgt.view(n_z_samples, batch_size, *grid_shape, self.r_dim)
