"""Convolutional (off-the-grid, translation-equivariant) neural processes.

Parity with /root/reference/npf/neuralproc/convnp.py (ConvCNP :26-181,
ConvLNP :184-335).  The two SetConvs (grid / ungrid) and the post-latent CNN
are the heaviest ops (SURVEY.md §3.2 "ConvLNP trace"); SetConv runs on the
fused HIP kernel (npf/ops), the CNN through MIOpen.
"""

import logging
from functools import partial

import torch
import torch.nn as nn

from npf.architectures import CNN, ResConvBlock, SetConv, discard_ith_arg

from .base import LatentNeuralProcessFamily, NeuralProcessFamily
from .helpers import (
    collapse_z_samples_batch,
    pool_and_replicate_middle,
    replicate_z_samples,
)

logger = logging.getLogger(__name__)

__all__ = ["ConvCNP", "ConvLNP"]


class ConvCNP(NeuralProcessFamily):
    """Convolutional CNP (Gordon et al. 2019).

    Context -> induced grid (SetConv) -> CNN -> targets (SetConv ungrid).
    Induced points are a regular grid on [-1.5, 1.5] (0.5 margin each side
    against boundary effects, reference convnp.py:102-104); the grid is a
    plain tensor attribute, NOT a buffer, so it is absent from `state_dict`
    exactly like the reference checkpoint format.
    """

    _valid_paths = ["deterministic"]

    def __init__(
        self,
        x_dim,
        y_dim,
        density_induced=128,
        Interpolator=SetConv,
        CNN=partial(
            CNN,
            ConvBlock=ResConvBlock,
            Conv=nn.Conv1d,
            n_blocks=3,
            Normalization=nn.Identity,
            is_chan_last=True,
            kernel_size=11,
        ),
        **kwargs,
    ):
        if "Decoder" in kwargs and kwargs["Decoder"] != nn.Identity:
            logger.warning(
                "`Decoder` was given to `ConvCNP`. To be translation equivariant "
                "you should disregard the first argument, e.g. via "
                "`discard_ith_arg(Decoder, i=0)` (the default when no Decoder "
                "is provided)."
            )

        kwargs["encoded_path"] = kwargs.get("encoded_path", "deterministic")
        super().__init__(x_dim, y_dim, x_transf_dim=None, XEncoder=nn.Identity, **kwargs)

        self.density_induced = density_induced
        self.X_induced = torch.linspace(-1.5, 1.5, int(self.density_induced * 3))
        self.CNN = CNN

        self.cntxt_to_induced = Interpolator(self.x_dim, self.y_dim, self.r_dim)
        self.induced_to_induced = CNN(self.r_dim)
        self.induced_to_trgt = Interpolator(self.x_dim, self.r_dim, self.r_dim)
        self.reset_parameters()

    @property
    def n_induced(self):
        # property because `set_extrapolation` regrids
        return len(self.X_induced)

    @property
    def dflt_Modules(self):
        dflt_Modules = NeuralProcessFamily.dflt_Modules.__get__(self)
        # decoder must not see x to stay translation equivariant
        dflt_Modules["Decoder"] = discard_ith_arg(dflt_Modules["SubDecoder"], i=0)
        return dflt_Modules

    def _get_X_induced(self, X):
        batch_size = X.size(0)
        self.X_induced = self.X_induced.to(X.device)  # one-time device move
        return self.X_induced.view(1, -1, 1).expand(
            batch_size, self.n_induced, self.x_dim
        )

    def encode_globally(self, X_cntxt, Y_cntxt):
        batch_size, n_cntxt, _ = X_cntxt.shape
        X_induced = self._get_X_induced(X_cntxt)

        if n_cntxt == 0:
            # empty context: zero representation (density channel is zero
            # too) — skip the SetConv entirely rather than launching a
            # degenerate K=0 kernel
            R_induced = torch.zeros(
                batch_size, self.n_induced, self.r_dim, device=X_cntxt.device
            )
        else:
            # context -> induced grid: [B, M, r_dim]
            R_induced = self.cntxt_to_induced(X_cntxt, X_induced, Y_cntxt)

        # induced -> induced: the CNN stack
        return self.induced_to_induced(R_induced)

    def trgt_dependent_representation(self, X_cntxt, z_samples, R_induced, X_trgt):
        X_induced = self._get_X_induced(X_cntxt)
        # induced grid -> targets: [B, T, r_dim]
        R_trgt = self.induced_to_trgt(X_induced, X_trgt, R_induced)
        return R_trgt.unsqueeze(0)

    def set_extrapolation(self, min_max):
        """Re-grid induced points over an extended range at train density
        (reference convnp.py:170-181)."""
        lo = min_max[0] - 0.5
        hi = min_max[1] + 0.5
        self.X_induced = torch.linspace(
            lo, hi, int(self.density_induced * (hi - lo))
        )


class ConvLNP(LatentNeuralProcessFamily, ConvCNP):
    """Convolutional LNP (Foong et al. 2020): per-induced-point latents with a
    post-sampling CNN for coherent samples, optional global latent."""

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

    @property
    def dflt_Modules(self):
        dflt_Modules = ConvCNP.dflt_Modules.__get__(self)
        dflt_Modules.update(LatentNeuralProcessFamily.dflt_Modules.__get__(self))
        # small linear decoder: the post-z CNN already mixes (reference :249)
        dflt_Modules["Decoder"] = discard_ith_arg(nn.Linear, i=0)
        return dflt_Modules

    def rep_to_lat_input(self, R):
        batch_size = R.size(0)
        if self.encoded_path == "latent":
            # one latent per induced point
            return R
        # "both": single pooled latent
        return R.view(batch_size, -1, self.r_dim).mean(dim=1, keepdim=True)

    def trgt_dependent_representation(self, X_cntxt, z_samples, R_induced, X_trgt):
        batch_size, n_trgt, _ = X_trgt.shape
        n_z_samples = z_samples.size(0)

        X_induced = self._get_X_induced(X_cntxt)
        # fold Z into batch for the CNN / SetConv (Z*B leading dim)
        X_induced = collapse_z_samples_batch(
            replicate_z_samples(X_induced, n_z_samples)
        )
        X_trgt = collapse_z_samples_batch(replicate_z_samples(X_trgt, n_z_samples))

        if self.encoded_path == "latent":
            z_samples = collapse_z_samples_batch(z_samples)
            if self.z_dim != self.r_dim:
                z_samples = self.reshaper_z(z_samples)

            # post-sampling mixing CNN => coherent function samples
            z_samples = self.induced_to_induced_post_sampling(z_samples)

            if self.is_global:
                z_samples = self.add_global_latent(z_samples)

            # [Z*B, T, r_dim]
            R_trgt = self.induced_to_trgt(X_induced, X_trgt, z_samples)

        else:  # "both"
            z_samples = z_samples.expand(
                n_z_samples, batch_size, self.n_induced, self.z_dim
            )
            R_induced = self.merge_r_z(R_induced, z_samples)
            R_induced = collapse_z_samples_batch(R_induced)
            R_induced = self.induced_to_induced_post_sampling(R_induced)
            R_trgt = self.induced_to_trgt(X_induced, X_trgt, R_induced)

        return R_trgt.view(n_z_samples, batch_size, n_trgt, self.r_dim)

    def add_global_latent(self, z_samples):
        """Split channels; mean-pool half into a global latent and broadcast
        back (reference convnp.py:322-335)."""
        local_z, global_z = z_samples.split(z_samples.shape[-1] // 2, dim=-1)
        global_z = pool_and_replicate_middle(global_z)
        return torch.cat([local_z, global_z], dim=-1)
