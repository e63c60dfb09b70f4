"""Vanilla (DeepSets mean-pool) conditional / latent neural processes.

Parity with /root/reference/npf/neuralproc/np.py (CNP :19-110, LNP :113-163).
The per-pair MLP + mean-pool encoder is SURVEY.md §2.3 "DeepSets encoder":
on GPU it runs as batched hipBLASLt GEMMs with the mean-pool as a fused
reduction epilogue.
"""

import logging
from functools import partial

import torch

from npf.architectures import MLP, merge_flat_input

from .base import LatentNeuralProcessFamily, NeuralProcessFamily

logger = logging.getLogger(__name__)

__all__ = ["CNP", "LNP"]


class CNP(NeuralProcessFamily):
    """Conditional Neural Process (Garnelo et al. 2018).

    `XYEncoder(x_transf_dim, y_dim, r_dim)` encodes each (x, y) context pair;
    the global representation is the mean over pairs (reference np.py:86-101).
    """

    _valid_paths = ["deterministic"]

    def __init__(self, x_dim, y_dim, XYEncoder=None, **kwargs):
        kwargs["encoded_path"] = kwargs.get("encoded_path", "deterministic")
        super().__init__(x_dim, y_dim, **kwargs)
        if XYEncoder is None:
            XYEncoder = self.dflt_Modules["XYEncoder"]
        self.xy_encoder = XYEncoder(self.x_transf_dim, self.y_dim, self.r_dim)
        self.reset_parameters()

    @property
    def dflt_Modules(self):
        dflt_Modules = NeuralProcessFamily.dflt_Modules.__get__(self)
        SubXYEncoder = partial(
            MLP, n_hidden_layers=2, is_force_hid_smaller=True, hidden_size=self.r_dim
        )
        dflt_Modules["XYEncoder"] = merge_flat_input(SubXYEncoder, is_sum_merge=True)
        return dflt_Modules

    def encode_globally(self, X_cntxt, Y_cntxt):
        batch_size, n_cntxt, _ = X_cntxt.shape
        # per-pair encodings [B, C, r_dim] -> mean-pool to [B, 1, r_dim]
        R_cntxt = self.xy_encoder(X_cntxt, Y_cntxt)
        R = torch.mean(R_cntxt, dim=1, keepdim=True)
        if n_cntxt == 0:
            # empty context => zero global representation (reference np.py:97-99)
            R = torch.zeros(batch_size, 1, self.r_dim, device=R_cntxt.device)
        return R

    def trgt_dependent_representation(self, _, __, R, X_trgt):
        batch_size, n_trgt, _ = X_trgt.shape
        # broadcast the single global representation over targets; Z dim = 1
        return R.expand(batch_size, n_trgt, self.r_dim).unsqueeze(0)


class LNP(LatentNeuralProcessFamily, CNP):
    """(Latent) Neural Process (Garnelo et al. 2018; Kim et al. 2019 for
    encoded_path="both")."""

    def __init__(self, x_dim, y_dim, encoded_path="latent", **kwargs):
        super().__init__(x_dim, y_dim, encoded_path=encoded_path, **kwargs)

    @property
    def dflt_Modules(self):
        # merged defaults (the reference's LNP cannot be built without an
        # explicit XYEncoder; this is a compat-safe extension)
        dflt_Modules = CNP.dflt_Modules.__get__(self)
        dflt_Modules.update(LatentNeuralProcessFamily.dflt_Modules.__get__(self))
        return dflt_Modules

    def trgt_dependent_representation(self, _, z_samples, R, X_trgt):
        batch_size, n_trgt, _ = X_trgt.shape
        n_z_samples = z_samples.size(0)

        if self.encoded_path == "both":
            # [Z, B, 1, r_dim]
            R_trgt = self.merge_r_z(R, z_samples)
        else:  # "latent"
            R_trgt = z_samples
            if self.z_dim != self.r_dim:
                R_trgt = self.reshaper_z(R_trgt)

        return R_trgt.expand(n_z_samples, batch_size, n_trgt, self.r_dim)
