"""Attentive conditional / latent neural processes.

Parity with /root/reference/npf/neuralproc/attnnp.py (AttnCNP :27-131,
AttnLNP :134-202).  The context->target cross-attention
(`attender(K=X_cntxt, Q=X_trgt, V=R)`) is the hot loop (SURVEY.md §2.3 row
"Cross-attention"): on GPU it runs through the fused HIP MHA kernel inside
the transformer attender.
"""

import logging

import torch

from npf.architectures import SelfAttention, get_attender, merge_flat_input

from .base import LatentNeuralProcessFamily, NeuralProcessFamily
from .np import CNP

__all__ = ["AttnCNP", "AttnLNP"]

logger = logging.getLogger(__name__)


class AttnCNP(NeuralProcessFamily):
    """Attentive CNP: deterministic Attentive Neural Process (Kim et al. 2019).

    Keeps one representation per context point (reference attnnp.py:105-116)
    and cross-attends targets to contexts (:118-131).
    """

    _valid_paths = ["deterministic"]

    def __init__(
        self,
        x_dim,
        y_dim,
        XYEncoder=None,
        attention="scaledot",
        attention_kwargs={},
        self_attention_kwargs={},
        is_self_attn=False,
        **kwargs,
    ):
        kwargs["encoded_path"] = kwargs.get("encoded_path", "deterministic")
        super().__init__(x_dim, y_dim, **kwargs)

        self.is_self_attn = is_self_attn
        if self.is_self_attn:
            XYEncoder = merge_flat_input(
                SelfAttention, is_sum_merge=True, **self_attention_kwargs
            )
        elif XYEncoder is None:
            XYEncoder = self.dflt_Modules["XYEncoder"]

        self.xy_encoder = XYEncoder(self.x_transf_dim, self.y_dim, self.r_dim)
        self.attender = get_attender(
            attention, self.x_transf_dim, self.r_dim, self.r_dim, **attention_kwargs
        )
        self.reset_parameters()

    dflt_Modules = CNP.dflt_Modules

    def encode_globally(self, X_cntxt, Y_cntxt):
        batch_size, n_cntxt, _ = X_cntxt.shape
        if n_cntxt == 0:
            # empty context => zero per-target representation downstream
            return torch.zeros(batch_size, 0, self.r_dim, device=X_cntxt.device)
        # one representation per context point: [B, C, r_dim]
        return self.xy_encoder(X_cntxt, Y_cntxt)

    def trgt_dependent_representation(self, X_cntxt, _, R, X_trgt):
        batch_size, n_cntxt, _ = X_cntxt.shape
        if n_cntxt == 0:
            R_trgt = torch.zeros(
                batch_size, X_trgt.size(1), self.r_dim, device=R.device
            )
        else:
            # cross-attention: keys = contexts, queries = targets, values = R
            R_trgt = self.attender(X_cntxt, X_trgt, R)
        return R_trgt.unsqueeze(0)  # Z dim = 1


class AttnLNP(LatentNeuralProcessFamily, AttnCNP):
    """Attentive (latent) Neural Process (Kim et al. 2019): deterministic
    cross-attention path merged with a mean-pooled latent path
    (encoded_path="both", reference attnnp.py:134-202)."""

    _valid_paths = ["both"]

    def __init__(self, x_dim, y_dim, **kwargs):
        super().__init__(x_dim, y_dim, encoded_path="both", **kwargs)

    @property
    def dflt_Modules(self):
        dflt_Modules = AttnCNP.dflt_Modules.__get__(self)
        dflt_Modules.update(LatentNeuralProcessFamily.dflt_Modules.__get__(self))
        return dflt_Modules

    def rep_to_lat_input(self, R):
        batch_size, n_cntxt, _ = R.shape
        if n_cntxt == 0:
            # empty context => zero pooled representation (reference :175-177)
            R = torch.zeros(batch_size, 1, self.r_dim, device=R.device)
        # per-context representations -> single latent input via mean-pool
        return torch.mean(R, dim=1, keepdim=True)

    def trgt_dependent_representation(self, X_cntxt, z_samples, R, X_trgt):
        batch_size, n_trgt, _ = X_trgt.shape
        n_z_samples = z_samples.size(0)

        # latent path broadcast over targets: [Z, B, T, z_dim]
        z_samples = z_samples.expand(n_z_samples, batch_size, n_trgt, self.z_dim)

        # deterministic path: the cross-attention output, ignoring z
        # (reference attnnp.py:183-196)
        R_trgt_det = AttnCNP.trgt_dependent_representation(
            self, X_cntxt, None, R, X_trgt
        ).squeeze(0)

        # merge both paths: [Z, B, T, r_dim]
        return self.merge_r_z(R_trgt_det, z_samples)
