"""Template-method core of the Neural Process Family.

Parity with /root/reference/npf/neuralproc/base.py: the fixed 5-step forward
(:177-239), decode with loc/scale transforms (:327-367), and the latent
machinery of `LatentNeuralProcessFamily` (:374-575).  Subclasses override only
`encode_globally` and `trgt_dependent_representation` (plus latent hooks).

MI355X notes: the predictive distribution is a thin (loc, scale) pair wrapped
in Independent(Normal); its log-prob + target-sum reduction runs through the
fused HIP kernel in `npf.ops.gaussian_nll_sum` (see npf/losses.py).  Latent
rsample is a single fused mul-add over a Philox draw via torch.
"""

import abc
import os
from functools import partial

import torch
import torch.nn as nn
import torch.nn.functional as F

from npf.architectures import MLP, merge_flat_input
from npf.utils.helpers import MultivariateNormalDiag, isin_range
from npf.utils.initialization import weights_init

from .helpers import pool_and_replicate_middle

__all__ = ["NeuralProcessFamily", "LatentNeuralProcessFamily"]


class NeuralProcessFamily(nn.Module, abc.ABC):
    """Base class for NPF members.

    Parameters (reference base.py:33-99):

    x_dim, y_dim : int
        Feature / value dimensions.
    encoded_path : {"deterministic", "latent", "both"}
        Which representation path(s) feed the decoder.
    r_dim : int, optional
        Representation width.
    x_transf_dim : int, optional
        Encoded-X width (`-1` -> r_dim, `None` -> x_dim).
    is_heteroskedastic : bool, optional
        If False, pools the predicted scales over the target set
        (exact on fixed grids).
    XEncoder, Decoder : constructors, optional
        `XEncoder(x_dim, x_transf_dim)`; `Decoder(x_transf_dim, r_dim, 2*y_dim)`.
    PredictiveDistribution : callable, optional
        Built from (loc, scale); default diagonal Gaussian.
    p_y_loc_transformer, p_y_scale_transformer : callable, optional
        Sufficient-stat transforms; the default scale floor 0.01 + 0.99*softplus
        follows Le et al. 2018 (reference base.py:116).
    """

    _valid_paths = ["deterministic", "latent", "both"]

    def __init__(
        self,
        x_dim,
        y_dim,
        encoded_path,
        r_dim=128,
        x_transf_dim=-1,
        is_heteroskedastic=True,
        XEncoder=None,
        Decoder=None,
        PredictiveDistribution=MultivariateNormalDiag,
        p_y_loc_transformer=nn.Identity(),
        p_y_scale_transformer=lambda y_scale: 0.01 + 0.99 * F.softplus(y_scale),
    ):
        super().__init__()
        self.x_dim = x_dim
        self.y_dim = y_dim
        self.r_dim = r_dim
        self.is_heteroskedastic = is_heteroskedastic

        if x_transf_dim is None:
            self.x_transf_dim = self.x_dim
        elif x_transf_dim == -1:
            self.x_transf_dim = self.r_dim
        else:
            self.x_transf_dim = x_transf_dim

        self.encoded_path = encoded_path.lower()
        if self.encoded_path not in self._valid_paths:
            raise ValueError(f"Unknown encoded_path={self.encoded_path}.")

        XEncoder = XEncoder if XEncoder is not None else self.dflt_Modules["XEncoder"]
        Decoder = Decoder if Decoder is not None else self.dflt_Modules["Decoder"]

        self.x_encoder = XEncoder(self.x_dim, self.x_transf_dim)
        # 2x out channels: predictive loc and scale sufficient statistics
        self.decoder = Decoder(self.x_transf_dim, self.r_dim, self.y_dim * 2)

        self.PredictiveDistribution = PredictiveDistribution
        self.p_y_loc_transformer = p_y_loc_transformer
        self.p_y_scale_transformer = p_y_scale_transformer

        self.reset_parameters()

    def reset_parameters(self):
        weights_init(self)

    @property
    def dflt_Modules(self):
        SubDecoder = partial(MLP, n_hidden_layers=4, hidden_size=self.r_dim)
        return {
            "XEncoder": partial(MLP, n_hidden_layers=1, hidden_size=self.r_dim),
            "SubDecoder": SubDecoder,
            "Decoder": merge_flat_input(SubDecoder, is_sum_merge=True),
        }

    def forward(self, X_cntxt, Y_cntxt, X_trgt, Y_trgt=None):
        """Posterior predictive of target values given the context set.

        Returns (p_yCc, z_samples, q_zCc, q_zCct) exactly as the reference
        (base.py:177-239): p_yCc has batch shape [n_z_samples, B, *n_trgt] and
        event shape [y_dim]; the three latent outputs are None on the
        deterministic path.
        """
        self._validate_inputs(X_cntxt, Y_cntxt, X_trgt, Y_trgt)

        # step 1: positional encoding of the features
        X_cntxt = self.x_encoder(X_cntxt)
        X_trgt = self.x_encoder(X_trgt)

        # step 2: global context representation {R^u}_u
        R = self.encode_globally(X_cntxt, Y_cntxt)

        # step 3: latent path (optional)
        if self.encoded_path in ("latent", "both"):
            z_samples, q_zCc, q_zCct = self.latent_path(X_cntxt, R, X_trgt, Y_trgt)
        else:
            z_samples, q_zCc, q_zCct = None, None, None

        if self.encoded_path == "latent":
            R = None  # decoder must not see the deterministic path

        # step 4: target-dependent representation [Z, B, *n_trgt, r_dim]
        R_trgt = self.trgt_dependent_representation(X_cntxt, z_samples, R, X_trgt)

        # step 5: decode to the predictive distribution
        p_yCc = self.decode(X_trgt, R_trgt)
        return p_yCc, z_samples, q_zCc, q_zCct

    def _validate_inputs(self, X_cntxt, Y_cntxt, X_trgt, Y_trgt):
        """Training features must be rescaled to [-1,1] (reference base.py:241-247).

        On GPU tensors the check is skipped by default: `.all()` forces a
        host<->device sync every training step (and is illegal inside a
        hipGraph capture).  Set NPF_VALIDATE_INPUTS=1 to force it.
        """
        if self.training:
            if X_cntxt.is_cuda and os.environ.get("NPF_VALIDATE_INPUTS") != "1":
                return
            if not (isin_range(X_cntxt, [-1, 1]) and isin_range(X_trgt, [-1, 1])):
                raise ValueError(
                    f"Features during training should be in [-1,1]. "
                    f"{X_cntxt.min()} <= X_cntxt <= {X_cntxt.max()} ; "
                    f"{X_trgt.min()} <= X_trgt <= {X_trgt.max()}."
                )

    @abc.abstractmethod
    def encode_globally(self, X_cntxt, Y_cntxt):
        """Context set -> global representation [B, *n_rep, r_dim]."""

    @abc.abstractmethod
    def trgt_dependent_representation(self, X_cntxt, z_samples, R, X_trgt):
        """-> per-target representation [n_z_samples, B, *n_trgt, r_dim]."""

    def latent_path(self, X_cntxt, R, X_trgt, Y_trgt):
        raise NotImplementedError(
            f"`latent_path` not implemented. Cannot use encoded_path="
            f"{self.encoded_path} in such case."
        )

    def decode(self, X_trgt, R_trgt):
        """Decode target representations into the predictive distribution.

        Reference base.py:327-367: decoder emits 2*y_dim sufficient stats,
        split loc/scale, transformed, optionally scale-pooled (homoskedastic),
        wrapped in Independent(Normal, 1).
        """
        # [n_z_samples, B, *n_trgt, 2*y_dim]
        p_y_suffstat = self.decoder(X_trgt, R_trgt)
        p_y_loc, p_y_scale = p_y_suffstat.split(self.y_dim, dim=-1)

        p_y_loc = self.p_y_loc_transformer(p_y_loc)
        p_y_scale = self.p_y_scale_transformer(p_y_scale)

        if not self.is_heteroskedastic:
            # pool all scales over the target set (exact on constant grids)
            n_z_samples, batch_size, *n_trgt, y_dim = p_y_scale.shape
            p_y_scale = p_y_scale.reshape(n_z_samples * batch_size, *n_trgt, y_dim)
            p_y_scale = pool_and_replicate_middle(p_y_scale)
            p_y_scale = p_y_scale.reshape(n_z_samples, batch_size, *n_trgt, y_dim)

        return self.PredictiveDistribution(p_y_loc, p_y_scale)

    def set_extrapolation(self, min_max):
        """Configure the model for out-of-range prediction (model-specific)."""
        pass


class LatentNeuralProcessFamily(NeuralProcessFamily):
    """Latent NPF base (reference base.py:374-575).

    Adds: `latent_encoder` (r -> 2*z_dim sufficient stats), the q(z|.) scale
    transform 0.1 + 0.9*sigmoid (base.py:432), optional `r_z_merger` for
    encoded_path="both" (:450), `reshaper_z` when z_dim != r_dim (:456-458),
    and stochastic sample counts (scipy frozen RVs allowed, :475-490).
    """

    _valid_paths = ["latent", "both"]

    def __init__(
        self,
        *args,
        is_q_zCct=False,
        n_z_samples_train=32,
        n_z_samples_test=32,
        LatentEncoder=None,
        LatentDistribution=MultivariateNormalDiag,
        q_z_loc_transformer=nn.Identity(),
        q_z_scale_transformer=lambda z_scale: 0.1 + 0.9 * torch.sigmoid(z_scale),
        z_dim=None,
        **kwargs,
    ):
        super().__init__(*args, **kwargs)
        self.is_q_zCct = is_q_zCct
        self.n_z_samples_train = n_z_samples_train
        self.n_z_samples_test = n_z_samples_test
        self.z_dim = self.r_dim if z_dim is None else z_dim

        if LatentEncoder is None:
            LatentEncoder = self.dflt_Modules["LatentEncoder"]
        self.latent_encoder = LatentEncoder(self.r_dim, self.z_dim * 2)

        if self.encoded_path == "both":
            self.r_z_merger = nn.Linear(self.r_dim + self.z_dim, self.r_dim)

        self.LatentDistribution = LatentDistribution
        self.q_z_loc_transformer = q_z_loc_transformer
        self.q_z_scale_transformer = q_z_scale_transformer

        if self.z_dim != self.r_dim and self.encoded_path == "latent":
            self.reshaper_z = nn.Linear(self.z_dim, self.r_dim)

        self.reset_parameters()

    @property
    def dflt_Modules(self):
        dflt_Modules = NeuralProcessFamily.dflt_Modules.__get__(self)
        dflt_Modules["LatentEncoder"] = partial(
            MLP, n_hidden_layers=1, hidden_size=self.r_dim
        )
        return dflt_Modules

    def forward(self, *args, **kwargs):
        # resolve the (possibly stochastic) sample count once per call
        n = self.n_z_samples_train if self.training else self.n_z_samples_test
        try:
            self.n_z_samples = n.rvs()  # scipy frozen RV (e.g. for SUMO)
        except AttributeError:
            self.n_z_samples = n
        return super().forward(*args, **kwargs)

    def latent_path(self, X_cntxt, R, X_trgt, Y_trgt):
        """Infer q(z|C) (and q(z|C,T) when posterior sampling) and rsample.

        Reference base.py:495-514.
        """
        q_zCc = self.infer_latent_dist(X_cntxt, R)

        if self.is_q_zCct and Y_trgt is not None:
            # posterior sampling: re-encode the (super)set of targets
            R_from_trgt = self.encode_globally(X_trgt, Y_trgt)
            q_zCct = self.infer_latent_dist(X_trgt, R_from_trgt)
            sampling_dist = q_zCct
        else:
            q_zCct = None
            sampling_dist = q_zCc

        # [n_z_samples, B, *n_lat, z_dim]
        z_samples = self._rsample(sampling_dist, self.n_z_samples)
        return z_samples, q_zCc, q_zCct

    @staticmethod
    def _rsample(dist, n_z_samples):
        """rsample that stays valid inside a hipGraph capture.

        With the noise pool enabled (npf.ops.noise), the reparameterized
        draw is `loc + scale * eps` over a static pool buffer refreshed by
        the host between replays — torch RNG kernels inside a capture would
        replay frozen Philox state.  Default path: plain rsample.
        """
        from npf.ops import noise as _noise

        base = getattr(dist, "base_dist", None)
        if _noise.is_noise_pool_enabled() and base is not None:
            eps = _noise.pool_noise(
                (n_z_samples, *base.loc.shape), base.loc.device, base.loc.dtype
            )
            return base.loc + base.scale * eps
        return dist.rsample([n_z_samples])

    def infer_latent_dist(self, X, R):
        """R -> q(z) (reference base.py:516-547)."""
        R_lat_inp = self.rep_to_lat_input(R)
        q_z_suffstat = self.latent_encoder(R_lat_inp)
        q_z_loc, q_z_scale = q_z_suffstat.split(self.z_dim, dim=-1)
        q_z_loc = self.q_z_loc_transformer(q_z_loc)
        q_z_scale = self.q_z_scale_transformer(q_z_scale)
        return self.LatentDistribution(q_z_loc, q_z_scale)

    def rep_to_lat_input(self, R):
        """Map the n_rep representations to latent-input positions (dflt: id)."""
        return R

    def merge_r_z(self, R, z_samples):
        """ReLU(Linear([R; z])) merge of both paths (reference base.py:554-575)."""
        if R.shape != z_samples.shape:
            R = R.unsqueeze(0).expand(*z_samples.shape[:-1], self.r_dim)
        return torch.relu(self.r_z_merger(torch.cat((R, z_samples), dim=-1)))
