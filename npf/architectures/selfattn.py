"""Stacked self-attention encoder (used by the image Attn models).

Parity with /root/reference/npf/architectures/selfattn.py:10-100.
"""

import torch.nn as nn

from npf.utils.initialization import weights_init

from .attention import get_attender
from .encoders import RelativeSinusoidalEncodings, SinusoidalEncodings

__all__ = ["SelfAttention"]


class SelfAttention(nn.Module):
    """`n_attn_layers` stacked attenders applied set-wise, with optional
    absolute or relative positional encodings (reference selfattn.py:44-100)."""

    def __init__(
        self,
        x_dim,
        out_dim=None,
        n_attn_layers=2,
        attention="transformer",
        positional=None,
        position_dim=None,
        max_len=2000,
        **kwargs,
    ):
        super().__init__()
        self.positional = positional

        if self.positional == "absolute":
            self.pos_encoder = SinusoidalEncodings(position_dim, x_dim)
        elif self.positional == "relative":
            self.rel_pos_encoder = RelativeSinusoidalEncodings(position_dim, x_dim)
            kwargs["is_relative_pos"] = True
        elif self.positional is not None:
            raise ValueError(f"Unknown positional={positional}.")

        self.attn_layers = nn.ModuleList(
            get_attender(attention, x_dim, x_dim, x_dim, **kwargs)
            for _ in range(n_attn_layers)
        )

        self.is_resize = out_dim is not None
        if self.is_resize:
            self.resize = nn.Linear(x_dim, out_dim)

        self.reset_parameters()

    def reset_parameters(self):
        weights_init(self)

    def forward(self, X, positions=None):
        add_to_keys = 0
        if self.positional == "absolute":
            X = X + self.pos_encoder(positions)
        elif self.positional == "relative":
            add_to_keys = self.rel_pos_encoder(positions, positions)

        out = X
        for attn_layer in self.attn_layers:
            # keep keys IS queries IS values when there is no positional
            # offset: skips an add kernel and lets the attender fuse its
            # three projections into one GEMM
            keys = out if isinstance(add_to_keys, int) and add_to_keys == 0 \
                else out + add_to_keys
            out = attn_layer(keys, out, out)

        if self.is_resize:
            out = self.resize(out)
        return out
