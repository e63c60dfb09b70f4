"""Positional encoders and the meta-modules that define encoder/decoder ABIs.

Parity with /root/reference/npf/architectures/encoders.py (cited per class).
`MergeFlatInputs.resizer` / `.flat_module` and `DiscardIthArg.destination`
attribute names are part of the checkpoint key space.
"""

import torch
import torch.nn as nn

from npf.utils.initialization import weights_init

from .mlp import MLP

__all__ = [
    "SinusoidalEncodings",
    "RelativeSinusoidalEncodings",
    "merge_flat_input",
    "discard_ith_arg",
]


class SinusoidalEncodings(nn.Module):
    """Sinusoidal positional encodings of [-1,1]^d inputs.

    Reference: encoders.py:17-75.  Each of the `x_dim` coordinates gets an
    `out_dim // x_dim`-dim sin/cos encoding; the frequency constant keeps the
    "Attention is All You Need" C/dim ratio.
    """

    def __init__(self, x_dim, out_dim):
        super().__init__()
        self.x_dim = x_dim
        self.sub_dim = out_dim // x_dim
        self._C = 10000 * (self.sub_dim / 512) ** 2

        if out_dim % x_dim != 0:
            raise ValueError(f"out_dim={out_dim} has to be dividable by x_dim={x_dim}.")
        if self.sub_dim % 2 != 0:
            raise ValueError(
                f"sum_dim=out_dim/x_dim={self.sub_dim} has to be dividable by 2."
            )

        two_i_d = torch.arange(0, self.sub_dim, 2, dtype=torch.float) / self.sub_dim
        denom = torch.repeat_interleave(torch.pow(self._C, two_i_d), 2).unsqueeze(0)
        self.denom = denom.expand(1, self.x_dim, self.sub_dim)

    def forward(self, x):
        shape = x.shape
        x = x.reshape(-1, shape[-1])
        self.denom = self.denom.to(x.device)
        # map [-1,1] to an NLP-like position range [1,51] (reference :68)
        x = (x.unsqueeze(-1) + 1) * 25 + 1
        out = x / self.denom
        out = torch.stack(
            [torch.sin(out[..., 0::2]), torch.cos(out[..., 1::2])], dim=-1
        ).flatten(-2)
        return out.reshape(*shape[:-1], self.sub_dim * self.x_dim)


class RelativeSinusoidalEncodings(nn.Module):
    """Windowed relative positional encodings (reference encoders.py:78-101)."""

    def __init__(self, x_dim, out_dim, window_size=2):
        super().__init__()
        self.pos_encoder = SinusoidalEncodings(x_dim, out_dim)
        self.weight = nn.Linear(out_dim, out_dim, bias=False)
        self.window_size = window_size
        self.out_dim = out_dim

    def forward(self, keys_pos, queries_pos):
        # [batch, n_queries, n_keys, x_dim]
        diff = (keys_pos.unsqueeze(1) - queries_pos.unsqueeze(2)).abs()
        # rescale |diff| in [0, window] to [-1, 1] for the sinusoidal encoder
        out = self.weight(self.pos_encoder(diff * 2 / self.window_size - 1))
        # zero-out beyond-window pairs (enables extrapolation)
        return out * (diff < self.window_size).float()


class DiscardIthArg(nn.Module):
    """Drop the i-th positional argument before delegating to `To`.

    Reference: encoders.py:105-120.  Attribute name `destination` is part of
    the checkpoint format (e.g. ConvCNP's decoder: `decoder.destination.*`).
    """

    def __init__(self, *args, i=0, To=nn.Identity, **kwargs):
        super().__init__()
        self.i = i
        self.destination = To(*self.filter_args(*args), **kwargs)

    def filter_args(self, *args):
        return [a for j, a in enumerate(args) if j != self.i]

    def forward(self, *args, **kwargs):
        return self.destination(*self.filter_args(*args), **kwargs)


def discard_ith_arg(module, i, **kwargs):
    """Factory returning a constructor that drops positional arg `i`
    (reference encoders.py:123-127)."""

    def make(*args, **kwargs2):
        return DiscardIthArg(*args, i=i, To=module, **kwargs, **kwargs2)

    return make


class MergeFlatInputs(nn.Module):
    """Adapt a single-input module to two flat inputs by sum- or cat-merging.

    Reference: encoders.py:130-183.  With `is_sum_merge`, the second input is
    resized by an MLP and sum-merged through a ReLU (encoders.py:175-183);
    this defines both the XY-encoder and decoder signatures of most models.
    """

    def __init__(self, FlatModule, x1_dim, x2_dim, n_out, is_sum_merge=False, **kwargs):
        super().__init__()
        self.is_sum_merge = is_sum_merge
        if self.is_sum_merge:
            dim = x1_dim
            self.resizer = MLP(x2_dim, dim)
        else:
            dim = x1_dim + x2_dim
        self.flat_module = FlatModule(dim, n_out, **kwargs)
        self.reset_parameters()

    def reset_parameters(self):
        weights_init(self)

    def forward(self, x1, x2):
        if self.is_sum_merge:
            # ReLU(x1 + resize(x2)): avoids two consecutive linear maps
            merged = torch.relu(x1 + self.resizer(x2))
        else:
            merged = torch.cat((x1, x2), dim=-1)
        return self.flat_module(merged)


def merge_flat_input(module, is_sum_merge=False, **kwargs):
    """Factory: `merge_flat_input(M)(x_dim, flat_dim, n_out, **kw)`
    (reference encoders.py:186-213)."""

    def make(x_shape, flat_dim, n_out, **kwargs2):
        assert isinstance(x_shape, int)
        return MergeFlatInputs(
            module, x_shape, flat_dim, n_out, is_sum_merge=is_sum_merge,
            **kwargs2, **kwargs,
        )

    return make
