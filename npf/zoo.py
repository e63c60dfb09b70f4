"""Canonical model zoo — the notebook configs of the reference
(/root/reference/jupyter/reproducibility/*.ipynb cells; parameter counts in
BASELINE.md), shared by the CLI, bench.py, examples and tests."""

from functools import partial

import torch

from npf import (
    CNP,
    LNP,
    AttnCNP,
    AttnLNP,
    ConvCNP,
    ConvLNP,
    GridConvCNP,
    GridConvLNP,
)
from npf.utils.helpers import CircularPad2d, make_abs_conv, make_padded_conv
from npf.architectures import (
    CNN,
    MLP,
    ResConvBlock,
    SetConv,
    discard_ith_arg,
    merge_flat_input,
)

R_DIM = 128

CNN_KWARGS = dict(ConvBlock=ResConvBlock, is_chan_last=True, n_conv_layers=2)


def mlp_kwargs(r_dim=R_DIM):
    return dict(
        r_dim=r_dim,
        XEncoder=partial(MLP, n_hidden_layers=1, hidden_size=r_dim),
        Decoder=merge_flat_input(
            partial(MLP, n_hidden_layers=4, hidden_size=r_dim), is_sum_merge=True
        ),
    )


def cnp_1d(r_dim=R_DIM):
    return CNP(
        x_dim=1, y_dim=1,
        XYEncoder=merge_flat_input(
            partial(MLP, n_hidden_layers=2, hidden_size=r_dim * 2), is_sum_merge=True
        ),
        **mlp_kwargs(r_dim),
    )


def cnp_2d(y_dim=3, r_dim=R_DIM):
    return CNP(
        x_dim=2, y_dim=y_dim,
        XYEncoder=merge_flat_input(
            partial(MLP, n_hidden_layers=2, hidden_size=r_dim * 3), is_sum_merge=True
        ),
        **mlp_kwargs(r_dim),
    )


def lnp_1d(r_dim=R_DIM):
    return LNP(
        x_dim=1, y_dim=1, is_q_zCct=True, n_z_samples_train=1, n_z_samples_test=32,
        XYEncoder=merge_flat_input(
            partial(MLP, n_hidden_layers=2, hidden_size=r_dim * 2), is_sum_merge=True
        ),
        **mlp_kwargs(r_dim),
    )


def lnp_2d(y_dim=3, r_dim=R_DIM):
    return LNP(
        x_dim=2, y_dim=y_dim, is_q_zCct=True, n_z_samples_train=1, n_z_samples_test=32,
        XYEncoder=merge_flat_input(
            partial(MLP, n_hidden_layers=2, hidden_size=r_dim * 3), is_sum_merge=True
        ),
        **mlp_kwargs(r_dim),
    )


def attncnp_1d(r_dim=R_DIM):
    return AttnCNP(
        x_dim=1, y_dim=1,
        XYEncoder=merge_flat_input(
            partial(MLP, n_hidden_layers=2, hidden_size=r_dim), is_sum_merge=True
        ),
        is_self_attn=False, attention="transformer",
        **mlp_kwargs(r_dim),
    )


def attncnp_2d(y_dim=3, r_dim=R_DIM):
    return AttnCNP(
        x_dim=2, y_dim=y_dim, is_self_attn=True, attention="transformer",
        **mlp_kwargs(r_dim),
    )


def attnlnp_1d(r_dim=R_DIM):
    return AttnLNP(
        x_dim=1, y_dim=1,
        XYEncoder=merge_flat_input(
            partial(MLP, n_hidden_layers=2, hidden_size=r_dim), is_sum_merge=True
        ),
        is_self_attn=False, is_q_zCct=True, n_z_samples_train=1, n_z_samples_test=8,
        r_dim=r_dim, attention="transformer",
    )


def attnlnp_2d(y_dim=3, r_dim=R_DIM):
    return AttnLNP(
        x_dim=2, y_dim=y_dim, is_self_attn=True, is_q_zCct=True,
        n_z_samples_train=1, n_z_samples_test=8, r_dim=r_dim, attention="transformer",
    )


def convcnp_1d(r_dim=R_DIM):
    return ConvCNP(
        x_dim=1, y_dim=1, Interpolator=SetConv,
        CNN=partial(
            CNN, Conv=torch.nn.Conv1d, Normalization=torch.nn.BatchNorm1d,
            n_blocks=5, kernel_size=19, **CNN_KWARGS,
        ),
        density_induced=64, r_dim=r_dim,
        Decoder=discard_ith_arg(partial(MLP, n_hidden_layers=4, hidden_size=r_dim), i=0),
    )


def gridconvcnp_2d(y_dim=3, r_dim=R_DIM, n_blocks=5):
    return GridConvCNP(
        x_dim=1, y_dim=y_dim,
        CNN=partial(
            CNN, Conv=torch.nn.Conv2d, Normalization=torch.nn.BatchNorm2d,
            n_blocks=n_blocks, kernel_size=9, **CNN_KWARGS,
        ),
        r_dim=r_dim,
        Decoder=discard_ith_arg(partial(MLP, n_hidden_layers=4, hidden_size=r_dim), i=0),
    )


def convlnp_1d(r_dim=R_DIM):
    return ConvLNP(
        x_dim=1, y_dim=1, Interpolator=SetConv,
        CNN=partial(
            CNN, Conv=torch.nn.Conv1d, Normalization=torch.nn.BatchNorm1d,
            kernel_size=19, n_blocks=4, **CNN_KWARGS,
        ),
        density_induced=64, is_global=True, is_q_zCct=False,
        n_z_samples_train=16, n_z_samples_test=32, r_dim=r_dim,
        Decoder=discard_ith_arg(torch.nn.Linear, i=0),
    )


def gridconvcnp_zsmms(y_dim=1, r_dim=R_DIM):
    """Fully translation-equivariant zsmms variant: circular padding in every
    conv incl. the density encoder (reference ConvCNP.ipynb model_2d_extrap;
    pretrained at results/pretrained/zsmms/ConvCNP)."""
    return GridConvCNP(
        x_dim=1, y_dim=y_dim,
        CNN=partial(
            CNN, Conv=make_padded_conv(torch.nn.Conv2d, CircularPad2d),
            Normalization=partial(torch.nn.BatchNorm2d, eps=1e-2),
            n_blocks=5, kernel_size=9, **CNN_KWARGS,
        ),
        Conv=lambda y_dim: make_padded_conv(
            make_abs_conv(torch.nn.Conv2d), CircularPad2d
        )(y_dim, y_dim, groups=y_dim, kernel_size=11, padding=11 // 2, bias=False),
        r_dim=r_dim,
        Decoder=discard_ith_arg(partial(MLP, n_hidden_layers=4, hidden_size=r_dim), i=0),
    )


def gridconvcnp_xl(y_dim=3, r_dim=R_DIM):
    """12-block celeba128 XL config (reference ConvCNP.ipynb model_2d_XL)."""
    return gridconvcnp_2d(y_dim=y_dim, r_dim=r_dim, n_blocks=12)


def gridconvlnp_2d(y_dim=3, r_dim=R_DIM):
    return GridConvLNP(
        x_dim=1, y_dim=y_dim,
        CNN=partial(
            CNN, Conv=torch.nn.Conv2d, Normalization=torch.nn.BatchNorm2d,
            kernel_size=9, n_blocks=4, **CNN_KWARGS,
        ),
        is_global=True, is_q_zCct=False, n_z_samples_train=16, n_z_samples_test=32,
        r_dim=r_dim, Decoder=discard_ith_arg(torch.nn.Linear, i=0),
    )


PUBLISHED_PARAM_COUNTS = {
    "cnp_1d": 252098,
    "cnp_2d": 367750,
    "lnp_1d": 301634,
    "lnp_2d": 417286,
    "attncnp_1d": 252738,
    "attncnp_2d": 386054,
    "attnlnp_1d": 335170,
    "attnlnp_2d": 468486,
    "convcnp_1d": 276612,
    "gridconvcnp_2d": 340721,
    "gridconvcnp_xl": 722417,
    "convlnp_1d": 376068,
    "gridconvlnp_2d": 487793,
}

BUILDERS = {
    "cnp_1d": cnp_1d,
    "cnp_2d": cnp_2d,
    "lnp_1d": lnp_1d,
    "lnp_2d": lnp_2d,
    "attncnp_1d": attncnp_1d,
    "attncnp_2d": attncnp_2d,
    "attnlnp_1d": attnlnp_1d,
    "attnlnp_2d": attnlnp_2d,
    "convcnp_1d": convcnp_1d,
    "gridconvcnp_2d": gridconvcnp_2d,
    "gridconvcnp_zsmms": gridconvcnp_zsmms,
    "gridconvcnp_xl": gridconvcnp_xl,
    "convlnp_1d": convlnp_1d,
    "gridconvlnp_2d": gridconvlnp_2d,
}
