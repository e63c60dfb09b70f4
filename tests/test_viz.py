"""Visualization-suite tests (capability parity with reference
utils/visualize/*): every public plot function renders on CPU with tiny
models/datasets, gifs are written, and the numeric helpers (points_to_grid,
marginal_log_like, sarle) are checked against direct computations."""

import os

import matplotlib

matplotlib.use("Agg")

import matplotlib.pyplot as plt
import numpy as np
import pytest
import torch

from npf.data import GPDataset
from npf.data.imgs import SyntheticImages
from npf.data.kernels import RBF
from npf.utils.datasplit import (
    CntxtTrgtGetter,
    GetRandomIndcs,
    GridCntxtTrgtGetter,
    RandomMasker,
    no_masker,
)
from npf.utils.helpers import MultivariateNormalDiag
from npf.viz import (
    fig2img,
    gen_p_y_pred,
    giffify,
    kdeplot,
    make_grid,
    marginal_log_like,
    plot_config,
    plot_dataset_samples_1d,
    plot_dataset_samples_imgs,
    plot_losses,
    plot_posterior_samples,
    plot_posterior_samples_1d,
    plot_prior_samples_1d,
    points_to_grid,
    sarle,
)

from model_zoo import cnp_1d, gridconvcnp_2d, lnp_1d


@pytest.fixture(scope="module")
def gp_ds():
    return GPDataset(kernel=RBF(0.2), n_samples=32, n_points=64)


@pytest.fixture(scope="module")
def img_ds():
    return SyntheticImages(shape=(3, 16, 16), n_samples=16)


def teardown_function(_):
    plt.close("all")


def test_plot_losses():
    hist = [
        {"train_loss": 10.0 - i, "valid_loss": 11.0 - i} for i in range(5)
    ]
    ax = plot_losses(hist)
    assert len(ax.lines) == 2
    assert ax.get_xlabel() == "Number of Epochs"


def test_plot_dataset_samples_1d(gp_ds):
    ax = plot_dataset_samples_1d(gp_ds, n_samples=3)
    assert len(ax.lines) == 3
    lo, hi = ax.get_xlim()
    assert lo <= gp_ds.min_max[0] and hi >= gp_ds.min_max[1]


def test_plot_posterior_samples_1d(gp_ds):
    model = cnp_1d()
    X, Y = gp_ds[0]
    getter = CntxtTrgtGetter(contexts_getter=GetRandomIndcs(a=5, b=10))
    ax = plot_posterior_samples_1d(
        X[None], Y[None], getter, model, seed=0, is_plot_std=True
    )
    labels = [ln.get_label() for ln in ax.lines]
    assert "Model" in labels
    assert "Target Function" in labels


def test_plot_posterior_extrapolation_boundary(gp_ds):
    model = cnp_1d()
    # targets beyond [-1,1] => extrapolation boundary line at train bound
    X = torch.linspace(-1.5, 1.5, 64).view(1, -1, 1)
    Y = torch.randn(1, 64, 1)
    getter = CntxtTrgtGetter(contexts_getter=GetRandomIndcs(a=5, b=10))
    ax = plot_posterior_samples_1d(X, Y, getter, model, seed=0)
    labels = [ln.get_label() for ln in ax.lines]
    assert "Extrapolation Boundary" in labels


def test_plot_prior_samples_1d_lnp():
    model = lnp_1d()
    ax = plot_prior_samples_1d(model, n_trgt=32, n_samples=4)
    assert len(ax.lines) >= 4


def test_gen_p_y_pred_latent_counts():
    model = lnp_1d()
    X = torch.rand(1, 16, 1) * 2 - 1
    Y = torch.randn(1, 16, 1)
    curves = list(gen_p_y_pred(model.eval(), X[:, :5], Y[:, :5], X, n_samples=7))
    assert len(curves) == 7
    assert model.n_z_samples_test != 7  # restored


def test_fig2img_and_giffify(tmp_path):
    def gen_fig(scale=1.0):
        fig, ax = plt.subplots(figsize=(2, 2))
        ax.plot([0, 1], [0, scale])
        return fig

    img = fig2img(gen_fig())
    assert img.ndim == 3 and img.shape[-1] == 4

    gif = os.path.join(tmp_path, "sweep.gif")
    giffify(gif, gen_fig, "scale", [0.5, 1.0, 2.0], fps=4)
    assert os.path.getsize(gif) > 0
    from PIL import Image

    with Image.open(gif) as im:
        assert im.n_frames == 3


def test_plot_config_restores_rc():
    before = dict(plt.rcParams)
    with plot_config(style="darkgrid", font_scale=2):
        assert plt.rcParams["axes.grid"] is True
    assert plt.rcParams["font.size"] == before["font.size"]


def test_make_grid_shape():
    t = torch.rand(5, 3, 8, 8)
    grid = make_grid(t, nrow=2, padding=2, pad_value=1.0)
    assert grid.shape == (3, 2 + 3 * 10, 2 + 2 * 10)
    assert float(grid[0, 0, 0]) == 1.0


def test_kdeplot():
    ax = kdeplot(np.random.RandomState(0).randn(200), label="x")
    assert ax.get_legend_handles_labels()[1] == ["x"]


def test_points_to_grid_roundtrip():
    # place 2 known pixels on a 4x4 grid
    X = torch.tensor([[[-1.0, -1.0], [1.0, 1.0]]])  # corners
    Y = torch.tensor([[[1.0, 2.0, 3.0], [4.0, 5.0, 6.0]]])
    grid, mask = points_to_grid(X, Y, (4, 4))
    assert torch.allclose(grid[0, 0, 0], torch.tensor([1.0, 2.0, 3.0]))
    assert torch.allclose(grid[0, 3, 3], torch.tensor([4.0, 5.0, 6.0]))
    assert mask.sum() == 2


def test_marginal_log_like_and_sarle():
    loc = torch.zeros(4, 1, 10, 1)
    scale = torch.ones(4, 1, 10, 1)
    pred = MultivariateNormalDiag(loc, scale)
    samples = torch.zeros(1, 100, 1, 1)
    ml = marginal_log_like(pred, samples)
    # z-marginal of N(0,1) at 0 with 4 identical comps = pdf(0) = 1/sqrt(2pi)
    assert torch.allclose(ml, torch.full_like(ml, 1 / float(np.sqrt(2 * np.pi))), atol=1e-4)

    unimodal = np.random.RandomState(0).randn(500, 8)
    s = sarle(unimodal)
    assert s.shape == (8,)
    assert (s < 0.6).all()  # unimodal gaussian ~ 0.33


def test_plot_dataset_samples_imgs(img_ds):
    fig, ax = plt.subplots()
    plot_dataset_samples_imgs(img_ds, n_plots=4, ax=ax)
    assert len(ax.images) == 1


def test_plot_posterior_samples_grid_model(img_ds):
    model = gridconvcnp_2d(y_dim=3)
    getter = GridCntxtTrgtGetter(
        context_masker=RandomMasker(a=0.1, b=0.3), target_masker=no_masker
    )
    grid = plot_posterior_samples(
        img_ds, getter, model,
        is_uniform_grid=True, n_plots=2, is_return=True,
    )
    assert grid.dim() == 3 and grid.size(0) == 3


def test_plot_img_marginal_pred(img_ds):
    from model_zoo import attnlnp_2d
    from npf.utils.datasplit import GridCntxtTrgtGetter, RandomMasker
    from npf.viz import plot_img_marginal_pred

    model = attnlnp_2d(y_dim=3)
    model.n_z_samples_test = 3
    getter = GridCntxtTrgtGetter(context_masker=RandomMasker(a=0.05, b=0.2))
    fig = plot_img_marginal_pred(
        model, img_ds, getter, n_samples=2, is_uniform_grid=False,
        n_plots_loop=1, n_marginals=3,
    )
    assert len(fig.axes) == 2


def test_plot_qualitative_with_kde(tmp_path, img_ds):
    import numpy as np

    from model_zoo import gridconvcnp_2d
    from npf import CNPFLoss
    from npf.train import NPFTrainer
    from npf.viz import plot_qualitative_with_kde

    trainer = NPFTrainer(
        gridconvcnp_2d(y_dim=3), CNPFLoss(), device="cpu",
        chckpnt_dirname=str(tmp_path),
    )
    rng = np.random.RandomState(0)
    np.savetxt(
        os.path.join(str(tmp_path), "eval.csv"),
        rng.randn(len(img_ds)) * 50 + 300,
    )
    fig = plot_qualitative_with_kde(
        ["GridConvCNP", trainer], img_ds, n_images=2, n_samples=1,
    )
    assert len(fig.axes) == 2
