"""Experiment-recipe layer tests (reference utils/ntbks_helpers.py parity):
dataset registries, renamer, y-dim injection, n_cntxt splitters and the
multi-model comparison plots."""

import matplotlib

matplotlib.use("Agg")

import matplotlib.pyplot as plt
import pytest
import torch

from npf import recipes
from npf.data import GPDataset

from model_zoo import attncnp_2d, cnp_1d, gridconvcnp_2d


def teardown_function(_):
    plt.close("all")


def test_get_datasets_single_gp_small():
    d, t, v = recipes.get_datasets_single_gp(
        n_samples=64, n_points=16, save_file=None
    )
    assert set(d) == {"RBF_Kernel", "Periodic_Kernel", "Noisy_Matern_Kernel"}
    for ds in d.values():
        assert not ds.is_reuse_across_epochs
    # test split is a fixed 10k draw in the reference; with save_file=None it
    # still freezes on one draw
    for ds in t.values():
        assert ds.is_reuse_across_epochs
    for ds in v.values():
        assert len(ds) == 64 // 10


def test_get_all_gp_datasets_names():
    d, t, v = recipes.get_all_gp_datasets(
        n_samples=32, n_points=8, save_file=None
    )
    assert set(d) == {
        "RBF_Kernel", "Periodic_Kernel", "Noisy_Matern_Kernel",
        "Variable_Matern_Kernel", "All_Kernels",
    }
    assert len(d["All_Kernels"]) == 3 * 32


def test_pretty_renamer():
    r = recipes.PRETTY_RENAMER
    assert r["celeba32"] == "CelebA32"
    assert r["AttnCNP"] == "AttnCNP"
    assert r["ConvLNP_ELBOTrue_LatLBTrue_SigLBTrue"].startswith("ConvLNP NPVI")
    assert r[3] == 3


def test_add_y_dim():
    models = {"m": cnp_1d}

    class FakeDs:
        shape = (3, 32, 32)

    out = recipes.add_y_dim(models, {"celeba32": FakeDs()})
    assert out["celeba32"]["m"].keywords == {"y_dim": 3}


def test_get_n_cntxt_1d():
    getter = recipes.get_n_cntxt(7)
    X = torch.rand(2, 32, 1) * 2 - 1
    Y = torch.randn(2, 32, 1)
    Xc, Yc, Xt, Yt = getter(X, Y)
    assert Xc.shape == (2, 7, 1)
    assert Xt.shape == (2, 32, 1)


def test_get_n_cntxt_2d():
    getter = recipes.get_n_cntxt(9, is_1d=False)
    X = torch.rand(2, 3, 8, 8)
    Xc, Yc, Xt, Yt = getter(X)
    assert Yc.shape == (2, 9, 3)


@pytest.fixture(scope="module")
def tiny_gp():
    from npf.data.kernels import RBF

    return GPDataset(
        kernel=RBF(0.2), n_samples=8, n_points=32, is_reuse_across_epochs=True
    )


def test_plot_multi_posterior_samples_1d(tiny_gp):
    fig = recipes.plot_multi_posterior_samples_1d(
        {"RBF_Kernel/CNP": cnp_1d()}, {"RBF_Kernel": tiny_gp}, n_cntxt=5,
        is_plot_generator=True,
    )
    assert len(fig.axes) == 1


def test_plot_multi_prior_samples_1d(tiny_gp):
    fig = recipes.plot_multi_prior_samples_1d(
        {"RBF_Kernel/CNP": cnp_1d()}, {"RBF_Kernel": tiny_gp}
    )
    assert len(fig.axes) == 1


def test_plot_multi_posterior_samples_imgs():
    from npf.data.imgs import SyntheticImages

    ds = SyntheticImages(shape=(3, 16, 16), n_samples=8)
    fig = recipes.plot_multi_posterior_samples_imgs(
        {"synthetic32/GridConvCNP": gridconvcnp_2d(y_dim=3)},
        {"synthetic32": ds},
        n_cntxt=0.2,
        n_plots=2,
    )
    assert len(fig.axes) == 1


def test_gp_dataset_generator_oracle(tiny_gp):
    gen = tiny_gp.generator
    import sklearn.gaussian_process

    assert isinstance(gen, sklearn.gaussian_process.GaussianProcessRegressor)
    # oracle can fit+predict on a context set
    gen.fit([[0.0], [0.5]], [[0.1], [0.2]])
    mean, std = gen.predict([[0.25]], return_std=True)
    assert mean.shape[0] == 1


def test_std_processing_kwargs_ablation_grid():
    """The LatLB/SigLB ablation knobs build working LNP variants."""
    from model_zoo import R_DIM
    from functools import partial
    from npf import LNP
    from npf.architectures import MLP, merge_flat_input

    for min_lat in (None, 0.1):
        for min_sig in (0.01, 1e-4):
            kw = recipes.get_std_processing_kwargs(
                min_sigma_pred=min_sig, min_lat=min_lat
            )
            m = LNP(
                x_dim=1, y_dim=1, is_q_zCct=False,
                n_z_samples_train=2, n_z_samples_test=2,
                XYEncoder=merge_flat_input(
                    partial(MLP, n_hidden_layers=2, hidden_size=2 * R_DIM),
                    is_sum_merge=True,
                ),
                XEncoder=partial(MLP, n_hidden_layers=1, hidden_size=R_DIM),
                Decoder=merge_flat_input(
                    partial(MLP, n_hidden_layers=4, hidden_size=R_DIM),
                    is_sum_merge=True,
                ),
                r_dim=R_DIM,
                **kw,
            )
            X = torch.rand(2, 9, 1) * 2 - 1
            p, *_ = m(X[:, :4], torch.randn(2, 4, 1), X)
            assert float(p.base_dist.scale.detach().min()) >= min_sig


def test_cli_ablation_knobs_apply(tmp_path):
    """--min-sigma-pred / --min-lat (the Losses.ipynb grid knobs) change the
    scale transformers and the model still trains a step."""
    import numpy as np

    from npf import cli

    cli.main([
        "train", "--model", "LNP", "--data", "RBF_Kernel",
        "--epochs", "1", "--n-tasks", "64", "--n-test-tasks", "16",
        "--batch-size", "8", "--chckpnt-dir", str(tmp_path) + "/",
        "--data-cache", str(tmp_path / "cache.npz"),
        "--min-sigma-pred", "0.1", "--min-lat", "0.1",
        "--loss", "elbo",
    ])
    run_dir = tmp_path / "RBF_Kernel" / "LNP" / "run_0"
    assert (run_dir / "params.pt").exists()
    ll = np.loadtxt(run_dir / "eval.csv", delimiter=",")
    assert ll.shape == (16,) and np.isfinite(ll).all()
