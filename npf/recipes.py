"""Canonical experiment recipes — the reference's `utils/ntbks_helpers.py`
layer (get_all_gp_datasets:61, get_img_datasets:52, add_y_dim:261,
get_n_cntxt:272, PRETTY_RENAMER:217, plot_multi_posterior_samples_1d:366,
plot_multi_posterior_samples_imgs:290, plot_multi_prior_samples_1d:444)
rebuilt on this framework's torch-native GP sampler and viz suite.

These are the dataset/model configurations every published baseline number
was produced with (BASELINE.md 'Training configuration')."""

import copy
from functools import partial

import matplotlib.pyplot as plt

from npf.data import GPDataset
from npf.data.imgs import get_train_test_img_dataset
from npf.data.helpers import DatasetMerger
from npf.data.imgs import SingleImage, get_test_upscale_factor
from npf.data.kernels import RBF, ExpSineSquared, Matern, WhiteKernel
from npf.neuralproc import GridConvCNP
from npf.utils.datasplit import (
    CntxtTrgtGetter,
    GetRandomIndcs,
    GridCntxtTrgtGetter,
    RandomMasker,
    SuperresolutionCntxtTrgtGetter,
    get_all_indcs,
    half_masker,
    no_masker,
)
from npf.viz import (
    plot_config,
    plot_posterior_samples,
    plot_posterior_samples_1d,
    plot_prior_samples_1d,
)

__all__ = [
    "get_img_datasets",
    "get_all_gp_datasets",
    "get_datasets_single_gp",
    "get_datasets_variable_hyp_gp",
    "get_datasets_variable_kernel_gp",
    "get_gp_datasets",
    "sample_gp_dataset_like",
    "StrFormatter",
    "PRETTY_RENAMER",
    "add_y_dim",
    "get_std_processing_kwargs",
    "get_n_cntxt",
    "plot_multi_posterior_samples_1d",
    "plot_multi_posterior_samples_imgs",
    "plot_multi_prior_samples_1d",
    "gif_explain",
    "select_labels",
]


# --------------------------------------------------------------------------- #
# datasets
# --------------------------------------------------------------------------- #


def get_img_datasets(datasets):
    """Instantiate (train, test) image datasets by name."""
    train_datasets, test_datasets = dict(), dict()
    for d in datasets:
        train_datasets[d], test_datasets[d] = get_train_test_img_dataset(d)
    return train_datasets, test_datasets


def get_datasets_single_gp(n_test=10000, **kwargs):
    """The three fixed-hyperparameter GP benchmarks (reference
    ntbks_helpers.py:78-99: RBF ls=0.2; periodic ls=0.5 p=0.5; noisy Matern
    White(0.1)+Matern ls=0.2 nu=1.5; 50k tasks x 128 points, fresh every
    epoch)."""
    kernels = {
        "RBF_Kernel": RBF(length_scale=0.2),
        "Periodic_Kernel": ExpSineSquared(length_scale=0.5, periodicity=0.5),
        "Noisy_Matern_Kernel": WhiteKernel(noise_level=0.1)
        + Matern(length_scale=0.2, nu=1.5),
    }
    defaults = dict(
        is_vary_kernel_hyp=False,
        n_samples=50000,
        n_points=128,
        is_reuse_across_epochs=False,
    )
    defaults.update(kwargs)
    return get_gp_datasets(kernels, n_test=n_test, **defaults)


def get_datasets_variable_hyp_gp(**kwargs):
    """Matern with per-task length-scale in (0.01, 0.3)."""
    kernels = {
        "Variable_Matern_Kernel": Matern(
            length_scale=0.1, length_scale_bounds=(0.01, 0.3), nu=1.5
        )
    }
    defaults = dict(
        is_vary_kernel_hyp=True,
        n_samples=50000,
        n_points=128,
        is_reuse_across_epochs=False,
    )
    defaults.update(kwargs)
    return get_gp_datasets(kernels, **defaults)


def get_datasets_variable_kernel_gp(n_test=10000, **kwargs):
    """All single-GP datasets merged into one task distribution."""
    datasets, test_datasets, valid_datasets = get_datasets_single_gp(
        n_test=n_test, **kwargs
    )
    return (
        dict(All_Kernels=DatasetMerger(datasets.values())),
        dict(All_Kernels=DatasetMerger(test_datasets.values())),
        dict(All_Kernels=DatasetMerger(valid_datasets.values())),
    )


def get_all_gp_datasets(**kwargs):
    """Train/test/valid dicts for every GP experiment."""
    datasets, test_datasets, valid_datasets = dict(), dict(), dict()
    for f in (
        get_datasets_single_gp,
        get_datasets_variable_hyp_gp,
        get_datasets_variable_kernel_gp,
    ):
        d, t, v = f(**kwargs)
        datasets.update(d)
        test_datasets.update(t)
        valid_datasets.update(v)
    return datasets, test_datasets, valid_datasets


def sample_gp_dataset_like(dataset, **kwargs):
    """Deep-copied dataset frozen on one `get_samples` draw."""
    new_dataset = copy.deepcopy(dataset)
    new_dataset.set_samples_(*dataset.get_samples(**kwargs))
    return new_dataset


def get_gp_datasets(kernels, save_file="data/gp_dataset.npz", n_test=10000,
                    **kwargs):
    """(train, test, valid) GPDataset dicts for each named kernel; test is a
    fixed draw of `n_test` tasks (cache chunk -1, 10k in the reference),
    valid n/10 (chunk -2) — reference ntbks_helpers.py:136-172."""

    def chunk_file(name):
        return (save_file, name) if save_file is not None else None

    datasets = {
        name: GPDataset(kernel=kernel, save_file=chunk_file(name), **kwargs)
        for name, kernel in kernels.items()
    }
    datasets_test = {
        k: sample_gp_dataset_like(
            ds, save_file=chunk_file(k), idx_chunk=-1, n_samples=n_test
        )
        for k, ds in datasets.items()
    }
    datasets_valid = {
        k: sample_gp_dataset_like(
            ds, save_file=chunk_file(k), idx_chunk=-2,
            n_samples=ds.n_samples // 10,
        )
        for k, ds in datasets.items()
    }
    return datasets, datasets_test, datasets_valid


# --------------------------------------------------------------------------- #
# naming / config helpers
# --------------------------------------------------------------------------- #


class StrFormatter:
    """Pretty-renamer: exact matches, then title-cased substring replaces,
    then upper-casing of listed words."""

    def __init__(self, exact_match={}, subtring_replace={}, to_upper=[]):
        self.exact_match = exact_match
        self.subtring_replace = subtring_replace
        self.to_upper = to_upper

    def __getitem__(self, key):
        if not isinstance(key, str):
            return key
        if key in self.exact_match:
            return self.exact_match[key]
        key = key.title()
        for match, replace in self.subtring_replace.items():
            key = key.replace(match, replace if replace is not None else "")
        for w in self.to_upper:
            key = key.replace(w, w.upper())
        return key


PRETTY_RENAMER = StrFormatter(
    exact_match={
        "celeba64": "CelebA64",
        "celeba32": "CelebA32",
        "zs-multi-mnist": "ZSMM",
        "zsmms": "ZSMM",
    },
    subtring_replace={
        "_": " ",
        "Elbofalse": "NPML",
        "Elbotrue": "NPVI",
        "Latlbtrue": "LB_Z",
        "Latlbfalse": "",
        "Siglbtrue": "LB_P",
        "Siglbfalse": "",
        "Vhalf": "Vert. Half",
        "hhalf": "Horiz. Half",
        "Attncnp": "AttnCNP",
        "Convcnp": "ConvCNP",
        "Attnlnp": "AttnLNP",
        "Convlnp": "ConvLNP",
        "Selfattn": "Attn",
    },
    to_upper=["Mnist", "Svhn", "Cnp", "Lnp", "Rbf"],
)


def get_std_processing_kwargs(min_sigma_pred=0.01, min_lat=None):
    """Predictive/latent std lower-bound kwargs — the knobs of the 24-model
    loss-ablation grid (reference Losses.ipynb `get_std_processing_kwargs`;
    results tabulated in BASELINE.md)."""
    import torch.nn.functional as F

    kwargs = dict(
        p_y_scale_transformer=lambda y_scale: min_sigma_pred
        + (1 - min_sigma_pred) * F.softplus(y_scale)
    )
    if min_lat is not None:
        kwargs["q_z_scale_transformer"] = lambda z_scale: min_lat + (
            1 - min_lat
        ) * F.softplus(z_scale)
    return kwargs


def add_y_dim(models, datasets):
    """{data: {model: partial(model, y_dim=channels)}} — inject the
    dataset-dependent output dimension into model factories."""
    return {
        data_name: {
            model_name: partial(model, y_dim=data_train.shape[0])
            for model_name, model in models.items()
        }
        for data_name, data_train in datasets.items()
    }


def get_n_cntxt(n_cntxt, is_1d=True, upscale_factor=1):
    """Context/target splitter with a fixed number of context points."""
    if is_1d:
        return CntxtTrgtGetter(
            contexts_getter=GetRandomIndcs(a=n_cntxt, b=n_cntxt),
            targets_getter=get_all_indcs,
            is_add_cntxts_to_trgts=False,
        )
    return GridCntxtTrgtGetter(
        context_masker=RandomMasker(a=n_cntxt, b=n_cntxt),
        target_masker=no_masker,
        is_add_cntxts_to_trgts=False,
        upscale_factor=upscale_factor,
    )


# --------------------------------------------------------------------------- #
# multi-model comparison plots
# --------------------------------------------------------------------------- #


def _module_of(trainer):
    """Accept either an NPFTrainer or a bare nn.Module."""
    return getattr(trainer, "module_", trainer)


def plot_multi_posterior_samples_1d(
    trainers,
    datasets,
    n_cntxt,
    trainers_compare=None,
    plot_config_kwargs={},
    title="Model : {model_name} | Data : {data_name} | Num. Context : {n_cntxt}",
    left_extrap=0,
    right_extrap=0,
    pretty_renamer=PRETTY_RENAMER,
    is_plot_generator=True,
    imgsize=(8, 3),
    **kwargs,
):
    """One posterior plot per '{data}/{model}' trainer, sharing axes."""
    with plot_config(**plot_config_kwargs):
        n_trainers = len(trainers)
        n_col = 1 if trainers_compare is None else 2
        fig, axes = plt.subplots(
            n_trainers, n_col,
            figsize=(imgsize[0] * n_col, imgsize[1] * n_trainers),
            sharex=True, sharey=True, squeeze=False,
        )
        for j, curr in enumerate([trainers, trainers_compare]):
            if curr is None:
                continue
            for i, (k, trainer) in enumerate(curr.items()):
                data_name, model_name = k.split("/")[0], k.split("/")[1]
                dataset = datasets[data_name]
                curr_title = (
                    title.format(
                        model_name=pretty_renamer[model_name],
                        n_cntxt=n_cntxt,
                        data_name=pretty_renamer[data_name],
                    )
                    if title is not None
                    else None
                )
                module = _module_of(trainer)
                test_min_max = dataset.min_max
                if left_extrap != 0 or right_extrap != 0:
                    test_min_max = (
                        dataset.min_max[0] - left_extrap,
                        dataset.min_max[1] + right_extrap,
                    )
                    module.set_extrapolation(test_min_max)
                X, Y = dataset.get_samples(
                    n_samples=1, n_points=3 * dataset.n_points,
                    test_min_max=test_min_max,
                )
                plot_posterior_samples_1d(
                    X, Y, get_n_cntxt(n_cntxt), module,
                    generator=dataset.generator if is_plot_generator else None,
                    train_min_max=dataset.min_max,
                    title=curr_title,
                    ax=axes[i, j],
                    scatter_label="Context Set",
                    **kwargs,
                )
        plt.tight_layout()
    return fig


def plot_multi_posterior_samples_imgs(
    trainers,
    datasets,
    n_cntxt,
    plot_config_kwargs={},
    title="{model_name} | {data_name} | C={n_cntxt}",
    pretty_renamer=PRETTY_RENAMER,
    n_plots=4,
    figsize=(3, 3),
    is_superresolution=False,
    **kwargs,
):
    """One image-completion plot per '{data}/{model}' trainer.

    `n_cntxt` may be an int, a fraction, "vhalf"/"hhalf" or (with
    is_superresolution) a resolution factor."""
    with plot_config(**plot_config_kwargs):
        n_trainers = len(trainers)
        fig, axes = plt.subplots(
            1, n_trainers,
            figsize=(figsize[0] * n_plots, figsize[1] * n_trainers),
            squeeze=False,
        )
        for i, (k, trainer) in enumerate(trainers.items()):
            data_name, model_name = k.split("/")[0], k.split("/")[1]
            dataset = datasets[data_name]

            if isinstance(n_cntxt, float) and n_cntxt < 1:
                if is_superresolution:
                    n_cntxt_title = (
                        f"{int(dataset.shape[1] * n_cntxt)}x"
                        f"{int(dataset.shape[2] * n_cntxt)}"
                    )
                else:
                    n_cntxt_title = f"{100 * n_cntxt:.1f}%"
            elif isinstance(n_cntxt, str):
                n_cntxt_title = pretty_renamer[n_cntxt]
            else:
                n_cntxt_title = n_cntxt

            curr_title = title.format(
                model_name=pretty_renamer[model_name],
                n_cntxt=n_cntxt_title,
                data_name=pretty_renamer[data_name],
            )

            upscale_factor = get_test_upscale_factor(data_name)
            if n_cntxt in ("vhalf", "hhalf"):
                getter = GridCntxtTrgtGetter(
                    context_masker=partial(
                        half_masker, dim=0 if n_cntxt == "hhalf" else 1
                    ),
                    upscale_factor=upscale_factor,
                )
            elif is_superresolution:
                getter = SuperresolutionCntxtTrgtGetter(
                    resolution_factor=n_cntxt, upscale_factor=upscale_factor
                )
            else:
                getter = get_n_cntxt(
                    n_cntxt, is_1d=False, upscale_factor=upscale_factor
                )

            module = _module_of(trainer).cpu()
            plot_posterior_samples(
                dataset, getter, module,
                is_uniform_grid=isinstance(module, GridConvCNP),
                ax=axes.flatten()[i],
                n_plots=n_plots if not isinstance(dataset, SingleImage) else 1,
                is_mask_cntxt=not is_superresolution,
                **kwargs,
            )
            axes.flatten()[i].set_title(curr_title)
    return fig


def plot_multi_prior_samples_1d(trainers, datasets, **kwargs):
    """Prior (no-context) function draws per trainer."""
    n_trainers = len(trainers)
    fig, axes = plt.subplots(
        n_trainers, 1, figsize=(8, 3 * n_trainers), sharex=True, squeeze=False
    )
    for i, (k, trainer) in enumerate(trainers.items()):
        data_name = k.split("/")[0]
        model_name = k.split("/")[1].replace("_", " ")
        dataset = datasets[data_name]
        plot_prior_samples_1d(
            _module_of(trainer),
            title=f"{model_name} Trained Prior : {data_name.replace('_', ' ')}",
            train_min_max=dataset.min_max,
            ax=axes.flatten()[i],
            n_samples=1,
            is_plot_std=True,
            **kwargs,
        )
    plt.tight_layout()
    return fig


# --------------------------------------------------------------------------- #
# ConvCNP anatomy gif (reference ntbks_helpers.py:485-720 gif_explain)
# --------------------------------------------------------------------------- #


def select_labels(dataset, label):
    """Subset an image dataset to one class label."""
    import torch as _torch

    dataset = copy.deepcopy(dataset)
    targets = (
        dataset.targets
        if _torch.is_tensor(dataset.targets)
        else _torch.as_tensor(dataset.targets)
    )
    filt = targets == label
    dataset.data = dataset.data[filt]
    dataset.targets = targets[filt]
    return dataset


def _convcnp_stages(model, X_cntxt, Y_cntxt, X_trgt):
    """Run a ConvCNP forward stage by stage, returning the intermediate
    representations the anatomy gif plots (reference's `splitted_forward`):
    pre-resizer SetConv output (values + density channel), the post-CNN
    induced representation, the target representation, and the predictive."""
    import torch as _torch

    X_cntxt_e = model.x_encoder(X_cntxt)
    X_trgt_e = model.x_encoder(X_trgt)
    X_induced = model._get_X_induced(X_cntxt_e)

    resizer = model.cntxt_to_induced.resizer
    model.cntxt_to_induced.resizer = _torch.nn.Identity()
    try:
        R_setconv = model.cntxt_to_induced(X_cntxt_e, X_induced, Y_cntxt)
    finally:
        model.cntxt_to_induced.resizer = resizer

    R = model.encode_globally(X_cntxt_e, Y_cntxt)
    R_trgt = model.trgt_dependent_representation(X_cntxt_e, None, R, X_trgt_e)
    p_yCc = model.decode(X_trgt_e, R_trgt)
    return X_induced, R_setconv, R, R_trgt, p_yCc


def gif_explain(
    save_filename,
    dataset,
    model,
    plot_config_kwargs=dict(),
    seed=123,
    n_cntxt=10,
    fps=0.5,
    length_scale_delta=0,
):
    """Step-by-step ConvCNP anatomy gif: context points -> SetConv values ->
    density channel -> post-CNN channels -> target representation ->
    predictive distribution."""
    import torch as _torch

    from npf.train.helpers import set_seed
    from npf.viz.helpers import fig2img
    from npf.viz.viz_1d import _plot_posterior_predefined_cntxt

    set_seed(seed)
    X, Y = dataset.get_samples(n_samples=1, n_points=dataset.n_points)
    X_cntxt, Y_cntxt, X_trgt, Y_trgt = get_n_cntxt(n_cntxt)(X, Y)

    model = copy.deepcopy(model).cpu().eval()
    if length_scale_delta:
        rbf = model.cntxt_to_induced.radial_basis_func
        rbf.length_scale_param = _torch.nn.Parameter(
            rbf.length_scale_param + length_scale_delta
        )

    X_plot = _torch.linspace(-1, 1, model.density_induced * 2).view(1, -1, 1)
    with _torch.no_grad():
        X_induced, R_setconv, R, R_trgt, p_yCc = _convcnp_stages(
            model, X_cntxt, Y_cntxt, X_plot
        )
    burn = model.density_induced // 2
    Xi = X_induced[0, burn:-burn, 0]

    def stage_fig(curves, labels, title, show_cntxt=True, text=None):
        with plot_config(**plot_config_kwargs):
            fig, ax = plt.subplots(1, 1, figsize=(11, 5))
            from npf.utils.helpers import rescale_range

            for cur, lab in zip(curves, labels):
                x, y, style = cur
                ax.plot(
                    rescale_range(x.numpy(), (-1, 1), dataset.min_max),
                    y.numpy(), style, label=lab, alpha=0.8,
                )
            if show_cntxt:
                ax.scatter(
                    rescale_range(X_cntxt[0, :, 0].numpy(), (-1, 1), dataset.min_max),
                    Y_cntxt[0, :, 0].numpy(), c="k", zorder=3,
                )
            if text is not None:
                ax.text(
                    0.5, 0.5, text, ha="center", va="center", fontsize=32,
                    transform=ax.transAxes,
                )
            ax.set_xlim(list(dataset.min_max))
            if title:
                ax.set_title(title)
            if labels and labels[0]:
                ax.legend()
        return fig

    figs = []

    def add(fig):
        figs.append(fig2img(fig))
        plt.close(fig)

    # 1. the context set alone
    add(stage_fig([], [], "Context set"))
    # 2. announce the SetConv
    add(stage_fig([], [], None, text="Apply SetConv"))
    # 3. SetConv values at induced points
    vals = R_setconv[0, burn:-burn, :-1].mean(-1)
    add(stage_fig([(Xi, vals, "-")], ["SetConv"], "SetConv output"))
    # 4. the density channel
    dens = R_setconv[0, burn:-burn, -1]
    add(stage_fig([(Xi, dens, "-")], ["density"], "Density channel"))
    # 5. announce the CNN
    add(stage_fig([], [], None, text="Apply CNN"))
    # 6. a few post-CNN channels
    chans = [(Xi, R[0, burn:-burn, c], "-") for c in range(3)]
    add(stage_fig(chans, [f"channel {c}" for c in range(3)], "Post-CNN channels"))
    # 7. the predictive
    loc = p_yCc.base_dist.loc[0, 0, :, 0]
    scale = p_yCc.base_dist.scale[0, 0, :, 0]
    with plot_config(**plot_config_kwargs):
        fig, ax = plt.subplots(1, 1, figsize=(11, 5))
        from npf.utils.helpers import rescale_range

        xs = rescale_range(X_plot[0, :, 0].numpy(), (-1, 1), dataset.min_max)
        ax.plot(xs, loc.numpy(), "b-", label="Predictive mean")
        ax.fill_between(
            xs, (loc - scale).numpy(), (loc + scale).numpy(),
            alpha=0.2, color="tab:blue",
        )
        ax.scatter(
            rescale_range(X_cntxt[0, :, 0].numpy(), (-1, 1), dataset.min_max),
            Y_cntxt[0, :, 0].numpy(), c="k", zorder=3,
        )
        ax.set_xlim(list(dataset.min_max))
        ax.set_title("Predictive distribution")
        ax.legend()
    add(fig)

    from PIL import Image

    frames = [Image.fromarray(f).convert("P", palette=Image.ADAPTIVE) for f in figs]
    w = min(f.size[0] for f in frames)
    h = min(f.size[1] for f in frames)
    frames = [f.resize((w, h)) for f in frames]
    frames[0].save(
        save_filename, save_all=True, append_images=frames[1:],
        duration=int(1000 / fps), loop=0,
    )
    return save_filename
