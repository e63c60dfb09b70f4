"""1D (GP-regression) visualization — capability of the reference's
utils/visualize/visualize_1d.py (plot_losses:24, plot_dataset_samples_1d:50,
plot_prior_samples_1d:80, plot_posterior_samples_1d:99, gen_p_y_pred:280,
_plot_posterior_predefined_cntxt:309), reimplemented for this framework's
trainer history and without seaborn."""

import matplotlib.pyplot as plt
import numpy as np
import torch

from npf.neuralproc.base import LatentNeuralProcessFamily
from npf.train.helpers import set_seed
from npf.utils.helpers import rescale_range

from .helpers import plot_config

DFLT_FIGSIZE = (11, 5)

__all__ = [
    "plot_dataset_samples_1d",
    "plot_prior_samples_1d",
    "plot_posterior_samples_1d",
    "plot_losses",
    "gen_p_y_pred",
]


def plot_losses(
    history, title=None, figsize=DFLT_FIGSIZE, ax=None, mode="both", label_sfx=""
):
    """Plot train/valid loss curves from an NPFTrainer history (list of epoch
    records with "train_loss"/"valid_loss"; reference visualize_1d.py:24
    consumed the skorch history instead)."""
    if ax is None:
        _, ax = plt.subplots(1, 1, figsize=figsize)
    if mode in ("both", "validation"):
        vals = [ep.get("valid_loss") for ep in history]
        if any(v is not None for v in vals):
            ax.plot(vals, label="Validation" + label_sfx)
    if mode in ("both", "training"):
        ax.plot([ep.get("train_loss") for ep in history], label="Training" + label_sfx)
    ax.legend()
    ax.set_ylabel("Negative Log Likelihood")
    ax.set_xlabel("Number of Epochs")
    if title is not None:
        ax.set_title(title)
    return ax


def plot_dataset_samples_1d(
    dataset,
    n_samples=10,
    title="Dataset",
    figsize=DFLT_FIGSIZE,
    ax=None,
    plot_config_kwargs={},
    seed=123,
):
    """Plot random function draws from a 1D meta-dataset."""
    rng = np.random.RandomState(seed)
    with plot_config(**plot_config_kwargs):
        if ax is None:
            _, ax = plt.subplots(1, 1, figsize=figsize)
        alpha = 0.5 + 1 / (n_samples ** 0.5 + 1)
        for _ in range(n_samples):
            x, y = dataset[rng.randint(len(dataset))]
            x = rescale_range(x, (-1, 1), dataset.min_max)
            ax.plot(x.numpy(), y.numpy(), alpha=alpha)
        ax.set_xlim(*dataset.min_max)
        if title is not None:
            ax.set_title(title, fontsize=14)
    return ax


def _assert_single_task(*tensors):
    for t in tensors:
        if t is not None and not (t.dim() == 3 and t.shape[0] == 1):
            raise ValueError(f"expected [1, n, dim] inputs, got {tuple(t.shape)}")


def _widen_ylim(y_min, y_max):
    y_min = y_min * 1.2 if y_min < 0 else y_min / 0.9
    y_max = y_max * 1.2 if y_max > 0 else y_max / 0.9
    return y_min, y_max


def gen_p_y_pred(model, X_cntxt, Y_cntxt, X_trgt, n_samples):
    """Yield (mean, std) curves of the model's posterior predictive.

    LNPF: one curve per latent sample (temporarily sets n_z_samples_test).
    CNPF with n_samples>1: yields noise draws (std=None) then the mean curve
    (reference visualize_1d.py:280-306)."""
    if X_cntxt is None:
        X_cntxt = torch.zeros(1, 0, model.x_dim)
        Y_cntxt = torch.zeros(1, 0, model.y_dim)

    if isinstance(model, LatentNeuralProcessFamily):
        old = model.n_z_samples_test
        model.n_z_samples_test = n_samples
        try:
            p_yCc, *_ = model.forward(X_cntxt, Y_cntxt, X_trgt)
        finally:
            model.n_z_samples_test = old
    else:
        p_yCc, *_ = model.forward(X_cntxt, Y_cntxt, X_trgt)
        if n_samples > 1:
            draws = p_yCc.sample((n_samples,)).detach().numpy()
            for i in range(draws.shape[0]):
                yield draws[i, 0, 0, :, 0].flatten(), None

    locs = p_yCc.base_dist.loc.detach().numpy()
    scales = p_yCc.base_dist.scale.detach().numpy()
    for i in range(locs.shape[0]):
        yield locs[i, 0, :, 0].flatten(), scales[i, 0, :, 0].flatten()


def _plot_posterior_predefined_cntxt(
    model,
    X_cntxt,
    Y_cntxt,
    X_trgt,
    Y_trgt=None,
    n_samples=1,
    is_plot_std=False,
    train_min_max=(-2, 2),
    model_label="Model",
    scatter_label=None,
    alpha_init=1,
    mean_std_colors=("b", "tab:blue"),
    title=None,
    figsize=DFLT_FIGSIZE,
    ax=None,
    is_smooth=True,
    is_legend=True,
    scatter_kwargs={},
    kwargs_std={},
    **kwargs,
):
    """Plot posterior-predictive curves for one fixed context set."""
    _assert_single_task(X_cntxt, Y_cntxt, X_trgt)
    mean_color, std_color = mean_std_colors
    is_conditioned = X_cntxt is not None and X_cntxt.shape[1] >= 1

    model.eval()
    model = model.cpu()

    xt = X_trgt.numpy()[0].flatten()
    interp_mask = (xt > -1) & (xt < 1)
    xt_plot = rescale_range(xt, (-1, 1), train_min_max)
    x_min, x_max = xt_plot.min(), xt_plot.max()
    if is_conditioned:
        xc_plot = rescale_range(X_cntxt.numpy()[0].flatten(), (-1, 1), train_min_max)

    alpha = alpha_init / n_samples ** 0.5
    if ax is None:
        _, ax = plt.subplots(1, 1, figsize=figsize)
        y_min, y_max = 0.0, 0.0
    else:
        y_min, y_max = ax.get_ylim()

    for i, (mean_y, std_y) in enumerate(
        gen_p_y_pred(model, X_cntxt, Y_cntxt, X_trgt, n_samples)
    ):
        if not is_smooth:
            kwargs["linestyle"] = ""
            kwargs.setdefault("marker", ".")
        label = {"label": model_label} if i == 0 else {}
        ax.plot(xt_plot, mean_y, alpha=alpha, c=mean_color, **label, **kwargs)
        if is_plot_std:
            if std_y is None:
                raise ValueError(
                    f"cannot plot std of noise draws from a CNPF (n_samples={n_samples})"
                )
            if is_smooth:
                kw = dict(alpha=alpha / 7)
                kw.update(kwargs_std)
                ax.fill_between(
                    xt_plot, mean_y - std_y, mean_y + std_y, color=std_color, **kw
                )
            else:
                kw = dict(alpha=alpha / 7, capsize=3, fmt="none", **kwargs)
                kw.update(kwargs_std)
                ax.errorbar(
                    xt_plot, mean_y, yerr=std_y, ecolor=std_color,
                    color=mean_color, **kw,
                )
            y_min = min(y_min, float((mean_y - std_y)[interp_mask].min()))
            y_max = max(y_max, float((mean_y + std_y)[interp_mask].max()))
        else:
            y_min = min(y_min, float(mean_y[interp_mask].min()))
            y_max = max(y_max, float(mean_y[interp_mask].max()))

    if Y_trgt is not None:
        _assert_single_task(Y_trgt)
        yt = Y_trgt.numpy()[0, :, 0].flatten()
        ax.plot(xt_plot, yt, "--k", alpha=0.7, label="Target Function")
        y_min = min(y_min, float(yt.min()))
        y_max = max(y_max, float(yt.max()))

    if is_conditioned:
        if scatter_label is not None:
            scatter_kwargs = dict(scatter_kwargs, label=scatter_label)
        ax.scatter(xc_plot, Y_cntxt[0, :, 0].numpy(), c="k", **scatter_kwargs)
        x_min = min(float(xc_plot.min()), x_min)
        x_max = max(float(xc_plot.max()), x_max)

    ax.set_xlim(x_min, x_max)
    ax.set_ylim(_widen_ylim(y_min, y_max))

    # mark the training range when extrapolating (reference :499-515)
    for bound, beyond in ((train_min_max[1], x_max > train_min_max[1]),
                          (train_min_max[0], x_min < train_min_max[0])):
        if beyond:
            ax.axvline(
                x=bound, color="r", linestyle=":", alpha=alpha_init / 2,
                label="Extrapolation Boundary",
            )

    if title is not None:
        ax.set_title(title, fontsize=14)
    if is_legend:
        ax.legend()
    return ax


def plot_prior_samples_1d(
    model, test_min_max=None, train_min_max=(-2, 2), n_trgt=256, **kwargs
):
    """Plot function draws from the model prior (no context)."""
    if test_min_max is None:
        test_min_max = train_min_max
    lo, hi = rescale_range(np.array(test_min_max), train_min_max, (-1, 1))
    X_trgt = torch.linspace(float(lo), float(hi), n_trgt).view(1, -1, 1)
    return _plot_posterior_predefined_cntxt(model, None, None, X_trgt, **kwargs)


def plot_posterior_samples_1d(
    X,
    Y,
    get_cntxt_trgt,
    model,
    compare_model=None,
    model_labels=dict(main="Model", compare="Compare", generator="Oracle GP"),
    generator=None,
    is_plot_real=True,
    train_min_max=(-2, 2),
    ax=None,
    seed=None,
    is_fill_generator_std=True,
    y_lim=(None, None),
    is_legend=True,
    plot_config_kwargs={},
    **kwargs,
):
    """Plot the posterior predictive of one (optionally two) models on a random
    context/target split of the task (X, Y), with an optional oracle-GP overlay
    (reference visualize_1d.py:99-255)."""
    with plot_config(**plot_config_kwargs):
        set_seed(seed)
        _assert_single_task(X, Y)
        X_cntxt, Y_cntxt, X_trgt, Y_trgt = get_cntxt_trgt(X, Y)
        alpha_init = 1 if compare_model is None else 0.5

        ax = _plot_posterior_predefined_cntxt(
            model, X_cntxt, Y_cntxt, X_trgt,
            train_min_max=train_min_max,
            Y_trgt=Y_trgt if is_plot_real else None,
            model_label=model_labels["main"],
            alpha_init=alpha_init,
            mean_std_colors=("b", "tab:blue"),
            ax=ax, is_legend=is_legend, **kwargs,
        )
        if compare_model is not None:
            ax = _plot_posterior_predefined_cntxt(
                compare_model, X_cntxt, Y_cntxt, X_trgt,
                train_min_max=train_min_max,
                model_label=model_labels["compare"],
                alpha_init=alpha_init,
                mean_std_colors=("m", "tab:pink"),
                ax=ax, is_legend=is_legend, **kwargs,
            )

        if generator is not None:
            import sklearn.base

            xc = rescale_range(X_cntxt, (-1, 1), train_min_max).numpy()[0]
            generator = sklearn.base.clone(generator)
            if X_cntxt.shape[1] > 0:
                generator.fit(xc, Y_cntxt.numpy()[0])
            xt = rescale_range(X, (-1, 1), train_min_max).numpy()[0].flatten()
            mean_y, std_y = generator.predict(xt[:, None], return_std=True)
            mean_y = mean_y.flatten()
            ax.plot(
                xt, mean_y, alpha=alpha_init / 1.5, c="g",
                label=model_labels["generator"],
            )
            if is_fill_generator_std:
                ax.fill_between(
                    xt, mean_y - std_y, mean_y + std_y,
                    alpha=alpha_init / 10, color="tab:green",
                )
            else:
                for sgn in (-1, 1):
                    ax.plot(
                        xt, mean_y + sgn * std_y, alpha=alpha_init / 2,
                        c="g", linestyle="--",
                    )
            if is_legend:
                ax.legend()
            ax.set_ylim([y_lim[0], y_lim[1]])
    return ax
