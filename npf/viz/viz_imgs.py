"""Image-completion visualization — capability of the reference's
utils/visualize/visualize_imgs.py (plot_dataset_samples_imgs:37,
get_posterior_samples:58, plot_img_marginal_pred:117, plot_posterior_samples:210,
plot_qualitative_with_kde:467, points_to_grid:889, marginal_log_like:839,
sarle:850), reimplemented for this framework (no torchvision/seaborn/griddata
hard dependency; KDE via scipy)."""

import logging
import math
import os
import random

import matplotlib.pyplot as plt
import matplotlib.ticker as ticker
import numpy as np
import torch

from npf.data.dataloader import cntxt_trgt_collate
from npf.neuralproc import GridConvCNP
from npf.neuralproc.base import LatentNeuralProcessFamily
from npf.train.helpers import set_seed
from npf.utils.datasplit import GridCntxtTrgtGetter
from npf.utils.helpers import MultivariateNormalDiag, channels_to_2nd_dim, prod
from npf.utils.predict import SamplePredictor

from .helpers import kdeplot, make_grid

DFLT_FIGSIZE = (17, 9)
EVAL_FILENAME = "eval.csv"

__all__ = [
    "plot_dataset_samples_imgs",
    "get_posterior_samples",
    "plot_img_marginal_pred",
    "plot_posterior_samples",
    "plot_qualitative_with_kde",
    "points_to_grid",
    "marginal_log_like",
    "sarle",
    "CntxtTrgtDict",
]

logger = logging.getLogger(__name__)


class CntxtTrgtDict(dict):
    """dict of precomputed context/target tensors carrying `upscale_factor`."""

    def __init__(self, *args, upscale_factor=1, **kwargs):
        self.upscale_factor = upscale_factor
        super().__init__(*args, **kwargs)


def plot_dataset_samples_imgs(
    dataset, n_plots=4, figsize=DFLT_FIGSIZE, ax=None, pad_value=1, seed=123, title=None
):
    """Show a grid of random dataset images."""
    set_seed(seed)
    if ax is None:
        _, ax = plt.subplots(figsize=figsize)
    imgs = torch.stack(
        [dataset[random.randint(0, len(dataset) - 1)][0] for _ in range(n_plots)]
    )
    grid = make_grid(imgs, nrow=2, pad_value=pad_value)
    ax.imshow(grid.permute(1, 2, 0).numpy())
    ax.axis("off")
    if title is not None:
        ax.set_title(title)


def keep_most_different_samples_(samples, n_samples, p=2):
    """In-place: keep the n_samples predictive draws farthest apart (mean Lp)."""
    n_avail = samples.batch_shape[0]
    assert n_samples <= n_avail
    loc, scale = samples.base_dist.loc, samples.base_dist.scale
    chosen = [0]
    pool = set(range(1, n_avail))
    for _ in range(n_samples - 1):
        dists = {
            j: float(np.mean([torch.dist(loc[c], loc[j], p=p) for c in chosen]))
            for j in pool
        }
        nxt = max(dists, key=dists.get)
        chosen.append(nxt)
        pool.remove(nxt)
    samples.base_dist.loc = loc[chosen]
    samples.base_dist.scale = scale[chosen]


def get_posterior_samples(
    data,
    get_cntxt_trgt,
    model,
    is_uniform_grid=True,
    img_indcs=None,
    n_plots=4,
    seed=123,
    n_samples=3,
    is_select_different=False,
):
    """Run the model on sampled (or provided) image context/target splits and
    return (predictive, mask_cntxt, Y_cntxt, mask_trgt)."""
    set_seed(seed)
    model.eval()

    if isinstance(get_cntxt_trgt, dict):
        device = next(model.parameters()).device
        mask_cntxt = get_cntxt_trgt["X_cntxt"].to(device)
        Y_cntxt = get_cntxt_trgt["Y_cntxt"].to(device)
        mask_trgt = get_cntxt_trgt["X_trgt"].to(device)
    else:
        if img_indcs is None:
            img_indcs = [random.randint(0, len(data) - 1) for _ in range(n_plots)]
        imgs = [data[i] for i in img_indcs]
        batch, _ = cntxt_trgt_collate(get_cntxt_trgt, is_return_masks=is_uniform_grid)(
            imgs
        )
        mask_cntxt, Y_cntxt, mask_trgt = (
            batch["X_cntxt"], batch["Y_cntxt"], batch["X_trgt"],
        )

    y_pred = SamplePredictor(model, is_dist=True)(mask_cntxt, Y_cntxt, mask_trgt)

    if is_select_different:
        keep_most_different_samples_(y_pred, n_samples)
    elif isinstance(n_samples, int):
        y_pred.base_dist.loc = y_pred.base_dist.loc[:n_samples]
        y_pred.base_dist.scale = y_pred.base_dist.scale[:n_samples]
    elif n_samples is not None:
        raise ValueError(f"unknown n_samples={n_samples}")
    return y_pred, mask_cntxt, Y_cntxt, mask_trgt


def marginal_log_like(predictive, samples):
    """exp of the z-marginalized log-likelihood of `samples`."""
    log_p = predictive.log_prob(samples)
    ll = torch.logsumexp(log_p, 0) - math.log(predictive.batch_shape[0])
    return ll.exp()


def sarle(out, axis=0):
    """Sarle's bimodality coefficient along `axis`."""
    import scipy.stats

    k = scipy.stats.kurtosis(out, axis=axis, fisher=True)
    g = scipy.stats.skew(out, axis=axis)
    n = out.shape[1]
    denom = k + 3 * (n - 1) ** 2 / ((n - 2) * (n - 2))
    return (g ** 2 + 1) / denom


def idcs_grid_to_idcs_flatten(idcs, grid_shape):
    """Grid-coordinate indices -> flat indices."""
    for i, _ in enumerate(grid_shape):
        idcs[:, :, i] *= prod(grid_shape[i + 1 :])
    return idcs.sum(-1)


def points_to_grid(
    X, Y, grid_shape, background=torch.tensor([0.0, 0.0, 0.0]), downscale_factor=1
):
    """Scatter (X in [-1,1] coords, Y values) back onto a pixel grid."""
    batch_size, _, y_dim = Y.shape
    X = X.clone() / downscale_factor
    for i, size in enumerate(grid_shape):
        X[:, :, i] = (X[:, :, i] + 1) * (size - 1) / 2
    idcs = idcs_grid_to_idcs_flatten(X.round().long(), grid_shape)

    canvas = (
        background.view(1, *(1 for _ in grid_shape), y_dim)
        .repeat(batch_size, *grid_shape, 1)
        .view(batch_size, -1, y_dim)
    )
    mask = torch.zeros(batch_size, canvas.size(1), 1).bool()
    for b in range(batch_size):
        canvas[b, idcs[b], :] = Y[b]
        mask[b, idcs[b], :] = True
    return (
        canvas.view(batch_size, *grid_shape, y_dim),
        mask.view(batch_size, *grid_shape, 1),
    )


def get_downscale_factor(get_cntxt_trgt):
    """Test-time upscale factor of a splitter (1 when absent)."""
    return getattr(get_cntxt_trgt, "upscale_factor", 1)


def remove_axis(ax, is_rm_ticks=True, is_rm_spines=True):
    if is_rm_spines:
        for side in ("right", "top", "bottom", "left"):
            ax.spines[side].set_visible(False)
        ax.set_frame_on(False)
    if is_rm_ticks:
        ax.tick_params(bottom=False, left=False)


def get_img_toplot(
    data, to_plot, mask, is_uniform_grid, downscale_factor=1, is_mask=True
):
    """Compose an image for display: masked values on missing-px background."""
    mask_toapply = mask if is_mask else torch.ones_like(mask).bool()
    if is_uniform_grid:
        background = (
            data.missing_px_color.view(1, 1, 1, 3).expand(*to_plot.shape).clone()
        )
        if mask.size(-1) == 1:
            out = torch.where(mask_toapply, to_plot, background)
        else:
            background[mask_toapply.squeeze(-1)] = to_plot.reshape(-1, 3)
            out = background.clone()
    else:
        out, _ = points_to_grid(
            mask_toapply, to_plot, data.shape[1:],
            background=data.missing_px_color,
            downscale_factor=downscale_factor,
        )
        _, mask = points_to_grid(
            mask, to_plot, data.shape[1:], downscale_factor=downscale_factor
        )
    return out, mask


def _grid_interpolate(mask_cntxt, out_cntxt, method):
    """Interpolation baselines over the context pixels (scipy griddata)."""
    from scipy.interpolate import griddata

    outs = []
    for i in range(mask_cntxt.shape[0]):
        m = mask_cntxt[i, :, :, 0]
        coord_y, coord_x = m.nonzero().unbind(1)
        grid_x, grid_y = np.meshgrid(
            np.arange(out_cntxt.shape[2]), np.arange(out_cntxt.shape[1])
        )
        interp = griddata(
            (coord_x.numpy(), coord_y.numpy()),
            out_cntxt[i, coord_y, coord_x].numpy(),
            (grid_x, grid_y),
            method=method,
        )
        outs.append(torch.from_numpy(np.nan_to_num(interp)).float())
    return torch.stack(outs, dim=0)


def plot_posterior_samples(
    data,
    get_cntxt_trgt,
    model,
    is_uniform_grid=True,
    img_indcs=None,
    n_plots=4,
    imgsize=(7, 4),
    ax=None,
    seed=123,
    is_return=False,
    is_hrztl_cat=False,
    n_samples=1,
    outs=None,
    is_select_different=False,
    is_plot_std=False,
    interp_baselines=[],
    is_add_annot=True,
    rotate_annot=None,
    is_mask_cntxt=True,
    labels=dict(mean="Pred. Mean", std="Pred. Std.", baseline="{baseline} Interp."),
):
    """Plot context / predicted mean (or samples) / std rows for image tasks.

    Mirrors reference visualize_imgs.py:210-464 behaviorally: rows are context,
    n_samples predictions, optional std, optional interpolation baselines;
    columns are the different images."""
    if outs is None:
        y_pred, mask_cntxt, X, mask_trgt = get_posterior_samples(
            data, get_cntxt_trgt, model,
            is_uniform_grid=is_uniform_grid,
            img_indcs=img_indcs, n_plots=n_plots, seed=seed,
            n_samples=n_samples, is_select_different=is_select_different,
        )
    else:
        y_pred, mask_cntxt, X, mask_trgt = outs

    if n_samples > 1 and not isinstance(model, LatentNeuralProcessFamily):
        if is_plot_std:
            raise ValueError("cannot plot std when sampling from a CNPF")
        mean_ys = y_pred.sample((n_samples,))[:, 0, ...]
    else:
        mean_ys = y_pred.base_dist.loc

    if n_samples > mean_ys.size(0):
        raise ValueError(
            f"n_samples={n_samples} > available latent samples {mean_ys.size(0)}"
        )

    if isinstance(get_cntxt_trgt, dict):
        n_plots = get_cntxt_trgt["X_cntxt"].size(0)

    dim_grid = 2 if is_uniform_grid else 1
    if is_uniform_grid:
        mean_ys = mean_ys.view(n_samples, *X.shape)
    if X.shape[-1] == 1:
        X = X.expand(-1, *[-1] * dim_grid, 3)
        mean_ys = mean_ys.expand(n_samples, -1, *[-1] * dim_grid, 3)
    std_ys = y_pred.base_dist.scale.expand(*mean_ys.shape)

    down = get_downscale_factor(get_cntxt_trgt)
    out_cntxt, mask_cntxt = get_img_toplot(
        data, X, mask_cntxt, is_uniform_grid,
        downscale_factor=down, is_mask=is_mask_cntxt,
    )

    rows = [out_cntxt]
    row_labels = ["Context"]
    for i in range(n_samples):
        out_pred, _ = get_img_toplot(
            data, mean_ys[i], mask_trgt, is_uniform_grid, downscale_factor=down
        )
        rows.append(out_pred)
        row_labels.append(f"Sample {i + 1}" if n_samples > 1 else labels["mean"])
    if is_plot_std:
        out_std, _ = get_img_toplot(
            data, std_ys[n_samples - 1], mask_trgt, is_uniform_grid,
            downscale_factor=down,
        )
        rows.append(out_std)
        row_labels.append(f"Std {n_samples}" if n_samples > 1 else labels["std"])
    for interp in interp_baselines:
        rows.append(_grid_interpolate(mask_cntxt, out_cntxt, interp))
        row_labels.append(labels["baseline"].format(baseline=interp.title()))

    outs_t = channels_to_2nd_dim(torch.cat(rows, dim=0)).detach()
    if is_hrztl_cat:
        tmp = []
        for i in range(n_plots):
            tmp.extend(outs_t[i::n_plots])
        outs_t = tmp

    n_per_row = n_plots
    n_per_col = len(row_labels)
    if is_hrztl_cat:
        n_per_row, n_per_col = n_per_col, n_per_row
    grid = make_grid(outs_t, nrow=n_per_row, pad_value=1.0)
    if is_return:
        return grid

    if ax is None:
        _, ax = plt.subplots(
            figsize=(imgsize[0] * n_per_row, imgsize[1] * n_per_col)
        )
    ax.imshow(grid.permute(1, 2, 0).numpy())

    if is_add_annot:
        idx_text = 2 if is_hrztl_cat else 1
        middle = data.shape[idx_text] // 2 + 1
        tick_pos = [middle * (2 * i + 1) for i in range(len(row_labels))]
        if is_hrztl_cat:
            rotate_annot = 20 if rotate_annot is None else rotate_annot
            ax.xaxis.set_major_locator(ticker.FixedLocator(tick_pos))
            ax.set_xticklabels(row_labels, rotation=rotate_annot, ha="right")
            ax.set_yticks([])
        else:
            rotate_annot = "vertical" if rotate_annot is None else rotate_annot
            ax.yaxis.set_major_locator(ticker.FixedLocator(tick_pos))
            ax.set_yticklabels(row_labels, rotation=rotate_annot, va="center")
            ax.set_xticks([])
        remove_axis(ax)
    else:
        ax.axis("off")


def plot_img_marginal_pred(
    model,
    data,
    get_cntxt_trgt,
    figsize=(11, 5),
    n_samples=5,
    is_uniform_grid=True,
    seed=123,
    n_plots_loop=1,
    wspace=0.3,
    n_marginals=5,
    n_columns=2,
    **kwargs,
):
    """Plot predictive samples next to per-pixel marginal densities; picks the
    most multimodal image (lowest median Sarle coefficient) over
    `n_plots_loop` candidates."""
    f, (ax0, ax1) = plt.subplots(
        1, 2, gridspec_kw={"width_ratios": [1, 1], "wspace": wspace}, figsize=figsize
    )
    predictive_all, mask_cntxt, X, mask_trgt = get_posterior_samples(
        data, get_cntxt_trgt, model,
        n_plots=n_plots_loop, is_uniform_grid=is_uniform_grid,
        seed=seed, n_samples=None,
    )
    if predictive_all.base_dist.loc.shape[0] == 1:
        logger.warning("single posterior sample: treating it as the marginal")

    arange = torch.linspace(0, 1, 1000)
    marg_shape = (1, 1000, 1, 1, 1) if is_uniform_grid else (1, 1000, 1, 1)
    arange_marg = arange.view(*marg_shape)

    best = float("inf")
    for i in range(n_plots_loop):
        predictive = MultivariateNormalDiag(
            predictive_all.base_dist.loc[:, i : i + 1, ...],
            predictive_all.base_dist.scale[:, i : i + 1, ...],
        )
        out = marginal_log_like(predictive, arange_marg).detach().reshape(1000, -1).numpy()
        sarles = np.nan_to_num(sarle(out), nan=np.inf)
        if i == 0 or np.median(sarles) < best:
            best = float(np.median(sarles))
            best_out, best_sarles = out, sarles
            best_pred = predictive
            best_pred.base_dist.loc = predictive.base_dist.loc[:n_samples]
            best_pred.base_dist.scale = predictive.base_dist.scale[:n_samples]
            best_sel = (
                mask_cntxt[i : i + 1], X[i : i + 1], mask_trgt[i : i + 1],
            )

    idx = np.argsort(best_sarles)[:n_marginals]
    ax1.plot(arange, best_out[:, idx], alpha=0.7)
    ax1.set_yticks([])
    ax1.set_ylabel("Marginal Predictive")
    ax1.set_xlabel("Pixel Intensity")
    ax1.set_xlim(-0.1, 1)
    ax1.set_xticks([0, 0.5, 1])

    plot_posterior_samples(
        data, get_cntxt_trgt, model,
        is_uniform_grid=is_uniform_grid, seed=seed, n_samples=n_samples,
        ax=ax0, outs=[best_pred, *best_sel],
        is_add_annot=False, n_plots=n_columns, **kwargs,
    )
    return f


def plot_qualitative_with_kde(
    named_trainer,
    dataset,
    named_trainer_compare=None,
    n_images=8,
    percentiles=None,
    figsize=DFLT_FIGSIZE,
    title=None,
    seed=123,
    height_ratios=[1, 3],
    font_size=12,
    h_pad=-3,
    x_lim={},
    is_smallest_xrange=False,
    kdeplot_kwargs={},
    n_samples=1,
    upscale_factor=1,
    **kwargs,
):
    """KDE of per-task test log-likelihood + qualitative predictions at chosen
    LL percentiles (reference visualize_imgs.py:467-770).

    `named_trainer` is `[name, trainer]` where trainer is an NPFTrainer whose
    checkpoint dir holds eval.csv."""
    kwargs["n_samples"] = n_samples
    kwargs["is_plot_std"] = False
    kwargs["is_add_annot"] = False

    if percentiles is not None:
        n_images = len(percentiles)

    plt.rcParams.update({"font.size": font_size})
    fig, axes = plt.subplots(
        2, 1, figsize=figsize, gridspec_kw={"height_ratios": height_ratios}
    )

    def _plot_kde_loglike(name, trainer):
        test_eval_file = os.path.join(trainer.chckpnt_dirname, EVAL_FILENAME)
        test_loglike = np.loadtxt(test_eval_file, delimiter=",")
        kdeplot(test_loglike, ax=axes[0], label=name, **kdeplot_kwargs)
        return test_loglike

    name, trainer = named_trainer
    test_loglike = _plot_kde_loglike(name, trainer)

    if named_trainer_compare is not None:
        _plot_kde_loglike(*named_trainer_compare)
        axes[0].legend()
    axes[0].set_xlabel("Test Log-Likelihood")
    axes[0].set_yticks([])
    if x_lim:
        axes[0].set_xlim(**x_lim)

    if percentiles is None:
        percentiles = np.linspace(0, 100, n_images)
    values = np.percentile(test_loglike, percentiles, method="nearest")
    img_indcs = [int(np.argwhere(test_loglike == v)[0, 0]) for v in values]
    if is_smallest_xrange:
        axes[0].set_xlim(values[0] - 1, values[-1] + 1)
    for v in values:
        axes[0].axvline(v, linestyle=":", alpha=0.7, c="tab:green")

    model = trainer.module_.cpu()
    is_uniform_grid = isinstance(model, GridConvCNP)
    getter = GridCntxtTrgtGetter(upscale_factor=upscale_factor)
    grids = []
    for i, idx in enumerate(img_indcs):
        g = plot_posterior_samples(
            dataset, getter, model,
            is_uniform_grid=is_uniform_grid, img_indcs=[idx],
            is_return=True, n_plots=1, **kwargs,
        )
        grids.append(g[..., 2:] if i != 0 else g)
    grid = torch.cat(grids, dim=-1)

    axes[1].imshow(grid.permute(1, 2, 0).numpy())
    axes[1].axis("off")
    if title is not None:
        axes[0].set_title(title)
    fig.tight_layout(h_pad=h_pad)
    return fig
