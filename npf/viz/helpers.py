"""Plot/gif infrastructure (capability of reference utils/visualize/helpers.py).

Implemented without seaborn/imageio/skimage (not in this image): styling is a
pure-matplotlib rc context, `fig2img` rasterizes through PIL, and `giffify`
writes GIFs with PIL's animated-save path.
"""

import contextlib
import io
import logging

import matplotlib.pyplot as plt
import numpy as np

__all__ = ["giffify", "plot_config", "fig2img", "make_grid", "kdeplot"]
logger = logging.getLogger(__name__)

# matplotlib rc approximations of the seaborn presets used by the reference
_STYLES = {
    "ticks": {
        "axes.grid": False,
        "xtick.direction": "out",
        "ytick.direction": "out",
        "axes.spines.top": False,
        "axes.spines.right": False,
    },
    "whitegrid": {"axes.grid": True, "grid.color": "0.9"},
    "darkgrid": {
        "axes.grid": True,
        "axes.facecolor": "#EAEAF2",
        "grid.color": "white",
    },
    "white": {"axes.grid": False},
    "dark": {"axes.grid": False, "axes.facecolor": "#EAEAF2"},
}
_CONTEXT_SCALE = {"paper": 0.8, "notebook": 1.0, "talk": 1.3, "poster": 1.6}

# colorblind-safe palette (the reference's default seaborn palette)
COLORBLIND = [
    "#0173B2", "#DE8F05", "#029E73", "#D55E00", "#CC78BC",
    "#CA9161", "#FBAFE4", "#949494", "#ECE133", "#56B4E9",
]


def fig2img(fig, dpi=200, format="png", is_transparent=False):
    """Rasterize a matplotlib figure to an RGB(A) uint8 numpy array."""
    from PIL import Image

    buf = io.BytesIO()
    fig.savefig(
        buf, dpi=dpi, bbox_inches="tight", format=format, transparent=is_transparent
    )
    buf.seek(0)
    return np.asarray(Image.open(buf).convert("RGBA"))


@contextlib.contextmanager
def plot_config(
    style="ticks",
    context="notebook",
    palette="colorblind",
    font_scale=1,
    font="sans-serif",
    is_ax_off=False,
    rc=dict(),
    set_kwargs=dict(),
    despine_kwargs=dict(),
):
    """Temporary matplotlib style/context (seaborn-free `plot_config` analog).

    Accepts the reference's signature; `palette` other than "colorblind" may be
    a list of colors; `despine_kwargs` is accepted for API compatibility (the
    "ticks"/"white" styles already hide top/right spines).
    """
    defaults = plt.rcParams.copy()
    try:
        updates = dict(_STYLES.get(style, {})) if isinstance(style, str) else dict(style)
        scale = _CONTEXT_SCALE.get(context, 1.0) * font_scale
        for key, base in [
            ("font.size", 10), ("axes.titlesize", 12), ("axes.labelsize", 11),
            ("xtick.labelsize", 10), ("ytick.labelsize", 10), ("legend.fontsize", 10),
        ]:
            updates[key] = base * scale
        colors = COLORBLIND if palette == "colorblind" else list(palette)
        updates["axes.prop_cycle"] = plt.cycler(color=colors)
        updates["font.family"] = font
        updates.update(rc)
        plt.rcParams.update(updates)
        yield
        last_fig = plt.gcf()
        for ax in last_fig.axes:
            if set_kwargs:
                ax.set(**set_kwargs)
            if is_ax_off:
                ax.axis("off")
    finally:
        plt.rcParams.update(defaults)


def _resize_img(img, hw):
    from PIL import Image

    return np.asarray(Image.fromarray(img).resize((hw[1], hw[0])))


def giffify(
    save_filename,
    gen_single_fig,
    sweep_parameter,
    sweep_values,
    fps=2,
    quality=70,
    is_transparent=False,
    **kwargs,
):
    """Make a gif by calling `gen_single_fig(**{sweep_parameter: v}, **kwargs)`
    for every v in `sweep_values` (reference utils/visualize/helpers.py:104)."""
    from PIL import Image

    frames = []
    size = None
    for i, v in enumerate(sweep_values):
        fig = gen_single_fig(**{sweep_parameter: v}, **kwargs)
        plt.close()
        img = fig2img(fig, is_transparent=is_transparent)
        if size is None:
            size = img.shape[:2]
        elif img.shape[:2] != size:
            img = _resize_img(img, size)
        frames.append(Image.fromarray(img).convert("P", palette=Image.ADAPTIVE))
    frames[0].save(
        save_filename,
        save_all=True,
        append_images=frames[1:],
        duration=int(1000 / fps),
        loop=0,
    )


def make_grid(tensor, nrow=8, padding=2, pad_value=0.0):
    """Tile a [N,C,H,W] tensor into one [C, H', W'] image grid
    (torchvision.utils.make_grid capability; torchvision is not installed)."""
    import torch

    if isinstance(tensor, (list, tuple)):
        tensor = torch.stack(tensor, dim=0)
    if tensor.dim() == 3:
        tensor = tensor.unsqueeze(0)
    n, c, h, w = tensor.shape
    ncols = min(nrow, n)
    nrows = (n + ncols - 1) // ncols
    grid = tensor.new_full(
        (c, padding + nrows * (h + padding), padding + ncols * (w + padding)),
        pad_value,
    )
    for i in range(n):
        r, col = divmod(i, ncols)
        grid[
            :,
            padding + r * (h + padding) : padding + r * (h + padding) + h,
            padding + col * (w + padding) : padding + col * (w + padding) + w,
        ] = tensor[i]
    return grid


def kdeplot(data, ax=None, label=None, shade=True, cut=0, bw_method=None, **kwargs):
    """Gaussian-KDE density plot (sns.kdeplot capability via scipy)."""
    from scipy.stats import gaussian_kde

    if ax is None:
        _, ax = plt.subplots()
    data = np.asarray(data, dtype=np.float64)
    data = data[np.isfinite(data)]
    if data.size < 2 or np.ptp(data) == 0:
        ax.axvline(data.mean() if data.size else 0.0, label=label, **kwargs)
        return ax
    kde = gaussian_kde(data, bw_method=bw_method)
    pad = 0 if cut == 0 else cut * kde.factor * data.std()
    xs = np.linspace(data.min() - pad, data.max() + pad, 512)
    ys = kde(xs)
    (line,) = ax.plot(xs, ys, label=label, **kwargs)
    if shade:
        ax.fill_between(xs, ys, alpha=0.25, color=line.get_color())
    return ax
