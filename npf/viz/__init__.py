"""Visualization suite (capability parity with reference utils/visualize/*)."""

from .helpers import fig2img, giffify, kdeplot, make_grid, plot_config  # noqa: F401
from .viz_1d import (  # noqa: F401
    gen_p_y_pred,
    plot_dataset_samples_1d,
    plot_losses,
    plot_posterior_samples_1d,
    plot_prior_samples_1d,
)
from .viz_imgs import (  # noqa: F401
    CntxtTrgtDict,
    get_posterior_samples,
    marginal_log_like,
    plot_dataset_samples_imgs,
    plot_img_marginal_pred,
    plot_posterior_samples,
    plot_qualitative_with_kde,
    points_to_grid,
    sarle,
)
