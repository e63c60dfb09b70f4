"""MI355X ops layer: fused CDNA4 HIP kernels with torch reference fallbacks."""

from ._backend import extension, has_extension, require_extension
from .functional import (attention_qkv, conv_block_1d, conv_block_2d,
                         gaussian_kl_sum, gaussian_nll_logmeanexp,
                         gaussian_nll_sum, grid_density, mlp_chain,
                         setconv_gaussian)

__all__ = [
    "attention_qkv",
    "conv_block_1d",
    "conv_block_2d",
    "grid_density",
    "gaussian_kl_sum",
    "mlp_chain",
    "setconv_gaussian",
    "gaussian_nll_sum",
    "gaussian_nll_logmeanexp",
    "extension",
    "has_extension",
    "require_extension",
]
