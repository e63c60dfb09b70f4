"""Tensor-shape helpers for folding latent samples into the batch dim.

Parity with /root/reference/npf/neuralproc/helpers.py:4-32.  These are the
zero-copy view discipline around the [n_z_samples * batch, ...] collapse that
every post-latent CNN/SetConv runs on.
"""

from npf.utils.helpers import prod

__all__ = [
    "collapse_z_samples_batch",
    "extract_z_samples_batch",
    "replicate_z_samples",
    "pool_and_replicate_middle",
]


def collapse_z_samples_batch(t):
    """[Z, B, ...] -> [Z*B, ...]."""
    n_z_samples, batch_size, *rest = t.shape
    return t.contiguous().view(n_z_samples * batch_size, *rest)


def extract_z_samples_batch(t, n_z_samples, batch_size):
    """Inverse of `collapse_z_samples_batch`."""
    _, *rest = t.shape
    return t.view(n_z_samples, batch_size, *rest)


def replicate_z_samples(t, n_z_samples):
    """Broadcast-replicate on a new leading dim (no copy)."""
    return t.unsqueeze(0).expand(n_z_samples, *t.shape)


def pool_and_replicate_middle(t):
    """Mean-pool all middle dims and broadcast back (no copy on the expand)."""
    first, *middle, last = t.shape
    pooled = t.reshape(first, prod(middle), last).mean(1, keepdim=True)
    return pooled.view(first, *([1] * len(middle)), last).expand(first, *middle, last)
