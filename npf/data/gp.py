"""GP meta-task dataset.

Capability parity with /root/reference/utils/data/gaussian_process.py:20-243
(GPDataset: fresh-function epochs, n_same_samples X/hyperparameter sharing,
chunked on-disk cache, get_samples/set_samples_, [-1,1] feature rescale).

MI355X-first: tasks are drawn with ONE batched torch Cholesky over
[n_groups, n_points, n_points] covariance instead of sklearn's per-group
serial solve — removing the CPU data-generation bottleneck that would starve
8 GPUs.  The cache uses .npz (h5py is not in this
image); the (file, group) save_file API is preserved.
"""

import logging
import os

import numpy as np
import torch
from torch.utils.data import Dataset

from npf.utils.helpers import rescale_range

from .kernels import Kernel, RBF, WhiteKernel, from_sklearn

__all__ = ["GPDataset"]


class NotLoadedError(Exception):
    pass


def _chunk_path(save_file, idx_chunk):
    if save_file is None:
        return None
    if isinstance(save_file, (tuple, list)):
        file, group = save_file
    else:
        file, group = save_file, "default"
    root, _ = os.path.splitext(file)
    return f"{root}__{group}__chunk{idx_chunk}.npz"


def load_chunk(keys, save_file, idx_chunk):
    """Load a cached chunk; raise NotLoadedError if absent/disabled."""
    path = _chunk_path(save_file, idx_chunk)
    if path is None or idx_chunk is None or not os.path.exists(path):
        raise NotLoadedError()
    with np.load(path) as z:
        return {k: torch.from_numpy(z[k]) for k in keys}


def save_chunk(to_save, save_file, idx_chunk, logger=None):
    path = _chunk_path(save_file, idx_chunk)
    if path is None or idx_chunk is None:
        return
    os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
    if logger is not None:
        logger.info(f"Saving chunk {idx_chunk} to {path}")
    np.savez(path, **{k: v.numpy() for k, v in to_save.items()})


class GPDataset(Dataset):
    """Meta-dataset of functions drawn from a GP prior.

    Parameters follow the reference (gaussian_process.py:20-77): `kernel`
    (torch-native or sklearn object), `min_max` evaluation range, `n_samples`,
    `n_points`, `is_vary_kernel_hyp` (uniform draw inside the kernel's
    `*_bounds`), `save_file` ((file, group) chunk cache),
    `n_same_samples` (functions sharing X and hyperparameters — here also the
    batched-Cholesky group), `is_reuse_across_epochs`.

    Extra (MI355X): `device` — where to run the batched sampling.
    """

    def __init__(
        self,
        kernel=(
            WhiteKernel(noise_level=0.1, noise_level_bounds=(0.1, 0.5))
            + RBF(length_scale=0.4, length_scale_bounds=(0.1, 1.0))
        ),
        min_max=(-2, 2),
        n_samples=1000,
        n_points=128,
        is_vary_kernel_hyp=False,
        save_file=None,
        logging_level=logging.INFO,
        n_same_samples=20,
        is_reuse_across_epochs=True,
        device="cpu",
        generator=None,
        defer_generation=False,
        **kwargs,
    ):
        self.n_samples = n_samples
        self.n_points = n_points
        self.min_max = min_max
        self.is_vary_kernel_hyp = is_vary_kernel_hyp
        self.logger = logging.getLogger("GPDataset")
        self.logger.setLevel(logging_level)
        self.save_file = save_file
        self.n_same_samples = n_same_samples
        self.is_reuse_across_epochs = is_reuse_across_epochs
        self.device = device

        self.kernel = from_sklearn(kernel) if not isinstance(kernel, Kernel) else kernel

        self._idx_precompute = 0
        self._idx_chunk = 0
        self._generator = generator

        self.data = self.targets = None
        if not defer_generation:
            # eager generation serves the DataLoader path; the
            # device-resident loader regenerates on-GPU and never reads it
            self.precompute_chunk_()

    @property
    def generator(self):
        """sklearn GaussianProcessRegressor with this dataset's kernel — the
        oracle model for posterior-overlay plots (reference
        gaussian_process.py exposes the sampling GPR the same way)."""
        if self._generator is None:
            from sklearn.gaussian_process import GaussianProcessRegressor

            from .kernels import to_sklearn

            self._generator = GaussianProcessRegressor(
                kernel=to_sklearn(self.kernel), alpha=1e-5, optimizer=None
            )
        return self._generator

    # ------------------------------------------------------------------ #
    # Dataset protocol
    # ------------------------------------------------------------------ #

    def __len__(self):
        return self.n_samples

    def __getitem__(self, index):
        if self.data is None:
            self.precompute_chunk_()
        if self.is_reuse_across_epochs:
            return self.data[index], self.targets[index]
        # fresh functions: serve sequentially, regenerate when exhausted
        self._idx_precompute += 1
        if self._idx_precompute == self.n_samples:
            self.precompute_chunk_()
        return self.data[self._idx_precompute], self.targets[self._idx_precompute]

    # ------------------------------------------------------------------ #
    # sampling
    # ------------------------------------------------------------------ #

    def get_samples(
        self, n_samples=None, test_min_max=None, n_points=None, save_file=None,
        idx_chunk=None,
    ):
        """Draw (or load cached) `n_samples` functions; returns (X, Y) with
        X in [-1,1] (rescaled from `min_max`)."""
        test_min_max = test_min_max if test_min_max is not None else self.min_max
        n_points = n_points if n_points is not None else self.n_points
        n_samples = n_samples if n_samples is not None else self.n_samples

        try:
            loaded = load_chunk({"data", "targets"}, save_file, idx_chunk)
            if loaded["data"].shape[0] == n_samples and loaded["data"].shape[1] == n_points:
                return loaded["data"], loaded["targets"]
            # stale cache from a different-sized run: regenerate (a smaller
            # chunk silently served as a full epoch corrupts the budget)
            self.logger.warning(
                "cached chunk %s has %s tasks, need %s: regenerating",
                idx_chunk, tuple(loaded["data"].shape), (n_samples, n_points),
            )
        except NotLoadedError:
            pass

        data, targets = self.sample_tasks(n_samples, n_points, test_min_max)
        save_chunk(
            {"data": data, "targets": targets}, save_file, idx_chunk,
            logger=self.logger,
        )
        return data, targets

    def sample_tasks(self, n_samples, n_points, min_max, out_device="cpu"):
        """Batched GP prior draws.

        Groups of `n_same_samples` functions share X and kernel
        hyperparameters (reference gaussian_process.py:202-231); all groups
        are factorized with one batched Cholesky.  `out_device` lets the
        device-resident loader keep the epoch in HBM (no host round-trip).
        """
        device = self.device
        n_groups = (n_samples + self.n_same_samples - 1) // self.n_same_samples

        # float64 factorization at the reference's jitter scale (sklearn
        # GaussianProcessRegressor alpha=1e-10): RBF ls=0.2 over 128 points
        # is too ill-conditioned for fp32, and a bigger jitter would raise
        # the data's noise floor above what the published LLs assume
        X = torch.empty(n_groups, n_points, 1, device=device, dtype=torch.float64)
        X.uniform_(min_max[0], min_max[1])
        X, _ = X.sort(dim=1)

        if self.is_vary_kernel_hyp:
            covs = []
            for g in range(n_groups):
                self.kernel.sample_hyperparameters_()
                covs.append(self.kernel(X[g]))
            cov = torch.stack(covs)
        else:
            cov = self.kernel(X)

        L = _robust_cholesky(cov.double(), jitter=1e-10)
        eps = torch.randn(
            n_groups, n_points, self.n_same_samples, device=device,
            dtype=torch.float64,
        )
        # [G, N, S] -> per-group S functions
        Y = torch.bmm(L, eps)

        X = X.expand(n_groups, n_points, 1).unsqueeze(1).expand(
            n_groups, self.n_same_samples, n_points, 1
        )
        Y = Y.permute(0, 2, 1).unsqueeze(-1)  # [G, S, N, 1]

        X = X.reshape(-1, n_points, 1)[:n_samples].to(out_device)
        Y = Y.reshape(-1, n_points, 1)[:n_samples].contiguous().to(out_device)

        # shuffle so same-group functions are not consecutive
        perm = torch.randperm(n_samples, device=X.device)
        X, Y = X[perm], Y[perm]

        X = rescale_range(X, self.min_max, (-1, 1)).float()
        return X, Y.float()

    def set_samples_(self, data, targets):
        """Freeze the dataset on a fixed (data, targets) draw."""
        self.is_reuse_across_epochs = True
        self.data = data
        self.targets = targets
        self.n_samples = self.data.size(0)

    def precompute_chunk_(self):
        """(Re)generate or load one epoch worth of tasks."""
        self._idx_precompute = 0
        self.data, self.targets = self.get_samples(
            save_file=self.save_file, idx_chunk=self._idx_chunk
        )
        self._idx_chunk += 1


def _robust_cholesky(cov, jitter=1e-6, max_tries=5):
    """Cholesky with escalating diagonal jitter (replaces the reference's
    resample-on-LinAlgError retry loop, gaussian_process.py:209-225)."""
    eye = torch.eye(cov.size(-1), dtype=cov.dtype, device=cov.device)
    j = jitter
    for _ in range(max_tries):
        try:
            L, info = torch.linalg.cholesky_ex(cov + j * eye)
            if int(info.max()) == 0:
                return L
        except RuntimeError:
            pass
        j *= 10
    raise np.linalg.LinAlgError(
        f"Cholesky failed after {max_tries} jitter escalations (jitter={j})."
    )
