"""Torch-native GP kernels (batched, device-capable).

The reference generates meta-tasks with sklearn's GaussianProcessRegressor on
CPU (reference utils/data/gaussian_process.py:100-104, 202-231) — a serial
Cholesky per sample group that becomes the training bottleneck at MI355X
speeds.  Here kernels evaluate batched covariance
matrices in torch, so the whole epoch's tasks are drawn with one batched
Cholesky (CPU or GPU).  Parametrization mirrors sklearn's
(RBF/ExpSineSquared/Matern/WhiteKernel and +/* composition), and
`from_sklearn` converts sklearn kernel objects so reference-style configs
work unchanged.
"""

import math

import numpy as np
import torch

__all__ = [
    "Kernel",
    "RBF",
    "ExpSineSquared",
    "Matern",
    "WhiteKernel",
    "SumKernel",
    "ProductKernel",
    "from_sklearn",
]


class Kernel:
    """Base covariance function.

    `__call__(X)` takes X [..., N, d] and returns [..., N, N].
    `sample_hyperparameters_()` draws uniformly inside the bounds in log-space
    is NOT what sklearn's reference loop does — it samples uniformly in the
    raw bounds (reference gaussian_process.py:233-243) — so we do the same.
    """

    def __call__(self, X):
        raise NotImplementedError

    def hyperparameter_bounds(self):
        """dict name -> (lo, hi) for the varying hyperparameters."""
        return {}

    def sample_hyperparameters_(self, rng=None):
        rng = rng or np.random
        for name, (lo, hi) in self.hyperparameter_bounds().items():
            setattr(self, name, float(rng.uniform(lo, hi)))

    def __add__(self, other):
        return SumKernel(self, other)

    def __mul__(self, other):
        return ProductKernel(self, other)


def _sqdist(X):
    # X [..., N, d] -> squared euclidean distances [..., N, N]
    d = X.unsqueeze(-2) - X.unsqueeze(-3)
    return (d * d).sum(-1)


class RBF(Kernel):
    def __init__(self, length_scale=1.0, length_scale_bounds=None):
        self.length_scale = float(length_scale)
        self.length_scale_bounds = length_scale_bounds

    def __call__(self, X):
        return torch.exp(-0.5 * _sqdist(X) / (self.length_scale**2))

    def hyperparameter_bounds(self):
        if self.length_scale_bounds is None:
            return {}
        return {"length_scale": self.length_scale_bounds}


class ExpSineSquared(Kernel):
    def __init__(self, length_scale=1.0, periodicity=1.0, length_scale_bounds=None,
                 periodicity_bounds=None):
        self.length_scale = float(length_scale)
        self.periodicity = float(periodicity)
        self.length_scale_bounds = length_scale_bounds
        self.periodicity_bounds = periodicity_bounds

    def __call__(self, X):
        dist = torch.sqrt(_sqdist(X).clamp_min(1e-30))
        s = torch.sin(math.pi * dist / self.periodicity) / self.length_scale
        return torch.exp(-2.0 * s * s)

    def hyperparameter_bounds(self):
        out = {}
        if self.length_scale_bounds is not None:
            out["length_scale"] = self.length_scale_bounds
        if self.periodicity_bounds is not None:
            out["periodicity"] = self.periodicity_bounds
        return out


class Matern(Kernel):
    """Matern kernel; nu in {0.5, 1.5, 2.5} (the closed forms)."""

    def __init__(self, length_scale=1.0, nu=1.5, length_scale_bounds=None):
        assert nu in (0.5, 1.5, 2.5), "only closed-form nu supported"
        self.length_scale = float(length_scale)
        self.nu = nu
        self.length_scale_bounds = length_scale_bounds

    def __call__(self, X):
        dist = torch.sqrt(_sqdist(X).clamp_min(1e-30)) / self.length_scale
        if self.nu == 0.5:
            return torch.exp(-dist)
        if self.nu == 1.5:
            a = math.sqrt(3) * dist
            return (1.0 + a) * torch.exp(-a)
        a = math.sqrt(5) * dist
        return (1.0 + a + a * a / 3.0) * torch.exp(-a)

    def hyperparameter_bounds(self):
        if self.length_scale_bounds is None:
            return {}
        return {"length_scale": self.length_scale_bounds}


class WhiteKernel(Kernel):
    def __init__(self, noise_level=1.0, noise_level_bounds=None):
        self.noise_level = float(noise_level)
        self.noise_level_bounds = noise_level_bounds

    def __call__(self, X):
        N = X.shape[-2]
        eye = torch.eye(N, dtype=X.dtype, device=X.device)
        return self.noise_level * eye.expand(*X.shape[:-2], N, N)

    def hyperparameter_bounds(self):
        if self.noise_level_bounds is None:
            return {}
        return {"noise_level": self.noise_level_bounds}


class _Composite(Kernel):
    def __init__(self, k1, k2):
        self.k1 = k1
        self.k2 = k2

    def sample_hyperparameters_(self, rng=None):
        self.k1.sample_hyperparameters_(rng)
        self.k2.sample_hyperparameters_(rng)


class SumKernel(_Composite):
    def __call__(self, X):
        return self.k1(X) + self.k2(X)


class ProductKernel(_Composite):
    def __call__(self, X):
        return self.k1(X) * self.k2(X)


def to_sklearn(kernel):
    """Convert a torch-native Kernel back to the sklearn equivalent (used for
    the oracle-GP overlay in 1D posterior plots)."""
    import sklearn.gaussian_process.kernels as SK

    if isinstance(kernel, SumKernel):
        return to_sklearn(kernel.k1) + to_sklearn(kernel.k2)
    if isinstance(kernel, ProductKernel):
        return to_sklearn(kernel.k1) * to_sklearn(kernel.k2)
    if isinstance(kernel, Matern):
        return SK.Matern(length_scale=float(kernel.length_scale), nu=kernel.nu)
    if isinstance(kernel, RBF):
        return SK.RBF(length_scale=float(kernel.length_scale))
    if isinstance(kernel, ExpSineSquared):
        return SK.ExpSineSquared(
            length_scale=float(kernel.length_scale),
            periodicity=float(kernel.periodicity),
        )
    if isinstance(kernel, WhiteKernel):
        return SK.WhiteKernel(noise_level=float(kernel.noise_level))
    raise ValueError(f"cannot convert {kernel!r} to sklearn")


def _bounds_of(sk, name):
    b = getattr(sk, f"{name}_bounds", "fixed")
    if isinstance(b, str):  # "fixed"
        return None
    b = np.asarray(b).squeeze()
    return (float(b[0]), float(b[1]))


def from_sklearn(sk_kernel):
    """Convert an sklearn.gaussian_process.kernels object (or return torch
    kernels unchanged) so reference-style dataset configs work verbatim."""
    if isinstance(sk_kernel, Kernel):
        return sk_kernel
    import sklearn.gaussian_process.kernels as SK

    if isinstance(sk_kernel, SK.Sum):
        return SumKernel(from_sklearn(sk_kernel.k1), from_sklearn(sk_kernel.k2))
    if isinstance(sk_kernel, SK.Product):
        return ProductKernel(from_sklearn(sk_kernel.k1), from_sklearn(sk_kernel.k2))
    # NOTE: sklearn's Matern subclasses RBF — check it first
    if isinstance(sk_kernel, SK.Matern):
        return Matern(np.squeeze(sk_kernel.length_scale), sk_kernel.nu,
                      _bounds_of(sk_kernel, "length_scale"))
    if isinstance(sk_kernel, SK.RBF):
        return RBF(np.squeeze(sk_kernel.length_scale),
                   _bounds_of(sk_kernel, "length_scale"))
    if isinstance(sk_kernel, SK.ExpSineSquared):
        return ExpSineSquared(
            sk_kernel.length_scale, sk_kernel.periodicity,
            _bounds_of(sk_kernel, "length_scale"),
            _bounds_of(sk_kernel, "periodicity"),
        )
    if isinstance(sk_kernel, SK.WhiteKernel):
        return WhiteKernel(sk_kernel.noise_level, _bounds_of(sk_kernel, "noise_level"))
    if isinstance(sk_kernel, SK.ConstantKernel):
        c = float(sk_kernel.constant_value)

        class _Const(Kernel):
            def __call__(self, X):
                N = X.shape[-2]
                return torch.full((*X.shape[:-2], N, N), c, dtype=X.dtype,
                                  device=X.device)

        return _Const()
    raise ValueError(f"Cannot convert sklearn kernel {sk_kernel!r}")
