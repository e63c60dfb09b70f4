"""Graph-safe latent sampling (the SURVEY §2.3 "Latent sampling" row).

hipGraph capture freezes the Philox offsets that torch's RNG kernels read,
so a captured `rsample` replays the same (and, observed on ROCm, corrupted)
noise on every replay — round 1 therefore ran every latent model eager,
paying full launch overhead (AttnLNP-2D 9.7 ms/step).

The MI355X answer: draw from a STATIC NOISE POOL owned outside the graph.
Inside the capture, sampling is the pure mul-add `loc + scale * eps` over a
fixed-address buffer; between replays the host refreshes the pool in-place
(`normal_()`), which the replayed kernels observe because graphs capture
addresses, not values.  Fresh noise per step => identical training
statistics to eager `rsample`.

Usage (see bench.py): `enable_noise_pool()` before warmup/capture, then
`refresh_noise_()` once per iteration outside the graph.
"""

import torch

__all__ = ["enable_noise_pool", "is_noise_pool_enabled", "pool_noise", "refresh_noise_"]

_POOLS = {}
_ENABLED = False


def enable_noise_pool(flag=True):
    """Globally switch latent sampling to the static-pool path."""
    global _ENABLED
    _ENABLED = flag


def is_noise_pool_enabled():
    return _ENABLED


def pool_noise(shape, device, dtype=torch.float32):
    """Return the pool buffer for (shape, device, dtype), creating (and
    filling) it on first use.  The same tensor object is returned for every
    later call with the same key — a requirement for graph capture, which
    bakes the buffer's address into the replay."""
    key = (tuple(shape), str(device), dtype)
    buf = _POOLS.get(key)
    if buf is None:
        buf = torch.randn(shape, device=device, dtype=dtype)
        _POOLS[key] = buf
    return buf


def refresh_noise_():
    """Refill every pool buffer in-place with fresh standard normals.

    Must be called OUTSIDE any graph capture/replay (it uses torch RNG);
    call once per training step before replaying the graph."""
    for buf in _POOLS.values():
        buf.normal_()


def clear_noise_pool():
    _POOLS.clear()
