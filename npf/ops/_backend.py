"""Loader for the in-tree HIP extension (gfx950).

The extension is built IN-TREE (see csrc/build.py / __graft_entry__.build) so the
resulting `.so` ships with the repo snapshot; nothing is JIT-compiled at import
time.  Policy:

- CPU tensors always use the pure-torch reference path (tests run without a GPU).
- CUDA (ROCm) tensors REQUIRE the HIP extension: if it is missing we raise
  instead of silently falling back to eager torch, so a GPU run can never
  "pass" on a non-native path.  Set ``NPF_ALLOW_EAGER_GPU=1`` to bypass (debug
  only).
"""

import importlib
import os

_EXT = None
_EXT_ERR = None


def _try_load():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        import torch  # noqa: F401  (extension links against torch)

        _EXT = importlib.import_module("npf._hip_C")
    except Exception as e:  # pragma: no cover - exercised only on GPU boxes
        _EXT_ERR = e
        _EXT = None
    return _EXT


def extension():
    """Return the loaded HIP extension module or None."""
    return _try_load()


def has_extension():
    return _try_load() is not None


def require_extension(op_name):
    """Return the extension, raising loudly if a GPU op has no native kernel."""
    if os.environ.get("NPF_FORCE_EAGER") == "1":  # debug: composed-torch path
        return None
    if op_name in os.environ.get("NPF_DISABLE_OPS", "").split(","):
        return None  # selective composed fallback (kernel bisection)
    ext = _try_load()
    if ext is None and os.environ.get("NPF_ALLOW_EAGER_GPU") != "1":
        raise RuntimeError(
            f"npf op '{op_name}' was called on a GPU tensor but the HIP "
            f"extension (npf._hip_C, gfx950) is not built/loadable: {_EXT_ERR!r}. "
            "Build it with `python csrc/build.py` (or __graft_entry__.build). "
            "Set NPF_ALLOW_EAGER_GPU=1 to force the eager torch path (debug only)."
        )
    return ext
