"""Weight initialization, matching the reference's *effective* semantics.

Reference: /root/reference/npf/utils/initialization.py:7-124.

An important subtlety of the reference `weights_init` (initialization.py:15-22):
it sets ``module.is_resetted = True`` *before* iterating over submodules, and the
loop's guard checks ``module`` (the outer module) rather than the iterated child.
Any caller that defines ``reset_parameters`` (every caller in the codebase does)
therefore short-circuits the whole loop, so only the *explicit* inits in each
class's ``reset_parameters`` ever run (e.g. ``MLP`` calls ``linear_init``
directly; ``MultiheadAttender`` re-inits its k/q/v transforms).  Everything else
keeps PyTorch defaults.  The shipped pretrained baselines were produced with
that behavior, so we reproduce the effective semantics rather than the bug's
surface form.
"""

import torch
from torch import nn

__all__ = ["weights_init", "linear_init", "init_param_"]

_ACTIVATION_NAMES = {
    nn.LeakyReLU: "leaky_relu",
    nn.ReLU: "relu",
    nn.Tanh: "tanh",
    nn.Sigmoid: "sigmoid",
    nn.Softmax: "sigmoid",
}


def _activation_name(activation):
    if isinstance(activation, str):
        return activation
    for cls, name in _ACTIVATION_NAMES.items():
        if isinstance(activation, cls):
            return name
    raise ValueError(f"Unknown activation type: {activation}")


def get_gain(activation):
    """Gain factor for an activation (reference initialization.py:53-64)."""
    if activation is None:
        return 1
    name = _activation_name(activation)
    param = None
    if name == "leaky_relu" and not isinstance(activation, str):
        param = activation.negative_slope
    return nn.init.calculate_gain(name, param)


def weights_init(module, **kwargs):
    """Mark a module initialized; init children only for reset-less modules.

    See module docstring: for every module that defines ``reset_parameters``
    (all in-tree callers) this is a marker-only operation, matching the
    reference's effective behavior (initialization.py:7-31).
    """
    module.is_resetted = True
    if hasattr(module, "reset_parameters"):
        return
    for m in module.modules():
        if isinstance(m, torch.nn.modules.conv._ConvNd):
            nn.init.kaiming_normal_(m.weight, mode="fan_out", **kwargs)
        elif isinstance(m, nn.Linear):
            linear_init(m, **kwargs)
        elif isinstance(m, nn.BatchNorm2d):
            m.weight.data.fill_(1)
            m.bias.data.zero_()


def linear_init(module, activation="relu"):
    """Initialize a linear layer for a given following activation.

    Reference: initialization.py:67-94 (kaiming-uniform for relu family,
    xavier for sigmoid/tanh, xavier-uniform when no activation).
    """
    if module.bias is not None:
        module.bias.data.zero_()

    w = module.weight
    if activation is None:
        return nn.init.xavier_uniform_(w)

    name = _activation_name(activation)
    if name == "leaky_relu":
        a = 0 if isinstance(activation, str) else activation.negative_slope
        return nn.init.kaiming_uniform_(w, a=a, nonlinearity="leaky_relu")
    if name == "relu":
        return nn.init.kaiming_uniform_(w, nonlinearity="relu")
    if name in ("sigmoid", "tanh"):
        return nn.init.xavier_uniform_(w, gain=get_gain(activation))


def init_param_(param, activation=None, is_positive=False, bound=0.05, shift=0):
    """Uniformly initialize a bare ``nn.Parameter`` (reference :97-124)."""
    gain = get_gain(activation)
    if is_positive:
        nn.init.uniform_(param, 1e-5 + shift, bound * gain + shift)
    else:
        nn.init.uniform_(param, -bound * gain + shift, bound * gain + shift)
