"""General MLP.

API/semantics parity with /root/reference/npf/architectures/mlp.py:12-115,
including the hidden-size clamping rules (mlp.py:64-79) which are part of the
parameter-count / checkpoint contract.  On MI355X the linear layers run as
hipBLASLt bf16 GEMMs under autocast; the (Linear -> activation) chains are
fusion candidates for the HIP epilogue kernels (SURVEY.md §2.3 "MLP decoder").
"""

import os
import warnings

import torch
import torch.nn as nn

from npf.utils.initialization import linear_init

__all__ = ["MLP"]


class MLP(nn.Module):
    """Multi-layer perceptron with optional residual connections and dropout.

    Parameters mirror the reference (mlp.py:12-55); attribute names
    (`to_hidden`, `linears`, `out`) are part of the checkpoint key space.
    """

    def __init__(
        self,
        input_size,
        output_size,
        hidden_size=32,
        n_hidden_layers=1,
        activation=nn.ReLU(),
        is_bias=True,
        dropout=0,
        is_force_hid_smaller=False,
        is_res=False,
    ):
        super().__init__()
        self.input_size = input_size
        self.output_size = output_size
        self.hidden_size = hidden_size
        self.n_hidden_layers = n_hidden_layers
        self.is_res = is_res

        # hidden-size clamping (reference mlp.py:64-79)
        hi, lo = max(output_size, input_size), min(output_size, input_size)
        if is_force_hid_smaller and self.hidden_size > hi:
            self.hidden_size = hi
            warnings.warn(
                f"hidden_size={hidden_size} larger than output={output_size} "
                f"and input={input_size}. Setting it to {self.hidden_size}."
            )
        elif self.hidden_size < lo:
            self.hidden_size = lo
            warnings.warn(
                f"hidden_size={hidden_size} smaller than output={output_size} "
                f"and input={input_size}. Setting it to {self.hidden_size}."
            )

        self.dropout = nn.Dropout(p=dropout) if dropout > 0 else nn.Identity()
        self.activation = activation

        self.to_hidden = nn.Linear(self.input_size, self.hidden_size, bias=is_bias)
        self.linears = nn.ModuleList(
            nn.Linear(self.hidden_size, self.hidden_size, bias=is_bias)
            for _ in range(self.n_hidden_layers - 1)
        )
        self.out = nn.Linear(self.hidden_size, self.output_size, bias=is_bias)

        self.reset_parameters()

    def _fused_ok(self, x):
        from npf.ops import has_extension

        return (
            x.is_cuda
            # bf16 inputs arrive from other fused ops (e.g. the fused
            # attender) outside autocast: same numerics, same kernel
            and (torch.is_autocast_enabled("cuda") or x.dtype == torch.bfloat16)
            and isinstance(self.activation, nn.ReLU)
            and isinstance(self.dropout, nn.Identity)
            and not self.is_res
            and self.input_size <= 128
            and self.hidden_size <= 128
            and self.output_size <= 128
            and self.to_hidden.bias is not None
            and self.to_hidden.weight.dtype == torch.float32
            and has_extension()
            and os.environ.get("NPF_FORCE_EAGER") != "1"
        )

    def forward(self, x):
        if self._fused_ok(x):
            # whole chain in ONE MFMA kernel (csrc/npf_hip/mlp_chain.hip):
            # bf16 compute + fp32 accumulate = the autocast numerics this
            # branch replaces, minus ~3 kernels per layer
            from npf.ops import mlp_chain

            layers = [self.to_hidden, *self.linears, self.out]
            return mlp_chain(
                x, [l.weight for l in layers], [l.bias for l in layers]
            )

        # first layer: linear -> act -> dropout (reference mlp.py:95-98)
        h = self.dropout(self.activation(self.to_hidden(x)))
        for linear in self.linears:
            out = self.activation(linear(h))
            if self.is_res:
                out = out + h
            h = self.dropout(out)
        return self.out(h)

    def reset_parameters(self):
        linear_init(self.to_hidden, activation=self.activation)
        for lin in self.linears:
            linear_init(lin, activation=self.activation)
        linear_init(self.out)
