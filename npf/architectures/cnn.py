"""Convolutional blocks, CNN stacks and the U-Net used between induced points.

Parity with /root/reference/npf/architectures/cnn.py (GaussianConv2d :24-53,
ConvBlock :56-123, ResConvBlock :126-215, ResNormalizedConvBlock :218-304,
CNN :307-380, UnetCNN :383-516).

The depthwise+pointwise (separable) blocks on [Z*B, r_dim, n_induced] /
[Z*B, r_dim, H, W] are the ConvNP hot loop (SURVEY.md §2.3 "ResConvBlock
stack"); they run through MIOpen on ROCm, with channel-last permutes kept at
the CNN boundary exactly as the reference does (cnn.py:363-375).
"""

import torch
import torch.nn as nn
from torch.nn import functional as F

from npf.utils.helpers import (
    channels_to_2nd_dim,
    channels_to_last_dim,
    make_depth_sep_conv,
)
from npf.utils.initialization import init_param_, weights_init

__all__ = [
    "GaussianConv2d",
    "ConvBlock",
    "ResNormalizedConvBlock",
    "ResConvBlock",
    "CNN",
    "UnetCNN",
]


class GaussianConv2d(nn.Module):
    """Separable Gaussian blur with learned per-axis widths
    (reference cnn.py:24-53)."""

    def __init__(self, kernel_size=5, **kwargs):
        super().__init__()
        self.kwargs = kwargs
        assert kernel_size % 2 == 1
        self.kernel_sizes = (kernel_size, kernel_size)
        self.exponent = -(
            (torch.arange(0, kernel_size).view(-1, 1).float() - kernel_size // 2) ** 2
        )
        self.reset_parameters()

    def reset_parameters(self):
        self.weights_x = nn.Parameter(torch.tensor([1.0]))
        self.weights_y = nn.Parameter(torch.tensor([1.0]))

    def forward(self, X):
        self.exponent = self.exponent.to(X.device)
        marginal_x = torch.softmax(self.exponent * self.weights_x, dim=0)
        marginal_y = torch.softmax(self.exponent * self.weights_y, dim=0).T
        in_chan = X.size(1)
        filters = (marginal_x @ marginal_y).view(1, 1, *self.kernel_sizes)
        filters = filters.expand(in_chan, 1, *self.kernel_sizes)
        return F.conv2d(X, filters, groups=in_chan, **self.kwargs)


class ConvBlock(nn.Module):
    """norm -> act -> depthwise-separable conv (reference cnn.py:56-123)."""

    def __init__(
        self,
        in_chan,
        out_chan,
        Conv,
        kernel_size=5,
        dilation=1,
        activation=nn.ReLU(),
        Normalization=nn.Identity,
        **kwargs,
    ):
        super().__init__()
        self.activation = activation
        Conv = make_depth_sep_conv(Conv)
        self.conv = Conv(in_chan, out_chan, kernel_size, padding=kernel_size // 2, **kwargs)
        self.norm = Normalization(in_chan)
        self.reset_parameters()

    def reset_parameters(self):
        weights_init(self)

    def forward(self, X):
        return self.conv(self.activation(self.norm(X)))


class ResConvBlock(nn.Module):
    """Pre-activation residual block with depthwise-separable convs
    (reference cnn.py:126-215).

    Attribute names (`conv1.depthwise`, `conv2_depthwise`, `conv2_pointwise`,
    `norm1`, `norm2`) are checkpoint-format-relevant.
    """

    def __init__(
        self,
        in_chan,
        out_chan,
        Conv,
        kernel_size=5,
        activation=nn.ReLU(),
        Normalization=nn.Identity,
        is_bias=True,
        n_conv_layers=1,
    ):
        super().__init__()
        self.activation = activation
        self.n_conv_layers = n_conv_layers
        assert self.n_conv_layers in (1, 2)
        if kernel_size % 2 == 0:
            raise ValueError(f"`kernel_size={kernel_size}`, but should be odd.")
        padding = kernel_size // 2

        if self.n_conv_layers == 2:
            self.norm1 = Normalization(in_chan)
            self.conv1 = make_depth_sep_conv(Conv)(
                in_chan, in_chan, kernel_size, padding=padding, bias=is_bias
            )
        self.norm2 = Normalization(in_chan)
        self.conv2_depthwise = Conv(
            in_chan, in_chan, kernel_size, padding=padding, groups=in_chan, bias=is_bias
        )
        self.conv2_pointwise = Conv(in_chan, out_chan, 1, bias=is_bias)
        self.reset_parameters()

    def reset_parameters(self):
        weights_init(self)

    @staticmethod
    def _is_plain_same_pad(conv, dims):
        """True only for stock zero same-padding convs.  Subclasses that pad
        externally (e.g. make_padded_conv's PaddedConv: native padding=0 plus
        a CircularPad2d `padder`) must NOT take the fused path — the fused
        kernel applies zero same-padding and would silently drop circular
        padding (and with it the zsmms variant's exact shift equivariance)."""
        k = conv.kernel_size
        return (
            not hasattr(conv, "padder")
            and conv.padding_mode == "zeros"
            and tuple(conv.padding) == tuple(ks // 2 for ks in k)
        )

    def _fused_op(self, X):
        from npf.ops import conv_block_1d, conv_block_2d, has_extension

        if not (X.is_cuda and isinstance(self.activation, nn.ReLU)
                and has_extension()):
            return None
        convs = [self.conv2_depthwise]
        if self.n_conv_layers == 2:
            convs.append(self.conv1.depthwise)
        if (X.dim() == 3 and isinstance(self.conv2_depthwise, nn.Conv1d)
                and isinstance(self.norm2, (nn.BatchNorm1d, nn.Identity))
                and all(self._is_plain_same_pad(c, 1) for c in convs)):
            return conv_block_1d
        if (X.dim() == 4 and isinstance(self.conv2_depthwise, nn.Conv2d)
                and isinstance(self.norm2, (nn.BatchNorm2d, nn.Identity))
                and all(self._is_plain_same_pad(c, 2) for c in convs)):
            return conv_block_2d
        return None

    @staticmethod
    def _pointwise_gemm(conv, x):
        """1x1 conv as a strided-batched GEMM: MIOpen falls back to naive
        bf16 NCHW kernels for these shapes (profiled 83ms/call on
        [256,128,64,64]); hipBLASLt does [Cout,Cin] @ [N,Cin,HW] directly."""
        n, cin = x.shape[0], x.shape[1]
        spatial = x.shape[2:]
        w = conv.weight.reshape(conv.out_channels, cin)
        y = torch.matmul(w, x.reshape(n, cin, -1))
        if conv.bias is not None:
            y = y + conv.bias.view(1, -1, 1)
        return y.view(n, conv.out_channels, *spatial)

    def forward(self, X):
        fused = self._fused_op(X)
        if fused is not None:
            # fused HIP path (csrc/npf_hip/convblock{,2d}.hip): bn+relu+dwconv
            # collapse to 2 kernels; the pointwise runs as a batched GEMM
            if self.n_conv_layers == 2:
                h = fused(
                    X,
                    self.conv1.depthwise,
                    bn=self.norm1 if not isinstance(self.norm1, nn.Identity) else None,
                )
                h = self._pointwise_gemm(self.conv1.pointwise, h)
            else:
                h = X
            out = fused(
                h,
                self.conv2_depthwise,
                bn=self.norm2 if not isinstance(self.norm2, nn.Identity) else None,
                residual=X,
            )
            return self._pointwise_gemm(self.conv2_pointwise, out)

        out = self.conv1(self.activation(self.norm1(X))) if self.n_conv_layers == 2 else X
        out = self.conv2_depthwise(self.activation(self.norm2(out)))
        # residual added BEFORE the pointwise so out_chan may differ
        out = out + X
        return self.conv2_pointwise(out.contiguous())


class ResNormalizedConvBlock(ResConvBlock):
    """Residual block with normalized convolutions and confidence channels
    (reference cnn.py:218-304)."""

    def __init__(
        self, in_chan, out_chan, Conv, kernel_size=5, activation=nn.ReLU(),
        is_bias=True, **kwargs,
    ):
        super().__init__(
            in_chan,
            out_chan,
            Conv,
            kernel_size=kernel_size,
            activation=activation,
            is_bias=is_bias,
            Normalization=nn.Identity,
            **kwargs,
        )

    def reset_parameters(self):
        weights_init(self)
        self.bias = nn.Parameter(torch.tensor([0.0]))
        self.temperature = nn.Parameter(torch.tensor([0.0]))
        init_param_(self.temperature)

    def forward(self, X):
        """Normalized convolution: the first half of the channels carries the
        signal, the second half a per-position confidence in [0, 1]."""
        signal, confidence = X.chunk(2, dim=1)
        confidence = confidence.clamp(0.0, 1.0)
        weighted = signal * confidence

        # smear the confidence-weighted signal and the confidence mass with
        # the same conv stack, then divide the mass back out (+ residual)
        smeared = self.conv2_depthwise(
            self.activation(self.conv1(self.activation(weighted)))
        )
        mass = self.conv2_depthwise(self.conv1(confidence))
        normalized = smeared / mass.clamp(min=1e-5) + weighted

        # confidence is monotone non-decreasing and saturates at 1
        gain = torch.sigmoid(mass * F.softplus(self.temperature) + self.bias)
        new_confidence = (confidence + gain).clamp(max=1.0)

        return torch.cat(
            [self.conv2_pointwise(normalized), self.conv2_pointwise(new_confidence)],
            dim=1,
        )


class CNN(nn.Module):
    """Stack of conv blocks with optional channel-last I/O
    (reference cnn.py:307-380)."""

    def __init__(self, n_channels, ConvBlock, n_blocks=3, is_chan_last=False, **kwargs):
        super().__init__()
        self.n_blocks = n_blocks
        self.is_chan_last = is_chan_last
        self.in_out_channels = self._get_in_out_channels(n_channels, n_blocks)
        self.conv_blocks = nn.ModuleList(
            ConvBlock(i, o, **kwargs) for i, o in self.in_out_channels
        )
        self.is_return_rep = False
        self.reset_parameters()

    def reset_parameters(self):
        weights_init(self)

    def _get_in_out_channels(self, n_channels, n_blocks):
        if isinstance(n_channels, int):
            channel_list = [n_channels] * (n_blocks + 1)
        else:
            channel_list = list(n_channels)
        assert len(channel_list) == n_blocks + 1, f"{len(channel_list)} != {n_blocks + 1}"
        return list(zip(channel_list, channel_list[1:]))

    def forward(self, X):
        if self.is_chan_last:
            X = channels_to_2nd_dim(X)
        X, representation = self.apply_convs(X)
        if self.is_chan_last:
            X = channels_to_last_dim(X)
        if self.is_return_rep:
            return X, representation
        return X

    def apply_convs(self, X):
        for conv_block in self.conv_blocks:
            X = conv_block(X)
        return X, None


class UnetCNN(CNN):
    """U-Net over induced points (reference cnn.py:383-516)."""

    def __init__(
        self,
        n_channels,
        ConvBlock,
        Pool,
        upsample_mode,
        max_nchannels=256,
        pooling_size=2,
        is_force_same_bottleneck=False,
        is_return_rep=False,
        **kwargs,
    ):
        self.max_nchannels = max_nchannels
        super().__init__(n_channels, ConvBlock, **kwargs)
        self.pooling_size = pooling_size
        self.pooling = Pool(self.pooling_size)
        self.upsample_mode = upsample_mode
        self.is_force_same_bottleneck = is_force_same_bottleneck
        self.is_return_rep = is_return_rep

    def apply_convs(self, X):
        n_down_blocks = self.n_blocks // 2
        residuals = [None] * n_down_blocks

        for i in range(n_down_blocks):
            X = self.conv_blocks[i](X)
            residuals[i] = X
            X = self.pooling(X)

        X = self.conv_blocks[n_down_blocks](X)
        # global-mean-pool bottleneck summary (reference cnn.py:464)
        representation = X.view(*X.shape[:2], -1).mean(-1)

        if self.is_force_same_bottleneck and self.training:
            # average bottlenecks of the two halves of the batch (the halves
            # hold different context/target draws of the SAME functions)
            batch_size = X.size(0)
            X_mean = (X[: batch_size // 2] + X[batch_size // 2 :]) / 2
            X = torch.cat([X_mean, X_mean], dim=0)

        for i in range(n_down_blocks + 1, self.n_blocks):
            X = F.interpolate(
                X,
                mode=self.upsample_mode,
                scale_factor=self.pooling_size,
                align_corners=True,
            )
            X = torch.cat((X, residuals[n_down_blocks - i]), dim=1)
            X = self.conv_blocks[i](X)

        return X, representation

    def _get_in_out_channels(self, n_channels, n_blocks):
        """Channel-doubling U-Net schedule: widths double per down level and
        mirror back up, interior widths capped at `max_nchannels` (the two
        endpoints are the caller's I/O and stay uncapped), and every
        up-path block's input is doubled by its skip concatenation."""
        assert n_blocks % 2 == 1, f"n_blocks={n_blocks} not odd"
        depth = n_blocks // 2
        down = [n_channels << lvl for lvl in range(depth + 1)]
        widths = down + down[::-1]
        last = len(widths) - 1
        widths = [
            w if i in (0, last) else min(w, self.max_nchannels)
            for i, w in enumerate(widths)
        ]
        return [
            (cin * 2 if idx > depth else cin, cout)
            for idx, (cin, cout) in enumerate(zip(widths[:-1], widths[1:]))
        ]
