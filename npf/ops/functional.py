"""Hot-path ops: fused HIP kernels on MI355X, pure-torch reference elsewhere.

Each op has (a) a composed-torch reference implementation used on CPU and as
the numerics oracle in tests, and (b) a custom autograd.Function dispatching to
the in-tree HIP extension for CUDA(ROCm) tensors.  The op inventory mirrors
SURVEY.md §2.3; reference call-sites cited per op.
"""

import math

import torch

from . import _backend

__all__ = ["attention_qkv", "setconv_gaussian", "gaussian_nll_sum", "gaussian_nll_logmeanexp", "conv_block_1d", "conv_block_2d", "grid_density", "gaussian_kl_sum", "mlp_chain"]


# --------------------------------------------------------------------------- #
# Scaled-dot cross attention (softmax over keys) + weighted value sum.
# Reference computation: attention.py:129-156 (BaseAttender.forward) with
# DotAttender.score (attention.py:204-220) — logits einsum, softmax over keys,
# bmm with values.
# --------------------------------------------------------------------------- #


def _attention_ref(keys, queries, values, scale):
    logits = torch.einsum("bkd,bqd->bqk", keys, queries) * scale
    attn = logits.softmax(dim=-1)
    return torch.bmm(attn, values)


class _AttentionFn(torch.autograd.Function):
    """Flash-style fused attention for the NP regime (small K, large Q)."""

    @staticmethod
    def forward(ctx, keys, queries, values, scale):
        ext = _backend.require_extension("attention_qkv")
        if ext is None:  # NPF_ALLOW_EAGER_GPU escape hatch
            out = _attention_ref(keys, queries, values, scale)
            ctx.save_for_backward(keys, queries, values)
            ctx.scale = scale
            ctx.lse = None
            return out
        out, lse = ext.attn_fwd(queries, keys, values, scale)
        ctx.save_for_backward(keys, queries, values, out, lse)
        ctx.scale = scale
        return out

    @staticmethod
    def backward(ctx, dout):
        scale = ctx.scale
        if len(ctx.saved_tensors) == 3:
            keys, queries, values = ctx.saved_tensors
            logits = torch.einsum("bkd,bqd->bqk", keys, queries) * scale
            attn = logits.softmax(dim=-1)
            out = None
            lse = None
        else:
            keys, queries, values, out, lse = ctx.saved_tensors
            attn = None

        ext = _backend.extension()
        if ext is not None and lse is not None:
            dout = dout.contiguous()
            dq, dk, dv = ext.attn_bwd(queries, keys, values, out, lse, dout, scale)
            return dk, dq, dv, None

        # composed fallback
        dv = torch.bmm(attn.transpose(1, 2), dout)
        dattn = torch.bmm(dout, values.transpose(1, 2))
        dlogits = attn * (dattn - (dattn * attn).sum(-1, keepdim=True))
        dq = torch.bmm(dlogits, keys) * scale
        dk = torch.bmm(dlogits.transpose(1, 2), queries) * scale
        return dk, dq, dv, None


def attention_qkv(keys, queries, values, scale=None):
    """softmax(scale * Q K^T) V, softmax over keys.

    keys: [N, K, D]; queries: [N, Q, D]; values: [N, K, Dv] -> [N, Q, Dv].

    The fused HIP kernel covers the NP regime (per-head D, Dv <= 32 — every
    shipped config uses 16); larger head dims use the composed rocBLAS path.
    """
    if scale is None:
        scale = 1.0 / math.sqrt(queries.size(-1))
    fusable = (
        queries.is_cuda
        and queries.size(-1) <= 32
        and values.size(-1) <= 32
        and keys.size(1) > 0
        and queries.dtype in (torch.float32, torch.bfloat16)
        and keys.dtype == queries.dtype == values.dtype
    )
    if fusable:
        return _AttentionFn.apply(
            keys.contiguous(), queries.contiguous(), values.contiguous(), scale
        )
    return _attention_ref(keys, queries, values, scale)


# --------------------------------------------------------------------------- #
# SetConv with Gaussian (p=2) softmax-normalized RBF + density channel.
# Reference computation: setcnn.py:234-268 (SetConv.forward) with ExpRBF
# (setcnn.py:126-142): pairwise diff -> -(d/sigma)^2 -> softmax over keys for
# the weights and raw exp-sum for the density; output [B, Q, C+1].
# --------------------------------------------------------------------------- #


def _setconv_ref(keys, queries, values, sigma):
    # keys [B,K,d], queries [B,Q,d], values [B,K,C]
    diff = keys.unsqueeze(1) - queries.unsqueeze(2)  # [B,Q,K,d]
    dist2 = (diff * diff).sum(-1)  # [B,Q,K]
    inp = -dist2 / (sigma * sigma)
    w = inp.softmax(dim=-1)
    density = inp.exp().sum(dim=-1, keepdim=True)  # [B,Q,1]
    out = torch.bmm(w, values)  # [B,Q,C]
    return torch.cat([out, density], dim=-1)


class _SetConvFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, keys, queries, values, sigma):
        ext = _backend.require_extension("setconv_gaussian")
        if ext is None:
            with torch.enable_grad():
                pass
            out = _setconv_ref(keys, queries, values, sigma)
            ctx.save_for_backward(keys, queries, values, sigma)
            ctx.fused = False
            return out
        # sigma rides as a device scalar: no host sync (hipGraph-capture
        # safe) and graph replays see the CURRENT learned length-scale
        sig32 = sigma.detach().reshape(1).float().contiguous()
        out = ext.setconv_fwd(keys, queries, values, sig32)
        ctx.save_for_backward(keys, queries, values, sigma, sig32)
        ctx.fused = True
        return out

    @staticmethod
    def backward(ctx, dout):
        if ctx.fused:
            keys, queries, values, sigma, sig32 = ctx.saved_tensors
        else:
            keys, queries, values, sigma = ctx.saved_tensors
        ext = _backend.extension()
        if ctx.fused and ext is not None:
            dout = dout.contiguous()
            dk, dq, dv, dsig = ext.setconv_bwd(keys, queries, values, sig32, dout)
            return dk, dq, dv, dsig.to(sigma.dtype).reshape(sigma.shape)

        # composed fallback (double-backward capable is not required)
        with torch.enable_grad():
            k = keys.detach().requires_grad_(keys.requires_grad)
            q = queries.detach().requires_grad_(queries.requires_grad)
            v = values.detach().requires_grad_(values.requires_grad)
            s = sigma.detach().requires_grad_(True)
            out = _setconv_ref(k, q, v, s)
            grads = torch.autograd.grad(
                out, [t for t in (k, q, v, s) if t.requires_grad], dout
            )
        it = iter(grads)
        return tuple(
            next(it) if t.requires_grad else None for t in (k, q, v, s)
        )


_SETCONV_MAX_K = 4096  # LDS stripe cap of the fused kernel (csrc setconv.hip)


def setconv_gaussian(keys, queries, values, sigma):
    """Fused Gaussian SetConv: returns [B, Q, C+1] (weighted values ++ density).

    `sigma` is the post-softplus length-scale (a 0-d/1-elem tensor so the
    learned parameter receives gradient).  Positions stay fp32 (length scales
    are ~4e-3: bf16 positions would destroy the RBF) and bf16 values are
    upcast before the kernel.
    """
    sigma = sigma.reshape(())
    B, K, _ = keys.shape
    if K == 0:
        # empty context: zero weighted sum AND zero density (matches the
        # reference's empty-softmax semantics, setcnn.py:253-266)
        Q, C = queries.size(1), values.size(2)
        return queries.new_zeros(B, Q, C + 1) * sigma  # keep sigma in graph
    if queries.is_cuda and keys.size(2) == 1 and K <= _SETCONV_MAX_K:
        return _SetConvFn.apply(
            keys.float().contiguous(),
            queries.float().contiguous(),
            values.float().contiguous(),
            sigma.float(),
        )
    return _setconv_ref(keys.float(), queries.float(), values.float(), sigma)


# --------------------------------------------------------------------------- #
# Fused diagonal-Gaussian log-likelihood summed over the target set.
# Reference computation: losses.py:18-24 (sum_log_prob) on
# Independent(Normal, 1) (npf/utils/helpers.py:125-129): elementwise normal
# log-prob then sum over every dim from 2.
# --------------------------------------------------------------------------- #

_HALF_LOG_2PI = 0.5 * math.log(2 * math.pi)


def _nll_ref(loc, scale, y):
    lp = (
        -((y - loc) ** 2) / (2 * scale * scale)
        - torch.log(scale)
        - _HALF_LOG_2PI
    )
    return lp.reshape(*lp.shape[:2], -1).sum(-1)


class _GaussLLFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, loc, scale, y):
        ext = _backend.require_extension("gaussian_nll_sum")
        if ext is None:
            ctx.save_for_backward(loc, scale, y)
            return _nll_ref(loc, scale, y)
        out = ext.gauss_ll_fwd(loc, scale, y)
        ctx.save_for_backward(loc, scale, y)
        return out

    @staticmethod
    def backward(ctx, dout):
        loc, scale, y = ctx.saved_tensors
        ext = _backend.extension()
        dout = dout.contiguous()
        if ext is not None and loc.is_cuda:
            dloc, dscale = ext.gauss_ll_bwd(loc, scale, y, dout)
            return dloc, dscale, None
        d = dout.reshape(*dout.shape, *([1] * (loc.dim() - 2)))
        diff = y - loc
        inv_s2 = 1.0 / (scale * scale)
        dloc = d * diff * inv_s2
        dscale = d * (diff * diff * inv_s2 / scale - 1.0 / scale)
        return dloc, dscale, None


def gaussian_nll_sum(loc, scale, y):
    """Sum over targets of diagonal-Gaussian log-prob: [Z,B,...,Y] -> [Z,B].

    `y` is broadcast against loc over the leading Z dim.
    """
    if y.dim() == loc.dim() - 1:
        y = y.unsqueeze(0)
    y = y.expand_as(loc)
    if loc.dtype not in (torch.float32, torch.float64):
        loc, scale = loc.float(), scale.float()  # bf16/half -> fp32 math
    y = y.to(loc.dtype)
    if loc.is_cuda and loc.dtype == torch.float32:
        return _GaussLLFn.apply(loc.contiguous(), scale.contiguous(), y.contiguous())
    return _nll_ref(loc, scale, y)


class _GaussLLLseFn(torch.autograd.Function):
    """NPML objective fused end-to-end: per-z target-summed log-lik then
    logmeanexp over z, [Z,B,...,Y] -> [B] (SURVEY.md §2.3 "NPML objective";
    reference losses.py:169-203 runs ~6 extra kernels for the logsumexp)."""

    @staticmethod
    def forward(ctx, loc, scale, y):
        ext = _backend.require_extension("gaussian_nll_logmeanexp")
        w = ext.gauss_ll_fwd(loc, scale, y)  # [Z, B]
        out = ext.lse_z_fwd(w)  # [B]
        ctx.save_for_backward(loc, scale, y, w, out)
        return out

    @staticmethod
    def backward(ctx, dout):
        ext = _backend.extension()
        loc, scale, y, w, out = ctx.saved_tensors
        dw = ext.lse_z_bwd(w, out, dout.contiguous())
        dloc, dscale = ext.gauss_ll_bwd(loc, scale, y, dw)
        return dloc, dscale, None


def gaussian_nll_logmeanexp(loc, scale, y):
    """log mean_z exp( sum_targets log N(y; loc, scale) ): [Z,B,..,Y] -> [B].

    The NPML training/eval objective without importance weights
    (q_zCct=None), one fused kernel pair instead of NLL + torch logsumexp.
    """
    if y.dim() == loc.dim() - 1:
        y = y.unsqueeze(0)
    y = y.expand_as(loc)
    if loc.dtype not in (torch.float32, torch.float64):
        loc, scale = loc.float(), scale.float()
    y = y.to(loc.dtype)
    if loc.is_cuda and loc.dtype == torch.float32 and loc.dim() >= 2 \
            and _backend.require_extension("gaussian_nll_logmeanexp") is not None:
        return _GaussLLLseFn.apply(
            loc.contiguous(), scale.contiguous(), y.contiguous()
        )
    w = _nll_ref(loc, scale, y)
    return torch.logsumexp(w, 0) - math.log(w.shape[0])


# --------------------------------------------------------------------------- #
# Fused TransformerAttender pieces (reference attention.py:375-527, :530-588).
# qkv_project_headsplit: the K/Q/V projection Linears as one MFMA kernel that
# writes the head-split [H*B, N, hs] layout directly (no permute kernels).
# add_layernorm: y = LayerNorm(a + b) with an optional head-split gather on
# `a`, so the attention output never needs a head-merge kernel either.
# --------------------------------------------------------------------------- #


class _QkvProjFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, n_heads, Bs, Ns, n_problems, *tensors):
        ext = _backend.require_extension("qkv_project_headsplit")
        xs = list(tensors[:n_problems])
        ws = list(tensors[n_problems : 2 * n_problems])
        bs = list(tensors[2 * n_problems :])
        outs = ext.qkv_fwd(
            xs, ws, [b if b is not None else torch.Tensor() for b in bs],
            list(Bs), list(Ns), n_heads,
        )
        ctx.save_for_backward(*xs, *ws)
        ctx.meta = (n_heads, tuple(Bs), tuple(Ns), n_problems,
                    tuple(b is not None for b in bs))
        return tuple(outs)

    @staticmethod
    def backward(ctx, *douts):
        ext = _backend.extension()
        n_heads, Bs, Ns, n, has_bias = ctx.meta
        xs = ctx.saved_tensors[:n]
        ws = ctx.saved_tensors[n:]
        rets = ext.qkv_bwd(
            [d.contiguous() for d in douts], list(ws), list(Bs), list(Ns),
            [int(h) for h in has_bias], n_heads,
        )
        dxs, dws, dbs = [], [], []
        for i in range(n):
            dx, dz, db = rets[3 * i], rets[3 * i + 1], rets[3 * i + 2]
            dxs.append(dx.view_as(xs[i]))
            # dW = dz^T @ x : K-large GEMM on hipBLASLt, fp32 master grad
            dws.append(torch.mm(dz.t(), xs[i].reshape(-1, xs[i].shape[-1])).float())
            dbs.append(db if has_bias[i] else None)
        return (None, None, None, None, *dxs, *dws, *dbs)


def qkv_project_headsplit(xs, weights, biases, n_heads):
    """[(B, N, D)] inputs -> [(H*B, N, D/H)] projected head-split outputs.

    One MFMA kernel for up to 3 projection problems (K, Q, V); torch Linear
    weight convention ([out, in]); bias may be None per problem.
    """
    Bs = [x.shape[0] for x in xs]
    Ns = [x.shape[1] for x in xs]
    xs = [x.reshape(-1, x.shape[-1]).to(torch.bfloat16).contiguous() for x in xs]
    ws = [w.float() for w in weights]
    bs = [b.float() if b is not None else None for b in biases]
    return _QkvProjFn.apply(n_heads, Bs, Ns, len(xs), *xs, *ws, *bs)


class _AddLNFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b, gamma, beta, B, N, H, eps):
        ext = _backend.require_extension("add_layernorm")
        y, s, mean, rstd = ext.add_ln_fwd(a, b, gamma, beta, B, N, H, eps)
        ctx.save_for_backward(s, gamma, mean, rstd)
        ctx.meta = (B, N, H)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _backend.extension()
        s, gamma, mean, rstd = ctx.saved_tensors
        B, N, H = ctx.meta
        da, db, dgamma, dbeta = ext.add_ln_bwd(
            s, dy.reshape(s.shape).contiguous(), gamma, mean, rstd, B, N, H
        )
        return da, db, dgamma, dbeta, None, None, None, None


def add_layernorm(a, b, gamma, beta, eps=1e-5, headsplit=None):
    """LayerNorm(a + b) over the last dim (<= 128), bf16 in/out, fp32 stats.

    `headsplit=(B, N, H)` reads `a` in the attention head-split layout
    [H*B, N, D/H] (row r of the plain view maps to (b=r//N, n=r%N)); the
    returned grad for `a` keeps that layout.  b: [B*N, D] or [B, N, D].
    """
    B, N, H = headsplit if headsplit is not None else (0, 0, 0)
    lead = b.shape[:-1]
    D = b.shape[-1]
    if H == 0:
        # plain layout: flatten so the returned grad matches a's shape
        a = a.reshape(-1, D)
    y = _AddLNFn.apply(
        a.to(torch.bfloat16).contiguous(),
        b.reshape(-1, D).to(torch.bfloat16).contiguous(),
        gamma.float(), beta.float(), B, N, H, eps,
    )
    return y.reshape(*lead, D)


# --------------------------------------------------------------------------- #
# Fused pre-activation depthwise conv block (1D).
# Reference computation: npf/architectures/cnn.py ResConvBlock.forward
# (reference cnn.py:204-215) — batchnorm -> relu -> depthwise conv1d
# [-> + residual].  The pointwise conv that follows stays a library GEMM.
# --------------------------------------------------------------------------- #


def _bump_num_batches(bn, training):
    """torch BN increments num_batches_tracked per training forward; the
    fused kernels update running stats directly, so mirror the counter here
    (a device-side add: it replays correctly inside captured graphs)."""
    if training and bn is not None and getattr(bn, "num_batches_tracked", None) is not None:
        bn.num_batches_tracked.add_(1)


def _conv_block_ref(x, weight, bias, bn, residual, training):
    a = x
    if bn is not None:
        a = torch.nn.functional.batch_norm(
            a, bn.running_mean, bn.running_var, bn.weight, bn.bias,
            training, bn.momentum, bn.eps,
        )
    a = torch.relu(a)
    out = torch.nn.functional.conv1d(
        a, weight, bias, padding=weight.shape[-1] // 2, groups=x.shape[1]
    )
    if residual is not None:
        out = out + residual
    return out


class _ConvBlock1dFn(torch.autograd.Function):
    """bn+relu+dwconv(+residual) in 2 fwd / 2 bwd kernels (convblock.hip)."""

    @staticmethod
    def forward(ctx, x, weight, bias, gamma, beta, running_mean, running_var,
                eps, momentum, training, residual):
        ext = _backend.require_extension("conv_block_1d")
        C = x.shape[1]
        w2d = weight.view(C, -1).contiguous()
        has_bn = gamma is not None
        und = torch.Tensor()  # undefined-tensor placeholder for the binding
        if has_bn:
            if training:
                mean, rstd, _ = ext.convblock_stats(
                    x, running_mean if running_mean is not None else und,
                    running_var if running_var is not None else und,
                    eps, momentum,
                )
            else:
                mean = running_mean
                rstd = torch.rsqrt(running_var + eps)
        else:
            mean = rstd = und
        y = ext.convblock_fwd(
            x, residual if residual is not None else und, w2d,
            bias if bias is not None else und,
            gamma if has_bn else und, beta if has_bn else und, mean, rstd,
        )
        ctx.save_for_backward(x, w2d, *( (gamma, beta, mean, rstd) if has_bn else () ))
        ctx.has_bn = has_bn
        ctx.has_bias = bias is not None
        ctx.has_res = residual is not None
        ctx.training_mode = training
        ctx.kshape = weight.shape
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _backend.require_extension("conv_block_1d")
        und = torch.Tensor()
        if ctx.has_bn:
            x, w2d, gamma, beta, mean, rstd = ctx.saved_tensors
        else:
            x, w2d = ctx.saved_tensors
            gamma = beta = mean = rstd = und
        dy = dy.contiguous()
        dx, dw, db, dgamma, dbeta = ext.convblock_bwd(
            x, w2d, dy, gamma, beta, mean, rstd,
            ctx.has_bias, ctx.training_mode,
        )
        return (
            dx,
            dw.view(ctx.kshape),
            db if ctx.has_bias else None,
            dgamma if ctx.has_bn else None,
            dbeta if ctx.has_bn else None,
            None, None, None, None, None,
            dy if ctx.has_res else None,
        )


def _cb1d_lds_ok(x, k):
    """The 1D backward stages 2*CB_TN(=4) padded rows of fp32 in dynamic LDS
    (convblock.hip npf_cb_bwd_launch): stay under the 64 KB limit, else fall
    back to the composed path (only reachable for L over ~2030)."""
    L = x.shape[-1]
    return 2 * 4 * (L + k - 1) * 4 <= 64 * 1024


def conv_block_1d(x, conv, bn=None, residual=None):
    """Fused norm->relu->depthwise-conv1d(+residual) on [N, C, L].

    `conv` is the depthwise nn.Conv1d (groups == C), `bn` an optional
    nn.BatchNorm1d (its running stats are updated in training mode exactly
    like torch), `residual` an optional tensor added to the output.
    """
    if (
        not x.is_cuda
        or not _cb1d_lds_ok(x, conv.weight.shape[-1])
        or _backend.require_extension("conv_block_1d") is None
    ):
        return _conv_block_ref(
            x, conv.weight, conv.bias, bn,
            residual, bn.training if bn is not None else conv.training,
        )
    training = bn.training if bn is not None else conv.training
    _bump_num_batches(bn, training)
    # fp32 compute: BN statistics and the stencil are precision-sensitive;
    # at these sizes the op is dispatch/HBM-bound so bf16 buys nothing
    xf = x.float().contiguous()
    res = residual.float().contiguous() if residual is not None else None
    y = _ConvBlock1dFn.apply(
        xf, conv.weight.float(), 
        conv.bias.float() if conv.bias is not None else None,
        bn.weight.float() if bn is not None else None,
        bn.bias.float() if bn is not None else None,
        bn.running_mean if bn is not None else None,
        bn.running_var if bn is not None else None,
        bn.eps if bn is not None else 1e-5,
        bn.momentum if bn is not None else 0.1,
        training, res,
    )
    return y


# --------------------------------------------------------------------------- #
# Fused pre-activation depthwise conv block (2D) — GridConv models.
# --------------------------------------------------------------------------- #


def _conv_block2d_ref(x, weight, bias, bn, residual, training):
    a = x
    if bn is not None:
        a = torch.nn.functional.batch_norm(
            a, bn.running_mean, bn.running_var, bn.weight, bn.bias,
            training, bn.momentum, bn.eps,
        )
    a = torch.relu(a)
    out = torch.nn.functional.conv2d(
        a, weight, bias, padding=weight.shape[-1] // 2, groups=x.shape[1]
    )
    if residual is not None:
        out = out + residual
    return out


class _ConvBlock2dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, gamma, beta, running_mean, running_var,
                eps, momentum, training, residual):
        ext = _backend.require_extension("conv_block_2d")
        C = x.shape[1]
        K = weight.shape[-1]
        w2d = weight.view(C, K, K).contiguous()
        has_bn = gamma is not None
        und = torch.Tensor()
        if has_bn:
            if training:
                mean, rstd, _ = ext.convblock_stats(
                    x.view(x.shape[0], C, -1),
                    running_mean if running_mean is not None else und,
                    running_var if running_var is not None else und,
                    eps, momentum,
                )
            else:
                mean = running_mean
                rstd = torch.rsqrt(running_var + eps)
        else:
            mean = rstd = und
        y = ext.convblock2d_fwd(
            x, residual if residual is not None else und, w2d,
            bias if bias is not None else und,
            gamma if has_bn else und, beta if has_bn else und, mean, rstd,
        )
        ctx.save_for_backward(x, w2d, *( (gamma, beta, mean, rstd) if has_bn else () ))
        ctx.has_bn = has_bn
        ctx.has_bias = bias is not None
        ctx.has_res = residual is not None
        ctx.training_mode = training
        ctx.kshape = weight.shape
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _backend.require_extension("conv_block_2d")
        und = torch.Tensor()
        if ctx.has_bn:
            x, w2d, gamma, beta, mean, rstd = ctx.saved_tensors
        else:
            x, w2d = ctx.saved_tensors
            gamma = beta = mean = rstd = und
        dy = dy.contiguous()
        dx, dw, db, dgamma, dbeta = ext.convblock2d_bwd(
            x, w2d, dy, gamma, beta, mean, rstd,
            ctx.has_bias, ctx.training_mode,
        )
        return (
            dx,
            dw.view(ctx.kshape),
            db if ctx.has_bias else None,
            dgamma if ctx.has_bn else None,
            dbeta if ctx.has_bn else None,
            None, None, None, None, None,
            dy if ctx.has_res else None,
        )


def _cb2d_lds_ok(x, k):
    """The 2D kernels stage 8-row tiles (+halo) in LDS; stay under the 64 KB
    default dynamic-LDS limit, else fall back to the composed path (only
    reachable for images wider than ~780 px)."""
    W = x.shape[-1]
    pad = k // 2
    return 2 * (8 + 2 * pad) * (W + 2 * pad) * 4 <= 64 * 1024


def conv_block_2d(x, conv, bn=None, residual=None):
    """Fused norm->relu->depthwise-conv2d(+residual) on [N, C, H, W].

    I/O stays in x's dtype (bf16 under autocast: HALVES the HBM traffic of
    the [Z*B, C, H, W] tensors and avoids 0.5 GB cast kernels); BN stats and
    the stencil accumulate in fp32 inside the kernels either way."""
    training = bn.training if bn is not None else conv.training
    k = conv.weight.shape[-1]
    if (
        not x.is_cuda
        or not _cb2d_lds_ok(x, k)
        or _backend.require_extension("conv_block_2d") is None
    ):
        return _conv_block2d_ref(
            x, conv.weight, conv.bias, bn, residual, training
        )
    _bump_num_batches(bn, training)
    xf = x.contiguous()
    res = residual.to(x.dtype).contiguous() if residual is not None else None
    return _ConvBlock2dFn.apply(
        xf, conv.weight.float(),
        conv.bias.float() if conv.bias is not None else None,
        bn.weight.float() if bn is not None else None,
        bn.bias.float() if bn is not None else None,
        bn.running_mean if bn is not None else None,
        bn.running_var if bn is not None else None,
        bn.eps if bn is not None else 1e-5,
        bn.momentum if bn is not None else 0.1,
        training, res,
    )


# --------------------------------------------------------------------------- #
# Grid density encoder (GridConvCNP/LNP cntxt_to_induced).
# Reference computation: gridconvnp.py:136-162 — abs-weight depthwise conv on
# the masked image and on the mask, divide (clamp 1e-5), concat density.
# --------------------------------------------------------------------------- #


def _grid_density_ref(x, mask, weight):
    w = weight.abs()
    C = x.shape[1]
    pad = weight.shape[-1] // 2
    signal = torch.nn.functional.conv2d(x * mask, w, padding=pad, groups=C)
    density = torch.nn.functional.conv2d(
        mask.expand_as(x), w, padding=pad, groups=C
    )
    out = signal / torch.clamp(density, min=1e-5)
    return torch.cat([out, density], dim=1)


class _GridDensityFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, mask, weight):
        ext = _backend.require_extension("grid_density")
        C = x.shape[1]
        K = weight.shape[-1]
        w2d = weight.view(C, K, K).contiguous()
        out = ext.griddensity_fwd(x, mask, w2d)
        ctx.save_for_backward(x, mask, w2d, out)
        ctx.kshape = weight.shape
        return out

    @staticmethod
    def backward(ctx, dout):
        ext = _backend.require_extension("grid_density")
        x, mask, w2d, out = ctx.saved_tensors
        dx, dw = ext.griddensity_bwd(x, mask, dout.contiguous(), out, w2d)
        return dx, None, dw.view(ctx.kshape)


def _gde_lds_ok(x, k):
    """griddensity.hip stages 2 padded 8-row tiles + the KxK filter in
    dynamic LDS; stay under the 64 KB limit, else compose (only reachable
    for images wider than ~780 px)."""
    W = x.shape[-1]
    pad = k // 2
    return (2 * (8 + 2 * pad) * (W + 2 * pad) + k * k) * 4 <= 64 * 1024


def grid_density(x, mask, weight):
    """Fused density encoder: x [B,C,H,W], mask [B,C or 1,H,W] (no grad),
    abs-conv weight [C,1,K,K] -> [B, 2C, H, W] (normalized signal ; density).
    """
    if (
        not x.is_cuda
        or not _gde_lds_ok(x, weight.shape[-1])
        or _backend.require_extension("grid_density") is None
    ):
        return _grid_density_ref(x, mask, weight)
    mask = mask.float().expand_as(x).contiguous()
    return _GridDensityFn.apply(
        x.float().contiguous(), mask, weight.float()
    )


# --------------------------------------------------------------------------- #
# Fused diagonal-Gaussian KL + reduce (NPVI ELBO term).
# Reference computation: losses.py:135-150 (kl_divergence(Normal, Normal)
# composed with sum over latent dims).
# --------------------------------------------------------------------------- #


def _kl_ref(mq, sq, mp, sp):
    kl = (
        (sp / sq).log().neg().neg()  # log(sp) - log(sq) spelled for clarity
        + (sq ** 2 + (mq - mp) ** 2) / (2 * sp ** 2)
        - 0.5
    )
    return kl.reshape(kl.shape[0], -1).sum(-1)


class _GaussKLFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, mq, sq, mp, sp):
        ext = _backend.require_extension("gaussian_kl_sum")
        ctx.save_for_backward(mq, sq, mp, sp)
        return ext.gauss_kl_fwd(mq, sq, mp, sp)

    @staticmethod
    def backward(ctx, dout):
        ext = _backend.require_extension("gaussian_kl_sum")
        mq, sq, mp, sp = ctx.saved_tensors
        dmq, dsq, dmp, dsp = ext.gauss_kl_bwd(mq, sq, mp, sp, dout.contiguous())
        return dmq, dsq, dmp, dsp


def gaussian_kl_sum(mq, sq, mp, sp):
    """KL( N(mq, sq) || N(mp, sp) ) summed over all dims but the first."""
    if not mq.is_cuda or _backend.require_extension("gaussian_kl_sum") is None:
        return _kl_ref(mq, sq, mp, sp)
    return _GaussKLFn.apply(
        mq.float().contiguous(), sq.float().contiguous(),
        mp.float().contiguous(), sp.float().contiguous(),
    )


# --------------------------------------------------------------------------- #
# Fused MFMA MLP chain (csrc/npf_hip/mlp_chain.hip).
# Reference computation: npf/architectures/mlp.py MLP.forward — a chain of
# torch Linears with ReLU between them (reference mlp.py:95-109).
# --------------------------------------------------------------------------- #


class _MLPChainFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, n_layers, *wb):
        ext = _backend.require_extension("mlp_chain")
        ws = list(wb[:n_layers])
        bs = list(wb[n_layers:])
        out = ext.mlp_chain_fwd(x, ws, bs)
        y, acts = out[0], out[1:]
        ctx.save_for_backward(x, *ws, *acts)
        ctx.n_layers = n_layers
        ctx.need_dx = x.requires_grad
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _backend.require_extension("mlp_chain")
        L = ctx.n_layers
        x = ctx.saved_tensors[0]
        ws = list(ctx.saved_tensors[1 : 1 + L])
        acts = list(ctx.saved_tensors[1 + L :])
        out = ext.mlp_chain_bwd(dy.to(torch.bfloat16).contiguous(), ws, acts,
                                ctx.need_dx)
        dx = out[0] if ctx.need_dx else None
        dzs = out[1 : 1 + L]
        dbs = out[1 + L :]
        # dW_l = dz_l^T @ a_{l-1}: K-large library GEMMs, fp32 master grads
        grads = [dx, None]
        ins = [x] + acts
        dws = [
            torch.mm(dzs[l].t(), ins[l]).float() for l in range(L)
        ]
        return tuple(grads + dws + list(dbs))


def mlp_chain(x, weights, biases):
    """Fused MLP chain: y = W_L(relu(... relu(W_1 x + b_1) ...)) + b_L.

    x [..., d0] (any float dtype; computed in bf16 with fp32 accumulation =
    autocast semantics), weights fp32 [out, in] per layer, out bf16.
    """
    lead = x.shape[:-1]
    d_out = weights[-1].shape[0]
    if x.numel() == 0:
        # zero-context episodes: no rows, no launch — but keep the output
        # connected to every parameter so .grad is zero-filled (not absent):
        # a rank whose whole batch is empty must still produce gradients for
        # the DDP flat all-reduce to stay aligned across ranks
        zero = x.reshape(-1).sum()
        for t in list(weights) + list(biases):
            zero = zero + t.sum() * 0.0
        out = x.new_zeros(*lead, d_out, dtype=torch.bfloat16)
        return out + zero.to(torch.bfloat16) * 0.0
    xb = x.reshape(-1, x.shape[-1]).to(torch.bfloat16).contiguous()
    y = _MLPChainFn.apply(xb, len(weights), *weights, *biases)
    return y.reshape(*lead, y.shape[-1])
