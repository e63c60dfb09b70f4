"""Property-based tests (hypothesis) for op-layer and splitter invariants.

These complement the fixed-case parity tests: random shapes/values probe the
reference implementations' contracts on CPU (the same code paths are the
GPU kernels' oracles)."""

import math

import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from npf.ops import functional as F_ops
from npf.utils.datasplit import CntxtTrgtGetter, GetRandomIndcs, get_all_indcs

SET = dict(max_examples=25, deadline=None)


@given(
    b=st.integers(1, 4),
    k=st.integers(1, 20),
    q=st.integers(1, 20),
    d=st.integers(1, 16),
    seed=st.integers(0, 2**16),
)
@settings(**SET)
def test_attention_rows_are_convex_combinations(b, k, q, d, seed):
    """softmax attention output lies in the convex hull of the values:
    min_k v <= out <= max_k v per dim."""
    g = torch.Generator().manual_seed(seed)
    keys = torch.randn(b, k, d, generator=g)
    queries = torch.randn(b, q, d, generator=g)
    values = torch.randn(b, k, d, generator=g)
    out = F_ops.attention_qkv(keys, queries, values, 1 / math.sqrt(d))
    lo = values.min(dim=1, keepdim=True).values
    hi = values.max(dim=1, keepdim=True).values
    assert (out >= lo - 1e-5).all() and (out <= hi + 1e-5).all()


@given(
    b=st.integers(1, 3),
    k=st.integers(1, 12),
    q=st.integers(1, 12),
    c=st.integers(1, 8),
    sigma=st.floats(0.05, 1.0),
    seed=st.integers(0, 2**16),
)
@settings(**SET)
def test_setconv_value_channels_are_convex_combinations(b, k, q, c, sigma, seed):
    """SetConv's value channels are softmax-weighted sums of the inputs; the
    density channel is positive and bounded by K."""
    g = torch.Generator().manual_seed(seed)
    xk = torch.rand(b, k, 1, generator=g) * 2 - 1
    xq = torch.rand(b, q, 1, generator=g) * 2 - 1
    v = torch.randn(b, k, c, generator=g)
    out = F_ops.setconv_gaussian(xk, xq, v, torch.tensor(sigma))
    vals, density = out[..., :c], out[..., c]
    lo = v.min(dim=1, keepdim=True).values
    hi = v.max(dim=1, keepdim=True).values
    assert (vals >= lo - 1e-5).all() and (vals <= hi + 1e-5).all()
    # distant queries at small sigma underflow to exactly 0 (legitimate)
    assert (density >= 0).all() and (density <= k + 1e-5).all()


@given(
    z=st.integers(1, 4),
    b=st.integers(1, 4),
    t=st.integers(1, 16),
    seed=st.integers(0, 2**16),
)
@settings(**SET)
def test_gaussian_nll_matches_torch_distribution(z, b, t, seed):
    g = torch.Generator().manual_seed(seed)
    loc = torch.randn(z, b, t, 1, generator=g)
    scale = torch.rand(z, b, t, 1, generator=g) + 0.05
    y = torch.randn(b, t, 1, generator=g)
    out = F_ops.gaussian_nll_sum(loc, scale, y)
    ref = (
        torch.distributions.Independent(torch.distributions.Normal(loc, scale), 1)
        .log_prob(y)
        .reshape(z, b, -1)
        .sum(-1)
    )
    assert torch.allclose(out, ref, atol=1e-4)


@given(
    mq=st.floats(-3, 3), sq=st.floats(0.1, 2.0),
    mp=st.floats(-3, 3), sp=st.floats(0.1, 2.0),
)
@settings(**SET)
def test_kl_nonnegative_and_zero_iff_equal(mq, sq, mp, sp):
    a = torch.tensor([[[mq]]])
    b_ = torch.tensor([[[sq]]])
    c = torch.tensor([[[mp]]])
    d = torch.tensor([[[sp]]])
    kl = F_ops.gaussian_kl_sum(a, b_, c, d)
    assert kl.item() >= -1e-6
    kl_self = F_ops.gaussian_kl_sum(a, b_, a, b_)
    assert abs(kl_self.item()) < 1e-6


@given(
    batch=st.integers(1, 4),
    n=st.integers(4, 64),
    a=st.integers(0, 10),
    extra=st.integers(0, 10),
    seed=st.integers(0, 2**16),
)
@settings(**SET)
def test_splitter_indices_valid_and_targets_complete(batch, n, a, extra, seed):
    """Context counts land in [a, b]; with get_all_indcs the targets are the
    full set in order."""
    b = min(a + extra, n)
    a = min(a, b)
    torch.manual_seed(seed)
    getter = CntxtTrgtGetter(
        contexts_getter=GetRandomIndcs(a=a, b=b), targets_getter=get_all_indcs
    )
    X = torch.rand(batch, n, 1) * 2 - 1
    Y = torch.randn(batch, n, 1)
    Xc, Yc, Xt, Yt = getter(X, Y)
    assert a <= Xc.shape[1] <= b
    assert torch.equal(Xt, X) and torch.equal(Yt, Y)
    # every context point exists in the full set (row-wise membership)
    for i in range(batch):
        for j in range(Xc.shape[1]):
            assert (X[i] == Xc[i, j]).all(dim=-1).any()


@given(seed=st.integers(0, 2**16), n=st.integers(2, 40))
@settings(**SET)
def test_logcumsumexp_matches_naive(seed, n):
    from npf.utils.helpers import logcumsumexp

    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n, 3, generator=g) * 5
    out = logcumsumexp(x, 0)
    naive = torch.stack(
        [torch.logsumexp(x[: i + 1], dim=0) for i in range(n)]
    )
    assert torch.allclose(out, naive, atol=1e-5)
