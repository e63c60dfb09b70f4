"""Model-layer tests: published parameter counts, forward contract, edge cases."""

import warnings

import pytest
import torch

from npf import CNPFLoss, ELBOLossLNPF, NLLLossLNPF

from model_zoo import BUILDERS, PUBLISHED_PARAM_COUNTS

warnings.filterwarnings("ignore")


@pytest.mark.parametrize("name", sorted(PUBLISHED_PARAM_COUNTS))
def test_param_counts_match_reference_published(name):
    """Parameter counts equal the counts printed by the reference notebooks
    (BASELINE.md 'Parameter counts') — a full structural check."""
    model = BUILDERS[name]()
    n = sum(p.numel() for p in model.parameters())
    assert n == PUBLISHED_PARAM_COUNTS[name]


def _set_batch(B=3, C=9, T=17, x_dim=1, y_dim=1):
    g = torch.Generator().manual_seed(0)
    Xc = torch.rand(B, C, x_dim, generator=g) * 2 - 1
    Yc = torch.randn(B, C, y_dim, generator=g)
    Xt = torch.rand(B, T, x_dim, generator=g) * 2 - 1
    Yt = torch.randn(B, T, y_dim, generator=g)
    return Xc, Yc, Xt, Yt


@pytest.mark.parametrize(
    "name,z",
    [("cnp_1d", 1), ("lnp_1d", 1), ("attncnp_1d", 1), ("attnlnp_1d", 1),
     ("convcnp_1d", 1), ("convlnp_1d", 16)],
)
def test_forward_contract_1d(name, z):
    model = BUILDERS[name]()
    model.train()
    Xc, Yc, Xt, Yt = _set_batch()
    p_y, z_samples, q_zCc, q_zCct = model(Xc, Yc, Xt, Yt)
    assert tuple(p_y.batch_shape) == (z, 3, 17)
    assert tuple(p_y.event_shape) == (1,)
    if name in ("cnp_1d", "attncnp_1d", "convcnp_1d"):
        assert z_samples is None and q_zCc is None and q_zCct is None
    else:
        assert z_samples is not None and q_zCc is not None


@pytest.mark.parametrize("name", ["gridconvcnp_2d", "gridconvlnp_2d"])
def test_forward_contract_grid(name):
    model = BUILDERS[name](y_dim=3)
    model.train()
    g = torch.Generator().manual_seed(0)
    B, H, W = 2, 16, 16
    mask = torch.rand(B, H, W, 1, generator=g) < 0.3
    img = torch.rand(B, H, W, 3, generator=g)
    full = torch.ones(B, H, W, 1).bool()
    p_y, *_ = model(mask, img, full, img)
    assert tuple(p_y.batch_shape)[1:] == (B, H, W)
    assert tuple(p_y.event_shape) == (3,)


@pytest.mark.parametrize("name", ["cnp_1d", "attncnp_1d", "convcnp_1d"])
def test_zero_context(name):
    """Zero-context tasks return a valid distribution (reference np.py:97-99,
    attnnp.py:108-125, convnp.py:146-151)."""
    model = BUILDERS[name]()
    model.eval()
    _, _, Xt, _ = _set_batch()
    p_y, *_ = model(torch.zeros(3, 0, 1), torch.zeros(3, 0, 1), Xt)
    assert tuple(p_y.batch_shape) == (1, 3, 17)
    assert torch.isfinite(p_y.base_dist.loc).all()


def test_training_input_range_validated():
    model = BUILDERS["cnp_1d"]()
    model.train()
    Xc, Yc, Xt, Yt = _set_batch()
    with pytest.raises(ValueError):
        model(Xc * 3, Yc, Xt, Yt)
    model.eval()
    model(Xc * 3, Yc, Xt)  # no validation at eval time


def test_backward_all_params_get_grads():
    model = BUILDERS["attnlnp_1d"]()
    model.train()
    Xc, Yc, Xt, Yt = _set_batch()
    out = model(Xc, Yc, Xt, Yt)
    loss = ELBOLossLNPF()(out, Yt)
    loss.backward()
    missing = [n for n, p in model.named_parameters() if p.grad is None]
    assert missing == []


def test_convlnp_nll_loss_backward():
    model = BUILDERS["convlnp_1d"]()
    model.train()
    Xc, Yc, Xt, Yt = _set_batch()
    out = model(Xc, Yc, Xt, Yt)
    loss = NLLLossLNPF()(out, Yt)
    loss.backward()
    assert torch.isfinite(loss)


def test_set_extrapolation_regrids():
    model = BUILDERS["convcnp_1d"]()
    n0 = model.n_induced
    model.set_extrapolation([-1, 3])
    assert model.n_induced != n0
    model.eval()
    Xc, Yc, _, _ = _set_batch()
    Xt = torch.rand(3, 7, 1) * 4 - 1  # beyond [-1, 1]
    p_y, *_ = model(Xc, Yc, Xt)
    assert torch.isfinite(p_y.base_dist.loc).all()


def test_heteroskedastic_flag_pools_scale():
    m = BUILDERS["cnp_1d"]()
    m.is_heteroskedastic = False
    m.eval()
    Xc, Yc, Xt, _ = _set_batch()
    p_y, *_ = m(Xc, Yc, Xt)
    s = p_y.base_dist.scale
    assert torch.allclose(s, s.mean(dim=2, keepdim=True).expand_as(s))


def test_state_dict_roundtrip():
    m1 = BUILDERS["convlnp_1d"]()
    m2 = BUILDERS["convlnp_1d"]()
    m2.load_state_dict(m1.state_dict())
    for (n1, p1), (n2, p2) in zip(m1.named_parameters(), m2.named_parameters()):
        assert n1 == n2
        assert torch.equal(p1, p2)


def test_gridconvcnp_zsmms_circular_translation_equivariance():
    """The CircularPad2d variant is EXACTLY equivariant to circular shifts:
    roll(inputs) -> roll(predictions) (the property zsmms training relies
    on, reference ConvCNP.ipynb 'full translation equivariance')."""
    import sys

    sys.path.insert(0, "tests")
    from model_zoo import gridconvcnp_zsmms

    torch.manual_seed(0)
    m = gridconvcnp_zsmms(y_dim=1).eval()
    g = torch.Generator().manual_seed(1)
    Y = torch.rand(1, 16, 16, 1, generator=g)
    mc = torch.rand(1, 16, 16, 1, generator=g) < 0.3
    mt = torch.ones(1, 16, 16, 1, dtype=torch.bool)
    with torch.no_grad():
        p, *_ = m(mc, Y, mt)
        loc = p.base_dist.loc[0, 0]  # [H, W, 1]
        sh = (5, 3)
        p2, *_ = m(torch.roll(mc, sh, dims=(1, 2)), torch.roll(Y, sh, dims=(1, 2)), mt)
        loc2 = p2.base_dist.loc[0, 0]
    assert torch.allclose(torch.roll(loc, sh, dims=(0, 1)), loc2, atol=1e-5), (
        (torch.roll(loc, sh, dims=(0, 1)) - loc2).abs().max()
    )


def test_convcnp_translation_equivariance_1d():
    """ConvCNP is (approximately, up to grid discretization) translation
    equivariant: shifting context+target x by a grid-aligned offset shifts
    predictions (the ConvCNP paper's core property)."""
    import sys

    sys.path.insert(0, "tests")
    from model_zoo import convcnp_1d

    torch.manual_seed(0)
    m = convcnp_1d().eval()
    g = torch.Generator().manual_seed(2)
    Xc = torch.rand(1, 9, 1, generator=g) * 0.8 - 0.6  # keep room to shift
    Yc = torch.randn(1, 9, 1, generator=g)
    Xt = torch.linspace(-0.6, 0.2, 33).view(1, -1, 1)
    # shift by an exact multiple of the induced-grid spacing
    delta = 16 / (m.density_induced * 3 - 1) * (3.0)  # grid step * 16... use steps
    step = 3.0 / (int(m.density_induced * 3) - 1)
    delta = 24 * step
    with torch.no_grad():
        p1, *_ = m(Xc, Yc, Xt)
        p2, *_ = m(Xc + delta, Yc, Xt + delta)
    a = p1.base_dist.loc[0, 0, :, 0]
    b = p2.base_dist.loc[0, 0, :, 0]
    assert torch.allclose(a, b, atol=1e-3), (a - b).abs().max()


def test_convcnp_with_unet_and_forced_bottleneck():
    """UNet induced-to-induced CNN incl. the batch-half bottleneck averaging
    (reference cnn.py:466-475 is_force_same_bottleneck) runs fwd+bwd."""
    from functools import partial

    from npf import CNP, ConvCNP  # noqa: F401
    from npf.architectures import MLP, ResConvBlock, SetConv, UnetCNN, discard_ith_arg

    m = ConvCNP(
        x_dim=1, y_dim=1, Interpolator=SetConv,
        CNN=partial(
            UnetCNN, ConvBlock=ResConvBlock, Conv=torch.nn.Conv1d,
            Normalization=torch.nn.Identity, n_blocks=3, kernel_size=5,
            is_chan_last=True, n_conv_layers=1, Pool=torch.nn.AvgPool1d,
            upsample_mode="linear", is_force_same_bottleneck=True,
        ),
        density_induced=16, r_dim=32,
        Decoder=discard_ith_arg(partial(MLP, n_hidden_layers=2, hidden_size=32), i=0),
    )
    m.train()
    # duplicated batch: two context draws of the same functions
    Xc = torch.rand(4, 6, 1) * 2 - 1
    Yc = torch.randn(4, 6, 1)
    Xt = torch.rand(4, 12, 1) * 2 - 1
    p, *_ = m(Xc, Yc, Xt)
    p.base_dist.loc.sum().backward()
    assert p.base_dist.loc.shape == (1, 4, 12, 1)


@pytest.mark.parametrize("name", ["cnp_1d", "attncnp_1d", "convcnp_1d"])
def test_context_permutation_invariance(name):
    """NP predictions are invariant to the ORDER of the context set."""
    import sys

    sys.path.insert(0, "tests")
    import model_zoo as zoo

    torch.manual_seed(0)
    m = zoo.BUILDERS[name]().eval()
    g = torch.Generator().manual_seed(3)
    Xc = torch.rand(2, 11, 1, generator=g) * 2 - 1
    Yc = torch.randn(2, 11, 1, generator=g)
    Xt = torch.rand(2, 17, 1, generator=g) * 2 - 1
    perm = torch.randperm(11, generator=g)
    with torch.no_grad():
        p1, *_ = m(Xc, Yc, Xt)
        p2, *_ = m(Xc[:, perm], Yc[:, perm], Xt)
    assert torch.allclose(
        p1.base_dist.loc, p2.base_dist.loc, atol=1e-5
    ), (p1.base_dist.loc - p2.base_dist.loc).abs().max()


@pytest.mark.gpu
def test_forward_bitwise_repeatable_on_gpu():
    """fixed-seed bitwise repeatability.  Eval-mode forwards
    through the HIP kernels are atomics-free (atomics live in backward and
    the training-mode BN stats kernel), so two identical forwards must be
    bitwise equal; and fixed-seed episode generation must reproduce."""
    import sys

    sys.path.insert(0, "tests")
    from model_zoo import attncnp_1d, convcnp_1d

    from npf.train import set_seed

    for builder in (attncnp_1d, convcnp_1d):
        torch.manual_seed(0)
        m = builder().cuda().eval()
        g = torch.Generator().manual_seed(5)
        Xc = (torch.rand(4, 17, 1, generator=g) * 2 - 1).cuda()
        Yc = torch.randn(4, 17, 1, generator=g).cuda()
        Xt = (torch.rand(4, 64, 1, generator=g) * 2 - 1).cuda()
        with torch.no_grad():
            p1, *_ = m(Xc, Yc, Xt)
            p2, *_ = m(Xc, Yc, Xt)
        assert torch.equal(p1.base_dist.loc, p2.base_dist.loc), builder
        assert torch.equal(p1.base_dist.scale, p2.base_dist.scale), builder

    # episode-generation determinism on device (splitter + GP sampler)
    from npf.data import GPDataset
    from npf.data.kernels import RBF
    from npf.utils.datasplit import CntxtTrgtGetter, GetRandomIndcs

    def draw():
        set_seed(7)
        ds = GPDataset(kernel=RBF(0.2), n_samples=8, n_points=32,
                       device="cuda", defer_generation=True)
        X, Y = ds.sample_tasks(8, 32, (-2, 2), out_device="cuda")
        sp = CntxtTrgtGetter(contexts_getter=GetRandomIndcs(a=2, b=6))
        return sp(X, Y)

    a, b = draw(), draw()
    for t1, t2 in zip(a, b):
        assert torch.equal(t1, t2)
