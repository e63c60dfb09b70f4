"""Golden parity vs the reference's shipped pretrained checkpoints.

These tests are the strongest structural checks we have: they load the
reference repo's `results/pretrained/.../params.pt` (data files, read-only)
into OUR modules and verify key-space identity; a subprocess runs the
REFERENCE implementation itself on fixed inputs and we compare outputs.
Skipped when /root/reference is absent.
"""

import os
import subprocess
import sys
import warnings

import pytest
import torch

from model_zoo import BUILDERS

warnings.filterwarnings("ignore")

REF = "/root/reference"
PRETRAINED = os.path.join(REF, "results", "pretrained", "RBF_Kernel")

needs_ref = pytest.mark.skipif(
    not os.path.isdir(PRETRAINED), reason="reference pretrained results unavailable"
)

REF_MODEL_DIRS = {
    "cnp_1d": "CNP",
    "lnp_1d": "LNP",
    "attncnp_1d": "AttnCNP",
    "attnlnp_1d": "AttnLNP",
    "convcnp_1d": "ConvCNP",
    "convlnp_1d": "ConvLNP",
}


@needs_ref
@pytest.mark.parametrize("name", sorted(REF_MODEL_DIRS))
def test_checkpoint_key_space_identical(name):
    sd_ref = torch.load(
        os.path.join(PRETRAINED, REF_MODEL_DIRS[name], "run_0", "params.pt"),
        map_location="cpu",
    )
    model = BUILDERS[name]()
    sd_mine = model.state_dict()
    assert set(sd_ref) == set(sd_mine)
    for k in sd_ref:
        assert sd_ref[k].shape == sd_mine[k].shape, k
    model.load_state_dict(sd_ref)  # must not raise


_REF_RUNNER = r"""
import sys, warnings, torch
warnings.filterwarnings("ignore")
sys.path.insert(0, "/root/reference")
from functools import partial
from npf import CNP, AttnCNP, ConvCNP
from npf.architectures import MLP, merge_flat_input, CNN, ResConvBlock, SetConv, discard_ith_arg

inp = torch.load(sys.argv[1], weights_only=False)
Xc, Yc, Xt = inp["Xc"], inp["Yc"], inp["Xt"]
R = 128
models = {
 "cnp_1d": CNP(x_dim=1, y_dim=1,
   XYEncoder=merge_flat_input(partial(MLP, n_hidden_layers=2, hidden_size=R*2), is_sum_merge=True),
   XEncoder=partial(MLP, n_hidden_layers=1, hidden_size=R),
   Decoder=merge_flat_input(partial(MLP, n_hidden_layers=4, hidden_size=R), is_sum_merge=True), r_dim=R),
 "attncnp_1d": AttnCNP(x_dim=1, y_dim=1,
   XYEncoder=merge_flat_input(partial(MLP, n_hidden_layers=2, hidden_size=R), is_sum_merge=True),
   is_self_attn=False, r_dim=R, attention="transformer",
   XEncoder=partial(MLP, n_hidden_layers=1, hidden_size=R),
   Decoder=merge_flat_input(partial(MLP, n_hidden_layers=4, hidden_size=R), is_sum_merge=True)),
 "convcnp_1d": ConvCNP(x_dim=1, y_dim=1, Interpolator=SetConv,
   CNN=partial(CNN, Conv=torch.nn.Conv1d, Normalization=torch.nn.BatchNorm1d, n_blocks=5,
               kernel_size=19, ConvBlock=ResConvBlock, is_chan_last=True, n_conv_layers=2),
   density_induced=64, r_dim=R,
   Decoder=discard_ith_arg(partial(MLP, n_hidden_layers=4, hidden_size=R), i=0)),
}
ckpt = {"cnp_1d": "CNP", "attncnp_1d": "AttnCNP", "convcnp_1d": "ConvCNP"}
out = {}
for name, m in models.items():
    sd = torch.load(f"/root/reference/results/pretrained/RBF_Kernel/{ckpt[name]}/run_0/params.pt",
                    map_location="cpu")
    m.load_state_dict(sd); m.eval()
    with torch.no_grad():
        p, *_ = m(Xc, Yc, Xt)
    out[name] = (p.base_dist.loc, p.base_dist.scale)
torch.save(out, sys.argv[2])
"""


@needs_ref
def test_forward_matches_reference_implementation(tmp_path):
    """Same pretrained weights + same inputs through the reference code and
    ours: CNP/AttnCNP bitwise, ConvCNP to fp32 roundoff."""
    g = torch.Generator().manual_seed(42)
    Xc = torch.rand(3, 11, 1, generator=g) * 2 - 1
    Yc = torch.randn(3, 11, 1, generator=g)
    Xt = torch.rand(3, 37, 1, generator=g) * 2 - 1
    inp_path = str(tmp_path / "inp.pt")
    out_path = str(tmp_path / "out.pt")
    torch.save({"Xc": Xc, "Yc": Yc, "Xt": Xt}, inp_path)

    env = dict(os.environ, PYTHONPATH=REF)
    subprocess.run(
        [sys.executable, "-c", _REF_RUNNER, inp_path, out_path],
        check=True, cwd="/tmp", env=env, capture_output=True,
    )
    golden = torch.load(out_path, weights_only=False)

    tol = {"cnp_1d": 0.0, "attncnp_1d": 0.0, "convcnp_1d": 1e-4}
    for name, builder in (
        ("cnp_1d", BUILDERS["cnp_1d"]),
        ("attncnp_1d", BUILDERS["attncnp_1d"]),
        ("convcnp_1d", BUILDERS["convcnp_1d"]),
    ):
        m = builder()
        sd = torch.load(
            os.path.join(PRETRAINED, REF_MODEL_DIRS[name], "run_0", "params.pt"),
            map_location="cpu",
        )
        m.load_state_dict(sd)
        m.eval()
        with torch.no_grad():
            p, *_ = m(Xc, Yc, Xt)
        rl, rs = golden[name]
        assert (p.base_dist.loc - rl).abs().max() <= tol[name]
        assert (p.base_dist.scale - rs).abs().max() <= tol[name]


@needs_ref
def test_eval_loglike_matches_published_mean():
    """Re-evaluating a shipped checkpoint on tasks drawn from the matching GP
    reproduces the published mean test log-likelihood (AttnCNP
    on RBF = 149.16).

    Counts are STRATIFIED (cycled 0..50, each exactly twice over 102 batches)
    so our estimator of the uniform-count expectation is tight: three
    different GP draws land within +-0.5 nats of each other (measured:
    159.7/160.4/160.5).  The remaining band is the published number's OWN
    count-draw noise — the reference evaluated 10k tasks in 157 batches of
    64 sharing one count draw each, so 149.16 = E[LL] + eps with
    sd(eps) ~ sd(E[LL|count])/sqrt(157) ~ 11 nats.  3 sigma => +-33.
    """
    import numpy as np

    from npf import CNPFLoss
    from npf.data import GPDataset, cntxt_trgt_collate
    from npf.data.kernels import RBF
    from npf.train import NPFTrainer, eval_loglike
    from npf.utils.datasplit import (
        CntxtTrgtGetter,
        StratifiedCountIndcs,
        get_all_indcs,
    )

    model = BUILDERS["attncnp_1d"]()
    sd = torch.load(
        os.path.join(PRETRAINED, "AttnCNP", "run_0", "params.pt"), map_location="cpu"
    )
    model.load_state_dict(sd)

    ds = GPDataset(kernel=RBF(length_scale=0.2), n_samples=1632, n_points=128)
    splitter = CntxtTrgtGetter(
        contexts_getter=StratifiedCountIndcs(a=0, b=50), targets_getter=get_all_indcs
    )
    collate = cntxt_trgt_collate(splitter)
    trainer = NPFTrainer(
        model, CNPFLoss(), collate_fn=collate, device="cpu", batch_size=16,
        valid_batch_size=16,
    )
    splitter.contexts_getter.reset()
    ll = eval_loglike(trainer, ds, seed=123)
    mean = float(np.mean(ll))
    assert 116 < mean < 182, mean


_REF_EVAL_RUNNER = r"""
import sys, warnings, torch
warnings.filterwarnings("ignore")
sys.path.insert(0, "/root/reference")
from functools import partial
from npf import AttnCNP
from npf.architectures import MLP, merge_flat_input

episodes = torch.load(sys.argv[1], weights_only=False)
R = 128
m = AttnCNP(x_dim=1, y_dim=1,
  XYEncoder=merge_flat_input(partial(MLP, n_hidden_layers=2, hidden_size=R), is_sum_merge=True),
  is_self_attn=False, r_dim=R, attention="transformer",
  XEncoder=partial(MLP, n_hidden_layers=1, hidden_size=R),
  Decoder=merge_flat_input(partial(MLP, n_hidden_layers=4, hidden_size=R), is_sum_merge=True))
sd = torch.load("/root/reference/results/pretrained/RBF_Kernel/AttnCNP/run_0/params.pt",
                map_location="cpu")
m.load_state_dict(sd); m.eval()
lls = []
with torch.no_grad():
    for ep in episodes:
        p, *_ = m(ep["X_cntxt"], ep["Y_cntxt"], ep["X_trgt"])
        # per-task test LL: log p(Y_trgt) summed over target points
        lls.append(p.log_prob(ep["Y_trgt"]).sum(-1).squeeze(0))
torch.save(torch.cat(lls), sys.argv[2])
"""


@needs_ref
def test_eval_loglike_matches_reference_implementation(tmp_path):
    """The WHOLE eval path (splitter -> collate -> forward -> unreduced loss
    -> row assembly) agrees per-task with the reference implementation run
    on the identical episodes — no count-noise barn door, <0.01 nat tight."""
    import numpy as np

    from npf import CNPFLoss
    from npf.data import GPDataset, cntxt_trgt_collate
    from npf.data.kernels import RBF
    from npf.train import NPFTrainer, eval_loglike
    from npf.utils.datasplit import (
        CntxtTrgtGetter,
        StratifiedCountIndcs,
        get_all_indcs,
    )

    model = BUILDERS["attncnp_1d"]()
    sd = torch.load(
        os.path.join(PRETRAINED, "AttnCNP", "run_0", "params.pt"), map_location="cpu"
    )
    model.load_state_dict(sd)

    ds = GPDataset(kernel=RBF(length_scale=0.2), n_samples=64, n_points=128)
    splitter = CntxtTrgtGetter(
        contexts_getter=StratifiedCountIndcs(a=1, b=50), targets_getter=get_all_indcs
    )
    recorded = []
    base_collate = cntxt_trgt_collate(splitter)

    def recording_collate(batch):
        inputs, y = base_collate(batch)
        recorded.append({k: v.clone() for k, v in inputs.items()})
        return inputs, y

    trainer = NPFTrainer(
        model, CNPFLoss(), collate_fn=recording_collate, device="cpu",
        batch_size=8, valid_batch_size=8,
    )
    ll = eval_loglike(trainer, ds, seed=123)

    inp, outp = str(tmp_path / "eps.pt"), str(tmp_path / "ll.pt")
    torch.save(recorded, inp)
    env = dict(os.environ, PYTHONPATH=REF)
    subprocess.run(
        [sys.executable, "-c", _REF_EVAL_RUNNER, inp, outp],
        check=True, cwd="/tmp", env=env, capture_output=True,
    )
    ll_ref = torch.load(outp, weights_only=False).numpy()
    assert ll.shape == ll_ref.shape
    assert np.abs(ll - ll_ref).max() < 1e-2, np.abs(ll - ll_ref).max()


PRETRAINED_2D = os.path.join(REF, "results", "pretrained", "celeba32")

REF_MODEL_DIRS_2D = {
    "cnp_2d": "CNP",
    "lnp_2d": "LNP",
    "attncnp_2d": "AttnCNP",
    "attnlnp_2d": "AttnLNP",
    "gridconvcnp_2d": "ConvCNP",
    "gridconvlnp_2d": "ConvLNP",
}


@needs_ref
@pytest.mark.parametrize("name", sorted(REF_MODEL_DIRS_2D))
def test_checkpoint_key_space_identical_2d(name):
    sd_ref = torch.load(
        os.path.join(PRETRAINED_2D, REF_MODEL_DIRS_2D[name], "run_0", "params.pt"),
        map_location="cpu",
    )
    model = BUILDERS[name](y_dim=3)
    sd_mine = model.state_dict()
    assert set(sd_ref) == set(sd_mine)
    for k in sd_ref:
        assert sd_ref[k].shape == sd_mine[k].shape, k
    model.load_state_dict(sd_ref)


_REF_RUNNER_2D = r"""
import sys, warnings, torch
warnings.filterwarnings("ignore")
sys.path.insert(0, "/root/reference")
from functools import partial
import torch.nn as nn
from npf import GridConvCNP
from npf.architectures import CNN, MLP, ResConvBlock, discard_ith_arg

inp_path, out_path = sys.argv[1], sys.argv[2]
d = torch.load(inp_path, weights_only=False)

model = GridConvCNP(
    x_dim=1, y_dim=3,
    CNN=partial(CNN, ConvBlock=ResConvBlock, Conv=nn.Conv2d,
                Normalization=nn.BatchNorm2d, n_blocks=5, kernel_size=9,
                is_chan_last=True, n_conv_layers=2),
    r_dim=128,
    Decoder=discard_ith_arg(partial(MLP, n_hidden_layers=4, hidden_size=128), i=0),
)
sd = torch.load(
    "/root/reference/results/pretrained/celeba32/ConvCNP/run_0/params.pt",
    map_location="cpu",
)
model.load_state_dict(sd)
model.eval()
with torch.no_grad():
    p, *_ = model(d["mc"], d["Y"], d["mt"])
torch.save((p.base_dist.loc, p.base_dist.scale), out_path)
"""


@needs_ref
def test_gridconvcnp_forward_matches_reference(tmp_path):
    """Shipped celeba32 GridConvCNP weights + identical mask batch through
    the reference implementation and ours."""
    g = torch.Generator().manual_seed(7)
    B, H, W = 2, 32, 32
    Y = torch.rand(B, H, W, 3, generator=g)
    mc = torch.rand(B, H, W, 1, generator=g) < 0.3
    mt = torch.ones(B, H, W, 1, dtype=torch.bool)
    inp, outp = str(tmp_path / "i.pt"), str(tmp_path / "o.pt")
    torch.save({"mc": mc, "Y": Y, "mt": mt}, inp)

    env = dict(os.environ, PYTHONPATH=REF)
    subprocess.run(
        [sys.executable, "-c", _REF_RUNNER_2D, inp, outp],
        check=True, cwd="/tmp", env=env, capture_output=True,
    )
    rl, rs = torch.load(outp, weights_only=False)

    m = BUILDERS["gridconvcnp_2d"](y_dim=3)
    sd = torch.load(
        os.path.join(PRETRAINED_2D, "ConvCNP", "run_0", "params.pt"),
        map_location="cpu",
    )
    m.load_state_dict(sd)
    m.eval()
    with torch.no_grad():
        p, *_ = m(mc, Y, mt)
    assert (p.base_dist.loc - rl).abs().max() <= 1e-5
    assert (p.base_dist.scale - rs).abs().max() <= 1e-5


@needs_ref
def test_zsmms_circularpad_checkpoint_parity():
    """Fully-translation-equivariant zsmms variant (CircularPad2d in every
    conv) matches the shipped zsmms/ConvCNP checkpoint key-for-key."""
    sd = torch.load(
        os.path.join(REF, "results", "pretrained", "zsmms", "ConvCNP",
                     "run_0", "params.pt"),
        map_location="cpu",
    )
    import model_zoo as zoo

    m = zoo.gridconvcnp_zsmms(y_dim=1)
    assert set(sd) == set(m.state_dict())
    m.load_state_dict(sd)
    # forward on a 56x56 zsmms-shaped batch
    g = torch.Generator().manual_seed(0)
    Y = torch.rand(2, 56, 56, 1, generator=g)
    mc = torch.rand(2, 56, 56, 1, generator=g) < 0.2
    mt = torch.ones(2, 56, 56, 1, dtype=torch.bool)
    m.eval()
    with torch.no_grad():
        p, *_ = m(mc, Y, mt)
    assert torch.isfinite(p.base_dist.loc).all()


def test_gridconvcnp_xl_param_count():
    import model_zoo as zoo

    m = zoo.gridconvcnp_xl(y_dim=3)
    assert sum(p.numel() for p in m.parameters()) == 722417


_REF_CNN_RUNNER = r"""
import sys, warnings, torch
warnings.filterwarnings("ignore")
sys.path.insert(0, "/root/reference")
from npf.architectures import CNN, UnetCNN, ResConvBlock
from npf.architectures.cnn import ResNormalizedConvBlock
import torch.nn as nn

inp, outp = sys.argv[1], sys.argv[2]
d = torch.load(inp, weights_only=False)

# channel schedules for several Unet configs
schedules = {}
for n_ch, n_blocks, max_ch in [(16, 5, 256), (32, 7, 64), (8, 9, 32)]:
    u = UnetCNN(n_ch, Conv=nn.Conv1d, ConvBlock=ResConvBlock, n_blocks=n_blocks,
                kernel_size=5, max_nchannels=max_ch, Pool=nn.MaxPool1d,
                upsample_mode="linear")
    schedules[(n_ch, n_blocks, max_ch)] = u.in_out_channels

# normalized-conv forward on fixed input/weights
m = ResNormalizedConvBlock(8, 8, nn.Conv1d, kernel_size=5, n_conv_layers=2)
m.load_state_dict(d["sd"])
m.eval()
with torch.no_grad():
    y = m(d["x"])
torch.save({"schedules": schedules, "y": y}, outp)
"""


@needs_ref
def test_unet_schedule_and_normalized_block_match_reference(tmp_path):
    """The rewritten UnetCNN channel schedule and ResNormalizedConvBlock
    forward agree with the reference implementation exactly."""
    import torch.nn as nn

    from npf.architectures import CNN, ResConvBlock, UnetCNN
    from npf.architectures.cnn import ResNormalizedConvBlock

    torch.manual_seed(0)
    ours_m = ResNormalizedConvBlock(8, 8, nn.Conv1d, kernel_size=5, n_conv_layers=2)
    x = torch.rand(3, 16, 32)  # [B, 2*chan (signal;confidence), L]
    inp, outp = str(tmp_path / "i.pt"), str(tmp_path / "o.pt")
    torch.save({"sd": ours_m.state_dict(), "x": x}, inp)

    env = dict(os.environ, PYTHONPATH=REF)
    subprocess.run(
        [sys.executable, "-c", _REF_CNN_RUNNER, inp, outp],
        check=True, cwd="/tmp", env=env, capture_output=True,
    )
    ref = torch.load(outp, weights_only=False)

    for (n_ch, n_blocks, max_ch), ref_sched in ref["schedules"].items():
        u = UnetCNN(
            n_ch, Conv=nn.Conv1d, ConvBlock=ResConvBlock, n_blocks=n_blocks,
            kernel_size=5, max_nchannels=max_ch, Pool=nn.MaxPool1d,
            upsample_mode="linear",
        )
        assert list(u.in_out_channels) == list(ref_sched), (n_ch, n_blocks, max_ch)

    ours_m.eval()
    with torch.no_grad():
        y = ours_m(x)
    assert torch.allclose(y, ref["y"], atol=1e-6), (y - ref["y"]).abs().max()
