"""Trainer tests: loop, checkpoint scheme, determinism, eval protocol."""

import os

import numpy as np
import pytest
import torch

from npf import CNP, CNPFLoss
from npf.data import GPDataset, cntxt_trgt_collate
from npf.data.kernels import RBF
from npf.train import CVSplit, NPFTrainer, eval_loglike, set_seed, train_models
from npf.utils.datasplit import CntxtTrgtGetter, GetRandomIndcs, get_all_indcs


def _collate():
    return cntxt_trgt_collate(
        CntxtTrgtGetter(
            contexts_getter=GetRandomIndcs(a=0.1, b=0.4), targets_getter=get_all_indcs
        )
    )


def _dataset(n=64):
    return GPDataset(kernel=RBF(0.2), n_samples=n, n_points=32)


def test_fit_reduces_loss():
    set_seed(0)
    ds = _dataset(128)
    trainer = NPFTrainer(
        CNP(1, 1, r_dim=32), CNPFLoss(), collate_fn=_collate(), device="cpu",
        batch_size=32, max_epochs=5, lr=1e-3, train_split=None, seed=0,
    )
    trainer.fit(ds)
    assert trainer.history[-1]["train_loss"] < trainer.history[0]["train_loss"]
    assert trainer.history[0]["tasks_per_sec"] > 0


def test_checkpoint_dir_scheme(tmp_path):
    ds = _dataset(32)
    test_ds = _dataset(16)
    train_models(
        {"RBF_Kernel": ds},
        {"CNP": CNP(1, 1, r_dim=16)},
        CNPFLoss,
        test_datasets={"RBF_Kernel": test_ds},
        chckpnt_dirname=str(tmp_path) + "/",
        is_retrain=True,
        train_split=None,
        max_epochs=1,
        batch_size=16,
        seed=123,
        iterator_train__collate_fn=_collate(),
        iterator_valid__collate_fn=_collate(),
    )
    run_dir = tmp_path / "RBF_Kernel" / "CNP" / "run_0"
    # same file set as the reference checkpoint format
    assert sorted(os.listdir(run_dir)) == [
        "eval.csv", "model_summary.txt", "optimizer.pt", "params.pt",
    ]
    sd = torch.load(run_dir / "params.pt", map_location="cpu")
    assert all(isinstance(k, str) for k in sd)  # plain state_dict, no wrapper
    ll = np.loadtxt(run_dir / "eval.csv", delimiter=",")
    assert ll.shape == (16,)


def test_checkpoint_monitor_best_valid(tmp_path):
    ds = _dataset(64)
    trainer = NPFTrainer(
        CNP(1, 1, r_dim=16), CNPFLoss(), collate_fn=_collate(), device="cpu",
        batch_size=16, max_epochs=3, train_split=CVSplit(0.25),
        chckpnt_dirname=str(tmp_path), seed=1,
    )
    trainer.fit(ds)
    assert (tmp_path / "params.pt").exists()
    assert any(r.get("valid_loss_best") for r in trainer.history)


def test_determinism_same_seed():
    def run():
        set_seed(123)
        ds = _dataset(32)
        t = NPFTrainer(
            CNP(1, 1, r_dim=16), CNPFLoss(), collate_fn=_collate(), device="cpu",
            batch_size=16, max_epochs=2, train_split=None, seed=123,
        )
        t.fit(ds)
        return torch.cat([p.flatten() for p in t.module.parameters()])

    assert torch.equal(run(), run())


def test_eval_loglike_seeded_and_per_task():
    ds = _dataset(24)
    trainer = NPFTrainer(
        CNP(1, 1, r_dim=16), CNPFLoss(), collate_fn=_collate(), device="cpu",
        batch_size=8, train_split=None,
    )
    ll1 = eval_loglike(trainer, ds, seed=123)
    ll2 = eval_loglike(trainer, ds, seed=123)
    assert ll1.shape == (24,)
    assert np.allclose(ll1, ll2)  # same seed -> same context draws
    # reduction restored
    assert trainer.criterion.reduction == "mean"


def test_load_params_roundtrip(tmp_path):
    t1 = NPFTrainer(
        CNP(1, 1, r_dim=16), CNPFLoss(), collate_fn=_collate(), device="cpu",
        chckpnt_dirname=str(tmp_path), train_split=None, max_epochs=1, batch_size=8,
    )
    t1.fit(_dataset(16))
    t1.save_params()
    t2 = NPFTrainer(
        CNP(1, 1, r_dim=16), CNPFLoss(), device="cpu", chckpnt_dirname=str(tmp_path)
    )
    t2.load_params()
    for p1, p2 in zip(t1.module.parameters(), t2.module.parameters()):
        assert torch.equal(p1, p2)


def test_cli_train_eval_roundtrip(tmp_path):
    """End-to-end CLI: tiny train writes the reference file set; eval reloads."""
    from npf import cli

    cli.main([
        "train", "--model", "CNP", "--data", "RBF_Kernel",
        "--epochs", "1", "--n-tasks", "64", "--n-test-tasks", "64",
        "--batch-size", "32",
        "--chckpnt-dir", str(tmp_path) + "/",
        "--data-cache", str(tmp_path / "gp.npz"),
        "--device", "cpu",
    ])
    run = tmp_path / "RBF_Kernel" / "CNP" / "run_0"
    for f in ("params.pt", "optimizer.pt", "model_summary.txt", "eval.csv"):
        assert (run / f).exists(), f


def test_load_all_results(tmp_path):
    """Aggregates {data}/{model}/run_*/eval.csv into the reference DataFrame
    schema (reference utils/helpers.py:22-32)."""
    import numpy as np

    from npf.train import load_all_results

    for data, model, vals in [
        ("RBF_Kernel", "CNP", [1.0, 3.0]),
        ("RBF_Kernel", "ConvCNP", [10.0, 20.0]),
    ]:
        d = tmp_path / data / model / "run_0"
        d.mkdir(parents=True)
        np.savetxt(d / "eval.csv", np.array(vals))
    df = load_all_results(str(tmp_path) + "/")
    assert set(df.columns) == {"Data", "Model", "Runs", "LogLike"}
    assert len(df) == 2
    row = df[df.Model == "ConvCNP"].iloc[0]
    assert row.LogLike == 15.0


def test_sample_predictor():
    import sys

    sys.path.insert(0, "tests")
    from model_zoo import cnp_1d
    from npf.utils.predict import SamplePredictor

    m = cnp_1d().eval()
    Xc = torch.rand(2, 5, 1) * 2 - 1
    Yc = torch.randn(2, 5, 1)
    Xt = torch.rand(2, 9, 1) * 2 - 1
    loc = SamplePredictor(m)(Xc, Yc, Xt)
    assert loc.shape == (1, 2, 9, 1) and not loc.requires_grad
    dist = SamplePredictor(m, is_dist=True)(Xc, Yc, Xt)
    assert torch.allclose(dist.base_dist.loc, loc)


def test_dataset_merger_attr_forwarding():
    from npf.data import GPDataset
    from npf.data.helpers import DatasetMerger
    from npf.data.kernels import RBF

    a = GPDataset(kernel=RBF(0.2), n_samples=4, n_points=8)
    b = GPDataset(kernel=RBF(0.4), n_samples=4, n_points=8)
    m = DatasetMerger([a, b])
    assert len(m) == 8
    assert m.min_max == a.min_max  # attr forwarded from the first dataset
    x, y = m[5]
    assert x.shape == (8, 1)


def test_trainer_validates_input_range_on_cpu():
    """Out-of-range training features fail loudly at the loader boundary
    (the model-side check is skipped on GPU for hipGraph safety)."""
    from npf import CNPFLoss
    from npf.train.trainer import NPFTrainer

    class BadDS(torch.utils.data.Dataset):
        def __len__(self):
            return 8

        def __getitem__(self, i):
            return torch.randn(16, 1) * 3, torch.randn(16, 1)

    def collate(batch):
        X = torch.stack([b[0] for b in batch])
        y = torch.stack([b[1] for b in batch])
        inputs = dict(X_cntxt=X[:, :4], Y_cntxt=y[:, :4], X_trgt=X, Y_trgt=y)
        return inputs, y

    from npf.zoo import cnp_1d

    tr = NPFTrainer(
        cnp_1d(), CNPFLoss(), collate_fn=collate, device="cpu",
        batch_size=4, max_epochs=1,
    )
    with pytest.raises(ValueError, match=r"\[-1,1\]"):
        tr.fit(BadDS())


class TestDeviceEpisodes:
    """Device-resident episode loader (npf/train/device_loader.py)."""

    def _ds(self, n=32, reuse=True):
        from npf.data import GPDataset
        from npf.data.kernels import RBF

        set_seed(5)
        return GPDataset(
            kernel=RBF(0.2), n_samples=n, n_points=16,
            is_reuse_across_epochs=reuse,
        )

    def test_fit_and_eval_run(self):
        from npf import CNPFLoss
        from npf.train.device_loader import DeviceEpisodes
        from npf.train.trainer import NPFTrainer, eval_loglike
        from npf.utils.datasplit import CntxtTrgtGetter, GetRandomIndcs
        from npf.zoo import cnp_1d

        ds = self._ds(reuse=False)
        eps = DeviceEpisodes(
            ds, CntxtTrgtGetter(contexts_getter=GetRandomIndcs(a=2, b=6)),
            device="cpu",
        )
        tr = NPFTrainer(
            cnp_1d(), CNPFLoss(), device="cpu", batch_size=8, max_epochs=2,
            seed=0,
        )
        tr.fit(eps)
        assert len(tr.history) == 2
        assert tr.history[0]["train_loss"] is not None
        ll = eval_loglike(tr, DeviceEpisodes(
            self._ds(), CntxtTrgtGetter(contexts_getter=GetRandomIndcs(a=2, b=6)),
            device="cpu"), seed=123)
        assert ll.shape == (32,)

    def test_eval_matches_dataloader_path(self):
        """Same frozen tasks + deterministic splitter: the device-resident
        eval equals the DataLoader/collate eval per task."""
        import numpy as np

        from npf import CNPFLoss
        from npf.data import cntxt_trgt_collate
        from npf.train.device_loader import DeviceEpisodes
        from npf.train.trainer import NPFTrainer, eval_loglike
        from npf.utils.datasplit import (
            CntxtTrgtGetter,
            GetRangeIndcs,
            get_all_indcs,
        )
        from npf.zoo import cnp_1d

        ds = self._ds()
        splitter = CntxtTrgtGetter(
            contexts_getter=GetRangeIndcs((0, 5)), targets_getter=get_all_indcs
        )
        model = cnp_1d()

        tr1 = NPFTrainer(
            model, CNPFLoss(), device="cpu", batch_size=8,
            collate_fn=cntxt_trgt_collate(splitter),
        )
        ll_loader = eval_loglike(tr1, ds, seed=123)

        tr2 = NPFTrainer(model, CNPFLoss(), device="cpu", batch_size=8)
        ll_device = eval_loglike(
            tr2, DeviceEpisodes(ds, splitter, device="cpu"), seed=123
        )
        assert np.allclose(ll_loader, ll_device, atol=1e-5)

    def test_fresh_epochs_regenerate(self):
        from npf.train.device_loader import DeviceEpisodes
        from npf.utils.datasplit import CntxtTrgtGetter, GetRangeIndcs

        ds = self._ds(reuse=False)
        eps = DeviceEpisodes(
            ds, CntxtTrgtGetter(contexts_getter=GetRangeIndcs((0, 4))),
            device="cpu",
        )
        first = [y.clone() for _, y in eps.batches(8, training=True)]
        second = [y.clone() for _, y in eps.batches(8, training=True)]
        assert not torch.equal(first[0], second[0])


@pytest.mark.gpu
class TestGraphedStepEquivalence:
    def _mk(self, hipgraphs):
        from npf import CNPFLoss
        from npf.train.trainer import NPFTrainer
        from npf.zoo import attncnp_1d

        torch.manual_seed(0)
        tr = NPFTrainer(
            attncnp_1d(), CNPFLoss(), device="cuda", batch_size=8,
            amp_dtype=torch.bfloat16, hipgraphs=hipgraphs, seed=0,
        )
        return tr

    def _episodes(self, n_steps, seed=7):
        g = torch.Generator().manual_seed(seed)
        eps = []
        for i in range(n_steps):
            if i < n_steps - 2:
                # 13 shapes incl. zero context — but NOT at step 0: a
                # zero-grad first step leaves Adam exp_avg_sq == 0 for the
                # attention params, and then kernel-atomic noise flips
                # sign(g)-scale updates (legit nondeterminism, not a
                # stepper bug), which would fail the tight early check
                n_c = ((i + 1) * 5) % 13
            else:
                n_c = 47  # a shape first seen late (capture mid-run)
            Xc = (torch.rand(8, n_c, 1, generator=g) * 2 - 1).cuda()
            Yc = torch.randn(8, n_c, 1, generator=g).cuda()
            Xt = (torch.rand(8, 64, 1, generator=g) * 2 - 1).cuda()
            Yt = torch.randn(8, 64, 1, generator=g).cuda()
            eps.append((dict(X_cntxt=Xc, Y_cntxt=Yc, X_trgt=Xt, Y_trgt=Yt), Yt))
        return eps

    def test_graphed_steps_match_eager_steps(self):
        """The per-shape captured step must track the eager trajectory over
        80 steps spanning 14 shapes (incl. zero-context and one first seen
        late, so a capture happens mid-run).  Kernel atomics make long
        trajectories non-bitwise, so: first 10 steps tight, then the
        late-window loss means must agree."""
        eps = self._episodes(80)
        t_e = self._mk(hipgraphs=False)
        t_g = self._mk(hipgraphs=True)
        # identical initial parameters
        t_g.module.load_state_dict(t_e.module.state_dict())

        losses_e, losses_g = [], []
        for i, (inputs, y) in enumerate(eps):
            losses_e.append(float(t_e.train_step(inputs, y, _first=i == 0)))
        for i, (inputs, y) in enumerate(eps):
            losses_g.append(float(t_g.train_step(inputs, y, _first=i == 0)))
        torch.cuda.synchronize()

        import numpy as np

        assert np.allclose(losses_e[:10], losses_g[:10], rtol=5e-3, atol=5e-3), (
            list(zip(losses_e[:10], losses_g[:10]))
        )
        m_e = float(np.mean(losses_e[-20:]))
        m_g = float(np.mean(losses_g[-20:]))
        assert abs(m_e - m_g) < 0.3 * abs(m_e) + 5.0, (m_e, m_g)
        for (n1, p1), (n2, p2) in zip(
            t_e.module.named_parameters(), t_g.module.named_parameters()
        ):
            assert torch.allclose(p1, p2, atol=5e-2), (
                n1, (p1 - p2).abs().max()
            )


def test_device_episodes_image_grid():
    """Grid (image) datasets run through the device-resident loader: masker
    splitting on-device, GridConvCNP steps."""
    from npf import CNPFLoss
    from npf.data.imgs import SyntheticImages
    from npf.train.device_loader import DeviceEpisodes
    from npf.train.trainer import NPFTrainer
    from npf.utils.datasplit import GridCntxtTrgtGetter, RandomMasker, no_masker
    from npf.zoo import gridconvcnp_2d

    from functools import partial

    set_seed(0)
    ds = SyntheticImages(shape=(3, 16, 16), n_samples=16)
    eps = DeviceEpisodes(
        ds,
        partial(
            GridCntxtTrgtGetter(
                context_masker=RandomMasker(a=0.05, b=0.2),
                target_masker=no_masker,
            ),
            is_return_masks=True,  # GridConv models consume mask episodes
        ),
        device="cpu",
    )
    tr = NPFTrainer(
        gridconvcnp_2d(y_dim=3), CNPFLoss(), device="cpu", batch_size=8,
        max_epochs=1, seed=0,
    )
    tr.fit(eps)
    assert tr.history[0]["train_loss"] is not None
    import math

    assert math.isfinite(tr.history[0]["train_loss"])


def test_train_models_device_episodes_with_valid_split(tmp_path):
    """train_models(device_episodes=...) with a CVSplit: the split happens
    BEFORE wrapping (Subset of the raw dataset feeds DeviceEpisodes) and
    validation keeps the collate path + best-valid checkpointing."""
    from functools import partial

    from npf import CNPFLoss
    from npf.data.dataloader import cntxt_trgt_collate
    from npf.data.imgs import SyntheticImages
    from npf.train import CVSplit, train_models
    from npf.utils.datasplit import GridCntxtTrgtGetter, RandomMasker, no_masker
    from npf.zoo import gridconvcnp_2d

    set_seed(0)
    ds = SyntheticImages(shape=(3, 16, 16), n_samples=24)
    test_ds = SyntheticImages(shape=(3, 16, 16), n_samples=8, split="test")
    splitter = GridCntxtTrgtGetter(
        context_masker=RandomMasker(a=0.05, b=0.2), target_masker=no_masker
    )
    collate = cntxt_trgt_collate(splitter, is_return_masks=True)
    train_models(
        {"synthetic": ds},
        {"GridConvCNP": partial(gridconvcnp_2d, y_dim=3)},
        CNPFLoss,
        test_datasets={"synthetic": test_ds},
        chckpnt_dirname=str(tmp_path) + "/",
        is_retrain=True,
        train_split=CVSplit(0.25),
        max_epochs=1,
        batch_size=8,
        seed=0,
        iterator_train__collate_fn=collate,
        iterator_valid__collate_fn=collate,
        device_episodes=partial(splitter, is_return_masks=True),
    )
    run = tmp_path / "synthetic" / "GridConvCNP" / "run_0"
    assert (run / "params.pt").exists()
    ll = np.loadtxt(run / "eval.csv", delimiter=",")
    assert ll.shape == (8,) and np.isfinite(ll).all()
