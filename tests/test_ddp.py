"""Distributed (gloo, world_size=2) tests of the flat-buffer data-parallel
backend: gradient-mean equivalence with a single process, parameter
broadcast, and order-preserving sharded evaluation."""

import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from npf import CNP, CNPFLoss
from npf.train import set_seed

WORLD = 2


def _free_port():
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _init(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    dist.init_process_group("gloo", rank=rank, world_size=world)


def _make_batch(seed):
    g = torch.Generator().manual_seed(seed)
    Xc = torch.rand(4, 7, 1, generator=g) * 2 - 1
    Yc = torch.randn(4, 7, 1, generator=g)
    Xt = torch.rand(4, 16, 1, generator=g) * 2 - 1
    Yt = torch.randn(4, 16, 1, generator=g)
    return Xc, Yc, Xt, Yt


def _grad_equiv_worker(rank, port, out):
    _init(rank, WORLD, port)
    from npf.parallel import FlatDDP

    set_seed(0)
    model = CNP(1, 1, r_dim=16)
    ddp = FlatDDP(model)  # broadcasts rank-0 params
    crit = CNPFLoss()
    crit.train()
    model.train()

    Xc, Yc, Xt, Yt = _make_batch(seed=rank)  # each rank its own tasks
    ddp.zero_grad_()
    loss = crit(model(Xc, Yc, Xt, Yt), Yt)
    loss.backward()
    ddp.reduce_()
    out[rank] = ddp.flat_grads.clone()
    dist.destroy_process_group()


def test_flat_ddp_grad_matches_single_process():
    port = _free_port()
    mgr = mp.Manager()
    out = mgr.dict()
    mp.spawn(_grad_equiv_worker, args=(port, out), nprocs=WORLD, join=True)

    # single-process oracle: mean loss over the union of both ranks' batches
    set_seed(0)
    model = CNP(1, 1, r_dim=16)
    crit = CNPFLoss()
    crit.train()
    model.train()
    b0 = _make_batch(0)
    b1 = _make_batch(1)
    Xc = torch.cat([b0[0], b1[0]])
    Yc = torch.cat([b0[1], b1[1]])
    Xt = torch.cat([b0[2], b1[2]])
    Yt = torch.cat([b0[3], b1[3]])
    loss = crit(model(Xc, Yc, Xt, Yt), Yt)
    loss.backward()
    flat = torch.cat([p.grad.flatten() for p in model.parameters()])

    g0, g1 = out[0], out[1]
    assert torch.allclose(g0, g1, atol=1e-7)  # replicas agree
    assert torch.allclose(g0, flat, atol=1e-5), (g0 - flat).abs().max()


def _eval_worker(rank, port, out):
    _init(rank, WORLD, port)
    from npf.data import GPDataset, cntxt_trgt_collate
    from npf.data.kernels import RBF
    from npf.train import NPFTrainer, eval_loglike
    from npf.utils.datasplit import CntxtTrgtGetter, GetRandomIndcs, get_all_indcs

    set_seed(7)
    ds = GPDataset(kernel=RBF(0.2), n_samples=24, n_points=16)
    collate = cntxt_trgt_collate(
        CntxtTrgtGetter(
            contexts_getter=GetRandomIndcs(a=2, b=6), targets_getter=get_all_indcs
        )
    )
    model = CNP(1, 1, r_dim=16)
    trainer = NPFTrainer(
        model, CNPFLoss(), collate_fn=collate, device="cpu", batch_size=8,
        train_split=None,
    )
    ll = eval_loglike(trainer, ds, seed=123)
    out[rank] = ll
    dist.destroy_process_group()


def test_sharded_eval_preserves_row_count_and_agreement():
    port = _free_port()
    mgr = mp.Manager()
    out = mgr.dict()
    mp.spawn(_eval_worker, args=(port, out), nprocs=WORLD, join=True)
    ll0, ll1 = out[0], out[1]
    assert ll0.shape == (24,)
    assert np.allclose(ll0, ll1)  # every rank reassembles the same full vector


def _replica_sync_worker(rank, port, out):
    """3 optimizer steps at world=2: replicas must stay bitwise identical."""
    _init(rank, WORLD, port)
    from npf.parallel import FlatDDP

    set_seed(0)
    model = CNP(1, 1, r_dim=16)
    ddp = FlatDDP(model)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    crit = CNPFLoss()
    crit.train()
    model.train()
    for step in range(3):
        Xc, Yc, Xt, Yt = _make_batch(seed=1000 * step + rank)
        ddp.zero_grad_()
        loss = crit(model(Xc, Yc, Xt, Yt), Yt)
        loss.backward()
        ddp.reduce_()
        opt.step()
    out[rank] = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    dist.destroy_process_group()


def test_replicas_stay_identical_across_steps():
    port = _free_port()
    man = mp.Manager()
    out = man.dict()
    mp.spawn(_replica_sync_worker, args=(port, out), nprocs=WORLD, join=True)
    assert torch.equal(out[0], out[1])


def _ckpt_worker(rank, port, tmpdir, out):
    """Checkpoint written under world=2 training must equal what a world-1
    load sees (DP-invariant format, rank-0-only writes)."""
    _init(rank, WORLD, port)
    from npf.data import GPDataset, cntxt_trgt_collate
    from npf.data.kernels import RBF
    from npf.train import NPFTrainer
    from npf.utils.datasplit import CntxtTrgtGetter, GetRandomIndcs

    set_seed(0)
    ds = GPDataset(kernel=RBF(0.2), n_samples=32, n_points=16)
    collate = cntxt_trgt_collate(
        CntxtTrgtGetter(contexts_getter=GetRandomIndcs(a=3, b=6))
    )
    trainer = NPFTrainer(
        CNP(1, 1, r_dim=16), CNPFLoss(), collate_fn=collate, device="cpu",
        batch_size=8, max_epochs=1, chckpnt_dirname=tmpdir, monitor=None,
        seed=7,
    )
    trainer.fit(ds)
    out[rank] = torch.cat(
        [p.detach().reshape(-1) for p in trainer.module_.parameters()]
    )
    dist.destroy_process_group()


def test_checkpoint_dp_invariant(tmp_path):
    port = _free_port()
    man = mp.Manager()
    out = man.dict()
    mp.spawn(
        _ckpt_worker, args=(port, str(tmp_path), out), nprocs=WORLD, join=True
    )
    # replicas ended identical and the rank-0 checkpoint matches them
    assert torch.equal(out[0], out[1])
    sd = torch.load(os.path.join(str(tmp_path), "params.pt"), map_location="cpu")
    flat = torch.cat([v.reshape(-1) for v in sd.values()])
    assert torch.equal(flat, out[0])


# --------------------------------------------------------------------------- #
# world-8 CPU (gloo) coverage: the driver's 8-GPU scaling run must not be the
# first time the code sees world_size 8
# --------------------------------------------------------------------------- #


def _grad_equiv_worker_w(rank, world, port, out):
    _init(rank, world, port)
    from npf.parallel import FlatDDP

    set_seed(0)
    model = CNP(1, 1, r_dim=16)
    ddp = FlatDDP(model)
    crit = CNPFLoss()
    crit.train()
    model.train()
    Xc, Yc, Xt, Yt = _make_batch(seed=rank)
    ddp.zero_grad_()
    loss = crit(model(Xc, Yc, Xt, Yt), Yt)
    loss.backward()
    ddp.reduce_()
    out[rank] = ddp.flat_grads.clone()
    dist.destroy_process_group()


def test_flat_ddp_grad_matches_single_process_world8():
    world = 8
    port = _free_port()
    mgr = mp.Manager()
    out = mgr.dict()
    mp.spawn(_grad_equiv_worker_w, args=(world, port, out), nprocs=world, join=True)

    set_seed(0)
    model = CNP(1, 1, r_dim=16)
    crit = CNPFLoss()
    crit.train()
    model.train()
    batches = [_make_batch(r) for r in range(world)]
    Xc, Yc, Xt, Yt = (torch.cat([b[i] for b in batches]) for i in range(4))
    loss = crit(model(Xc, Yc, Xt, Yt), Yt)
    loss.backward()
    flat = torch.cat([p.grad.flatten() for p in model.parameters()])

    for r in range(1, world):
        assert torch.allclose(out[0], out[r], atol=1e-7)
    assert torch.allclose(out[0], flat, atol=1e-5), (out[0] - flat).abs().max()


def _ragged_eval_worker(rank, world, port, data, targets, out):
    if world > 1:
        _init(rank, world, port)
    from npf.data import GPDataset, cntxt_trgt_collate
    from npf.data.kernels import RBF
    from npf.train import NPFTrainer, eval_loglike
    from npf.utils.datasplit import CntxtTrgtGetter, GetRangeIndcs, get_all_indcs

    set_seed(3)
    ds = GPDataset(kernel=RBF(0.2), n_samples=4, n_points=16)
    ds.set_samples_(data, targets)
    # deterministic episodes: fixed context range => world-N eval must equal
    # world-1 eval elementwise after order restoration
    collate = cntxt_trgt_collate(
        CntxtTrgtGetter(
            contexts_getter=GetRangeIndcs((0, 5)), targets_getter=get_all_indcs
        )
    )
    model = CNP(1, 1, r_dim=16)
    trainer = NPFTrainer(
        model, CNPFLoss(), collate_fn=collate, device="cpu", batch_size=4,
        train_split=None, seed=0,
    )
    ll = eval_loglike(trainer, ds, seed=123)
    out[rank] = ll
    if world > 1:
        dist.destroy_process_group()


def test_sharded_eval_ragged_world8_matches_single_process():
    """27 tasks over 8 ranks (padded shards) reassemble to exactly the
    single-process per-task vector, in dataset order."""
    from npf.data import GPDataset
    from npf.data.kernels import RBF

    set_seed(11)
    src = GPDataset(kernel=RBF(0.2), n_samples=27, n_points=16)
    data, targets = src.data, src.targets

    world = 8
    port = _free_port()
    mgr = mp.Manager()
    out = mgr.dict()
    mp.spawn(
        _ragged_eval_worker, args=(world, port, data, targets, out),
        nprocs=world, join=True,
    )
    solo = mgr.dict()
    _ragged_eval_worker(0, 1, port, data, targets, solo)

    for r in range(world):
        assert out[r].shape == (27,)
        assert np.allclose(out[r], solo[0], atol=1e-6), r
