"""Native training loop (replaces the reference's skorch harness).

The reference drives training through `skorch.NeuralNet`
(/root/reference/utils/train.py:34-305); this trainer reproduces the
observable contract — checkpoint files (`params.pt` = plain module
state_dict, `optimizer.pt`, `model_summary.txt`, `eval.csv`), the
`(inputs_dict, Y_trgt)` batch ABI, `valid_loss_best` monitoring, exponential
LR decay, seed handling — with an MI355X-first loop:

- one process per GPU (RCCL/xGMI data parallel via npf.parallel.FlatDDP:
  single flat all-reduce per step);
- optional bf16 autocast compute (`amp_dtype`);
- epoch records include tasks/sec (the headline throughput metric).
"""

import logging
import os
import time
import numpy as np
import torch
from torch.optim import Adam
from torch.utils.data import DataLoader, Subset

from npf.parallel import ddp as dist_utils
from npf.train.helpers import get_exponential_decay_gamma, set_seed

__all__ = ["NPFTrainer", "CVSplit", "predefined_split", "train_models", "eval_loglike"]

logger = logging.getLogger(__name__)

EVAL_FILENAME = "eval.csv"
MOD_SUMM_FILENAME = "model_summary.txt"


class CVSplit:
    """Random train/valid split by fraction (skorch.dataset.CVSplit analog)."""

    def __init__(self, valid_fraction=0.1, seed=123):
        self.valid_fraction = valid_fraction
        self.seed = seed

    def __call__(self, dataset):
        n = len(dataset)
        n_valid = int(n * self.valid_fraction)
        g = torch.Generator().manual_seed(self.seed)
        perm = torch.randperm(n, generator=g).tolist()
        return Subset(dataset, perm[n_valid:]), Subset(dataset, perm[:n_valid])


def predefined_split(valid_dataset):
    """Use a fixed validation dataset (skorch.helper.predefined_split analog)."""

    def split(dataset):
        return dataset, valid_dataset

    return split


def _validate_episode(inputs):
    """Training features must be rescaled to [-1,1] (reference
    base.py:241-247).  The model skips the check on GPU tensors (a `.all()`
    there forces a device sync every step and is illegal under hipGraph
    capture), so the trainer enforces it HERE, on the loader's CPU tensors,
    before the H2D copy — bad data fails loudly without touching the hot
    path.  Device-resident episodes (GPU-side sampler) are trusted: the
    sampler constructs coordinates in-range by design."""
    for key in ("X_cntxt", "X_trgt"):
        X = inputs.get(key) if isinstance(inputs, dict) else None
        if X is not None and torch.is_tensor(X) and not X.is_cuda \
                and torch.is_floating_point(X) and X.numel():
            lo, hi = float(X.min()), float(X.max())
            if lo < -1 - 1e-6 or hi > 1 + 1e-6:
                raise ValueError(
                    f"Features during training should be in [-1,1]: "
                    f"{lo} <= {key} <= {hi}."
                )


def _move(obj, device, non_blocking=True):
    if torch.is_tensor(obj):
        return obj.to(device, non_blocking=non_blocking)
    if isinstance(obj, dict):
        return {k: _move(v, device, non_blocking) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return type(obj)(_move(v, device, non_blocking) for v in obj)
    return obj


class NPFTrainer:
    """Train / evaluate / checkpoint an NPF model.

    Parameters
    ----------
    module : NeuralProcessFamily (instance or zero-arg constructor)
    criterion : BaseLossNPF instance or class
    collate_fn : callable, optional
        `cntxt_trgt_collate(...)` output; maps raw batches to the
        ({X_cntxt, Y_cntxt, X_trgt, Y_trgt}, Y_trgt) ABI.
    train_split : callable, optional
        dataset -> (train, valid); e.g. `CVSplit(0.1)` / `predefined_split(v)`.
    monitor : {"valid_loss_best", None}
        Save the best-validation checkpoint, or overwrite every epoch.
    decay_lr : float, optional
        Total LR decay factor over training (exponential schedule).
    amp_dtype : torch.dtype, optional
        Autocast compute dtype on GPU (e.g. torch.bfloat16).
    is_ddp : bool
        Use the flat-buffer RCCL data-parallel reducer when a process group
        is initialized.
    """

    def __init__(
        self,
        module,
        criterion,
        *,
        optimizer=Adam,
        lr=1e-3,
        batch_size=32,
        max_epochs=10,
        device=None,
        collate_fn=None,
        valid_collate_fn=None,
        valid_batch_size=None,
        shuffle=True,
        train_split=None,
        monitor="valid_loss_best",
        chckpnt_dirname=None,
        seed=None,
        decay_lr=None,
        patience=None,
        grad_clip_norm=None,
        amp_dtype=None,
        is_ddp=True,
        num_workers=0,
        is_progressbar=False,
        profile=False,
        hipgraphs=False,
    ):
        self.rank = dist_utils.get_rank()
        self.world_size = dist_utils.get_world_size()

        if seed is not None:
            # rank-offset seed: each rank samples different tasks
            set_seed(seed + self.rank * 1000)
        self.seed = seed

        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)

        self.module = module() if callable(module) and not isinstance(
            module, torch.nn.Module
        ) else module
        self.module.to(self.device)

        self.criterion = criterion() if isinstance(criterion, type) else criterion

        self.lr = lr
        self.batch_size = batch_size
        self.max_epochs = max_epochs
        self.collate_fn = collate_fn
        self.valid_collate_fn = valid_collate_fn or collate_fn
        self.valid_batch_size = valid_batch_size or batch_size * 2
        self.shuffle = shuffle
        self.train_split = train_split
        self.monitor = monitor
        self.chckpnt_dirname = chckpnt_dirname
        self.decay_lr = decay_lr
        self.patience = patience
        self.grad_clip_norm = grad_clip_norm
        if isinstance(amp_dtype, str):  # "bfloat16" / "float16" from the CLI
            amp_dtype = getattr(torch, amp_dtype)
        self.amp_dtype = amp_dtype
        self.num_workers = num_workers
        self.is_progressbar = is_progressbar
        # rocprofv3-visible roctx ranges around step phases (SURVEY.md §5.1:
        # run under `rocprofv3 --kernel-trace --stats -- python ...` and the
        # ranges delimit forward/backward/optimizer per-kernel attribution)
        self.profile = profile and torch.cuda.is_available()

        # MI355X hot path: fused multi-tensor Adam (one kernel instead of
        # ~10 foreach launches), capturable with a device lr tensor so the
        # LR schedule survives hipGraph replay
        self.hipgraphs = bool(hipgraphs) and self.device.type == "cuda"
        self._lr_tensor = None
        if self.device.type == "cuda" and optimizer is Adam:
            self._lr_tensor = torch.tensor(float(lr), device=self.device)
            try:
                self.optimizer = Adam(
                    self.module.parameters(), lr=self._lr_tensor,
                    fused=True, capturable=True,
                )
            except Exception:
                self._lr_tensor = None
                self.optimizer = optimizer(self.module.parameters(), lr=lr)
        else:
            self.optimizer = optimizer(self.module.parameters(), lr=lr)
        self.scheduler = None
        self._lr_gamma = None
        if decay_lr is not None:
            gamma = get_exponential_decay_gamma(decay_lr, max_epochs)
            self._lr_gamma = gamma
            if self._lr_tensor is None:
                self.scheduler = torch.optim.lr_scheduler.ExponentialLR(
                    self.optimizer, gamma=gamma
                )
        self._step_graphs = None

        self.ddp = None
        if is_ddp and dist_utils.is_distributed():
            self.ddp = dist_utils.FlatDDP(self.module)

        self.history = []
        self._best_valid = float("inf")

    @property
    def module_(self):
        """skorch-compatible alias (reference code reads `trainer.module_`)."""
        return self.module

    # ------------------------------------------------------------------ #
    # loops
    # ------------------------------------------------------------------ #

    def _autocast(self):
        if self.amp_dtype is not None and self.device.type == "cuda":
            return torch.autocast(device_type="cuda", dtype=self.amp_dtype)
        import contextlib

        return contextlib.nullcontext()

    def _loader(self, dataset, training):
        from torch.utils.data.distributed import DistributedSampler

        from npf.train.device_loader import DeviceEpisodes

        if isinstance(dataset, DeviceEpisodes):
            # device-resident episodes: no DataLoader, no collate, no H2D;
            # fresh-task ranks draw from their own RNG stream (rank-offset
            # seed), so no sampler sharding either
            return dataset.batches(
                self.batch_size if training else self.valid_batch_size,
                training=training,
            )

        sampler = None
        shuffle = self.shuffle if training else False
        if self.world_size > 1 and getattr(dataset, "is_rank_sharded", False) is False:
            # map-style datasets are sharded across ranks; fresh-task datasets
            # (is_rank_sharded=True) already draw rank-local tasks
            sampler = DistributedSampler(
                dataset, num_replicas=self.world_size, rank=self.rank,
                shuffle=shuffle, drop_last=training,
            )
            shuffle = False
        return DataLoader(
            dataset,
            batch_size=self.batch_size if training else self.valid_batch_size,
            shuffle=shuffle,
            sampler=sampler,
            collate_fn=self.collate_fn if training else self.valid_collate_fn,
            num_workers=self.num_workers,
            drop_last=training,
        )

    def _range_push(self, name):
        if self.profile:
            torch.cuda.nvtx.range_push(name)  # roctx range on ROCm

    def _range_pop(self):
        if self.profile:
            torch.cuda.nvtx.range_pop()

    def _step_body(self, inputs, y):
        """fwd + loss + bwd + reduce + clip + optimizer — replay-safe."""
        if self.ddp is not None:
            self.ddp.zero_grad_()
        else:
            self.optimizer.zero_grad(set_to_none=True)
        with self._autocast():
            outputs = self.module(**inputs)
        loss = self.criterion(outputs, y)  # loss math in fp32
        loss.backward()
        if self.ddp is not None:
            self.ddp.reduce_()
        if self.grad_clip_norm is not None:
            torch.nn.utils.clip_grad_norm_(
                self.module.parameters(), self.grad_clip_norm
            )
        self.optimizer.step()
        return loss.detach()

    def _maybe_graph_step(self, inputs, y):
        """Replay the step as one hipGraph when eligible (fixed-address
        buffers, RNG via the noise pool); returns None to run eager."""
        if not self.hipgraphs:
            return None
        if self._step_graphs is None:
            # the first eager step (run by the caller) has primed optimizer
            # state; SUMO-style stochastic z counts are not graph-safe
            n_z = getattr(self.module, "n_z_samples_train", None)
            if hasattr(n_z, "rvs"):
                self.hipgraphs = False
                return None
            from npf.train.step_graphs import GraphedStepper

            self._step_graphs = GraphedStepper(
                self.module, self.optimizer, self._step_body
            )
        return self._step_graphs.step(inputs, y)

    def train_step(self, inputs, y, _first=False):
        """One optimization step; returns the loss (0-dim device tensor)."""
        self.module.train()
        self.criterion.train()
        self._range_push("npf/step")
        loss = None if _first else self._maybe_graph_step(inputs, y)
        if loss is None:
            loss = self._step_body(inputs, y)
        else:
            loss = loss.detach().clone()  # static buffer: next replay overwrites
        self._range_pop()
        return loss

    def validation_step(self, inputs, y):
        self.module.eval()
        self.criterion.eval()
        with torch.no_grad(), self._autocast():
            outputs = self.module(**inputs)
            loss = self.criterion(outputs, y)
        return loss.detach()

    def fit(self, dataset, valid_dataset=None):
        """Full training run with per-epoch validation / checkpointing."""
        if valid_dataset is None and self.train_split is not None:
            dataset, valid_dataset = self.train_split(dataset)

        epochs_no_improve = 0
        for epoch in range(self.max_epochs):
            t0 = time.perf_counter()
            n_tasks = 0
            train_losses = []
            loader = self._loader(dataset, training=True)
            if hasattr(getattr(loader, "sampler", None), "set_epoch"):
                loader.sampler.set_epoch(epoch)
            first = epoch == 0
            for inputs, y in loader:
                _validate_episode(inputs)
                inputs = _move(inputs, self.device)
                y = _move(y, self.device)
                # losses stay on-device until epoch end: a float() here
                # would sync the pipeline every step
                loss = self.train_step(inputs, y, _first=first)
                first = False
                train_losses.append(loss)
                n_tasks += y.shape[0] * self.world_size
            if self.device.type == "cuda":
                torch.cuda.synchronize()
            dur = time.perf_counter() - t0

            record = {
                "epoch": epoch + 1,
                "train_loss": (
                    float(torch.stack(train_losses).float().mean())
                    if train_losses else None
                ),
                "dur": dur,
                "tasks_per_sec": n_tasks / dur if dur > 0 else None,
                "lr": float(self.optimizer.param_groups[0]["lr"]),
            }

            if valid_dataset is not None:
                valid_losses = []
                for inputs, y in self._loader(valid_dataset, training=False):
                    inputs = _move(inputs, self.device)
                    y = _move(y, self.device)
                    valid_losses.append(float(self.validation_step(inputs, y)))
                record["valid_loss"] = float(np.mean(valid_losses))
                is_best = record["valid_loss"] < self._best_valid
                record["valid_loss_best"] = is_best
                if is_best:
                    self._best_valid = record["valid_loss"]
                    epochs_no_improve = 0
                else:
                    epochs_no_improve += 1

            self.history.append(record)

            # checkpointing policy mirrors the reference (train.py:203-221):
            # best-valid when monitored, else overwrite-every-epoch
            if self.chckpnt_dirname is not None and self.rank == 0:
                if self.monitor is None or valid_dataset is None or record.get(
                    "valid_loss_best", False
                ):
                    self.save_params()

            if self.scheduler is not None:
                self.scheduler.step()
            elif self._lr_gamma is not None and self._lr_tensor is not None:
                # write the decayed lr INTO the device tensor the (possibly
                # graph-captured) fused Adam reads on every step
                self._lr_tensor.fill_(self.lr * self._lr_gamma ** (epoch + 1))

            if self.is_progressbar and self.rank == 0:
                msg = f"epoch {epoch + 1}/{self.max_epochs} " + " ".join(
                    f"{k}={v:.4g}" for k, v in record.items()
                    if isinstance(v, (int, float)) and k != "epoch"
                )
                print(msg, flush=True)

            if self.patience is not None and epochs_no_improve >= self.patience:
                break

        dist_utils.barrier()
        return self

    # ------------------------------------------------------------------ #
    # checkpointing (reference format: SURVEY.md §5.4)
    # ------------------------------------------------------------------ #

    def save_params(self, dirname=None):
        dirname = dirname or self.chckpnt_dirname
        os.makedirs(dirname, exist_ok=True)
        torch.save(self.module.state_dict(), os.path.join(dirname, "params.pt"))
        torch.save(self.optimizer.state_dict(), os.path.join(dirname, "optimizer.pt"))

    def load_params(self, dirname=None):
        dirname = dirname or self.chckpnt_dirname
        sd = torch.load(os.path.join(dirname, "params.pt"), map_location=self.device)
        self.module.load_state_dict(sd)
        opt_path = os.path.join(dirname, "optimizer.pt")
        if os.path.exists(opt_path):
            try:
                self.optimizer.load_state_dict(
                    torch.load(opt_path, map_location=self.device)
                )
            except ValueError:
                logger.warning("optimizer.pt incompatible; keeping fresh optimizer")
        return self


def eval_loglike(trainer, dataset, seed=123):
    """Per-task test log-likelihood vector (reference utils/evaluate.py:9-28).

    Seed 123 fixes the context/target draws; the criterion runs unreduced and
    in eval mode (NPML forced).  Under DDP the dataset is rank-sharded and
    the rows are re-assembled in original order on every rank.
    """
    set_seed(seed)
    old_reduction = trainer.criterion.reduction
    trainer.criterion.reduction = None
    trainer.module.to(trainer.device)

    world = trainer.world_size
    all_ll = []
    loader = trainer._loader(dataset, training=False)
    for inputs, y in loader:
        inputs = _move(inputs, trainer.device)
        y = _move(y, trainer.device)
        loss = trainer.validation_step(inputs, y)
        all_ll.append(-loss.float().cpu())
    trainer.criterion.reduction = old_reduction
    ll = torch.cat(all_ll, dim=0)

    from npf.train.device_loader import DeviceEpisodes

    if isinstance(dataset, DeviceEpisodes):
        # device episodes are not rank-sharded: every rank evaluated the
        # whole set already
        return ll.numpy()

    if world > 1:
        # DistributedSampler (shuffle=False) pads the index list to a
        # multiple of `world` by repeating its head, then deals it
        # round-robin; invert exactly that assignment so rows come back in
        # dataset order (padded duplicates collapse onto their slot).
        ll_dev = ll.to(trainer.device)
        gathered = dist_utils.all_gather_cat(ll_dev, dim=0).cpu()
        n_total = len(dataset)
        shard_len = gathered.shape[0] // world
        padded = np.arange(world * shard_len) % n_total
        ll = torch.empty(n_total, dtype=gathered.dtype)
        for r in range(world):
            for j in range(shard_len):
                ll[padded[r + j * world]] = gathered[r * shard_len + j]

    return ll.numpy()


def train_models(
    datasets,
    models,
    criterion,
    test_datasets=dict(),
    valid_datasets=dict(),
    chckpnt_dirname=None,
    is_continue_train=False,
    is_retrain=False,
    runs=1,
    starting_run=0,
    train_split=CVSplit(0.1),
    device=None,
    max_epochs=100,
    batch_size=16,
    lr=1e-3,
    optimizer=Adam,
    patience=None,
    decay_lr=None,
    is_reeval=False,
    seed=None,
    datasets_kwargs=dict(),
    models_kwargs=dict(),
    device_episodes=None,
    **kwargs,
):
    """Grid-train {datasets} x {models} x runs (reference train.py:34-305).

    Accepts the reference's double-underscore kwargs
    (`iterator_train__collate_fn`, `iterator_valid__batch_size`, ...) and
    writes the same per-run directory scheme:
    `{chckpnt_dirname}{data}/{model}/run_{k}/{params.pt,optimizer.pt,
    model_summary.txt,eval.csv}`.

    `device_episodes`: a CntxtTrgtGetter — wraps each training dataset in
    DeviceEpisodes (tasks resident on the GPU, on-device splitting, fresh
    epochs regenerated by the batched GPU sampler) instead of the
    DataLoader/collate path.  Evaluation keeps the collate path (protocol
    fidelity; it is off the hot path).
    """
    trainers = dict()

    def to_trainer_kwargs(kw):
        out = {}
        mapping = {
            "iterator_train__collate_fn": "collate_fn",
            "iterator_valid__collate_fn": "valid_collate_fn",
            "iterator_valid__batch_size": "valid_batch_size",
            "iterator_train__shuffle": "shuffle",
        }
        for k, v in kw.items():
            out[mapping.get(k, k)] = v
        return out

    for data_name, data_train in datasets.items():
        current_models = (
            models[data_name]
            if isinstance(next(iter(models.values())), dict)
            else models
        )
        data_test = test_datasets.get(data_name, None)
        data_valid = valid_datasets.get(data_name, None)
        curr_split = train_split if data_valid is None else predefined_split(data_valid)

        for model_name, model in current_models.items():
            for run in range(starting_run, starting_run + runs):
                suffix = f"{data_name}/{model_name}/run_{run}"
                if dist_utils.get_rank() == 0:
                    print(
                        f"\n--- {'Training' if is_retrain else 'Loading'} {suffix} ---\n",
                        flush=True,
                    )
                run_dir = (chckpnt_dirname + suffix) if chckpnt_dirname else None

                run_kwargs = dict(kwargs)
                run_kwargs.update(datasets_kwargs.get(data_name, dict()))
                run_kwargs.update(models_kwargs.get(model_name, dict()))
                run_kwargs = to_trainer_kwargs(run_kwargs)

                trainer = NPFTrainer(
                    model,
                    criterion,
                    optimizer=optimizer,
                    lr=lr,
                    batch_size=batch_size,
                    max_epochs=max_epochs,
                    device=device,
                    train_split=curr_split,
                    monitor="valid_loss_best" if curr_split is not None else None,
                    chckpnt_dirname=run_dir,
                    seed=seed + run if seed is not None else None,
                    decay_lr=decay_lr,
                    patience=patience,
                    **run_kwargs,
                )

                if is_continue_train and run_dir and os.path.exists(
                    os.path.join(run_dir, "params.pt")
                ):
                    trainer.load_params()

                if is_retrain:
                    fit_data, fit_valid = data_train, None
                    if device_episodes is not None:
                        from npf.train.device_loader import DeviceEpisodes

                        # split BEFORE wrapping (Subset of the raw dataset);
                        # validation keeps the DataLoader/collate path
                        if trainer.train_split is not None:
                            fit_data, fit_valid = trainer.train_split(fit_data)
                            trainer.train_split = None
                        fit_data = DeviceEpisodes(
                            fit_data, device_episodes, device=trainer.device
                        )
                    trainer.fit(fit_data, valid_dataset=fit_valid)
                    if run_dir and dist_utils.get_rank() == 0:
                        with open(os.path.join(run_dir, MOD_SUMM_FILENAME), "w") as f:
                            f.write(str(trainer.module))
                        if not os.path.exists(os.path.join(run_dir, "params.pt")):
                            trainer.save_params()

                if run_dir and os.path.exists(os.path.join(run_dir, "params.pt")):
                    trainer.load_params()

                test_loglike = None
                if data_test is not None:
                    eval_file = (
                        os.path.join(run_dir, EVAL_FILENAME) if run_dir else None
                    )
                    ll = None
                    if (
                        eval_file
                        and os.path.exists(eval_file)
                        and not (is_retrain or is_reeval)
                    ):
                        ll = np.loadtxt(eval_file, delimiter=",")
                    if ll is None:
                        ll = eval_loglike(trainer, data_test)
                        if eval_file and dist_utils.get_rank() == 0:
                            np.savetxt(eval_file, ll, delimiter=",")
                    test_loglike = float(np.mean(ll))

                valid_loss, best_epoch = _best_loss(trainer, "valid")
                train_loss, _ = _best_loss(trainer, "train")
                if dist_utils.get_rank() == 0:
                    print(
                        suffix,
                        "| best epoch:", best_epoch,
                        "| train loss:", _round(train_loss),
                        "| valid loss:", _round(valid_loss),
                        "| test log likelihood:", _round(test_loglike),
                        flush=True,
                    )

                trainer.module.cpu()
                if torch.cuda.is_available():
                    torch.cuda.empty_cache()
                trainers[suffix] = trainer

    return trainers


def _round(x, n=4):
    return None if x is None else float(f"{x:.{n}f}")


def _best_loss(trainer, mode="valid"):
    try:
        best_epoch, best = None, None
        for rec in trainer.history:
            if mode == "valid" and rec.get("valid_loss_best"):
                best, best_epoch = rec["valid_loss"], rec["epoch"]
            elif mode == "train" and rec.get("train_loss") is not None:
                if best is None or rec["train_loss"] < best:
                    best, best_epoch = rec["train_loss"], rec["epoch"]
        return best, best_epoch
    except Exception:
        return None, None
