"""Thin CLI over the programmatic training API (`python -m npf.cli ...`).

The reference has no CLI (SURVEY.md §5.6 — notebooks compose
functools.partial trees); this maps the same canonical configurations
(npf.recipes + the model zoo in npf.zoo) onto flags:

    python -m npf.cli train --model ConvCNP --data RBF_Kernel \
        --epochs 10 --batch-size 32 --chckpnt-dir results/

    python -m npf.cli eval  --model ConvCNP --data RBF_Kernel \
        --chckpnt-dir results/

    python -m npf.cli list  # available models / datasets

Multi-GPU: launch under `torch.distributed.run --nproc-per-node N` — the
trainer shards tasks per rank and all-reduces gradients over RCCL/xGMI.
"""

import argparse

MODELS_1D = {
    "CNP": "cnp_1d",
    "LNP": "lnp_1d",
    "AttnCNP": "attncnp_1d",
    "AttnLNP": "attnlnp_1d",
    "ConvCNP": "convcnp_1d",
    "ConvLNP": "convlnp_1d",
}
MODELS_2D = {
    "CNP": "cnp_2d",
    "AttnCNP": "attncnp_2d",
    "AttnLNP": "attnlnp_2d",
    "GridConvCNP": "gridconvcnp_2d",
    "GridConvLNP": "gridconvlnp_2d",
}
GP_DATASETS = [
    "RBF_Kernel", "Periodic_Kernel", "Noisy_Matern_Kernel",
    "Variable_Matern_Kernel", "All_Kernels",
]
LATENT_MODELS = {"LNP", "AttnLNP", "ConvLNP", "GridConvLNP"}


def _build_loss(model_name, loss_name):
    from npf import CNPFLoss, ELBOLossLNPF, NLLLossLNPF, SUMOLossLNPF

    if loss_name == "auto":
        loss_name = "elbo" if model_name in LATENT_MODELS else "cnpf"
        if model_name in ("ConvLNP", "GridConvLNP"):
            loss_name = "nll"
    return {
        "cnpf": CNPFLoss, "elbo": ELBOLossLNPF,
        "nll": NLLLossLNPF, "sumo": SUMOLossLNPF,
    }[loss_name]()


def _build_data(args):
    """Returns (train, test, valid, is_1d)."""
    from npf.recipes import get_gp_datasets, get_img_datasets
    from npf.data.kernels import RBF, ExpSineSquared, Matern, WhiteKernel

    if args.data in GP_DATASETS:
        kernels = {
            "RBF_Kernel": RBF(length_scale=0.2),
            "Periodic_Kernel": ExpSineSquared(length_scale=0.5, periodicity=0.5),
            "Noisy_Matern_Kernel": WhiteKernel(noise_level=0.1)
            + Matern(length_scale=0.2, nu=1.5),
            "Variable_Matern_Kernel": Matern(
                length_scale=0.1, length_scale_bounds=(0.01, 0.3), nu=1.5
            ),
        }
        if args.data == "All_Kernels":
            from npf.recipes import get_datasets_variable_kernel_gp

            d, t, v = get_datasets_variable_kernel_gp(
                n_samples=args.n_tasks, save_file=args.data_cache,
                n_test=args.n_test_tasks,
            )
        else:
            d, t, v = get_gp_datasets(
                {args.data: kernels[args.data]},
                save_file=args.data_cache,
                n_samples=args.n_tasks,
                n_points=128,
                is_vary_kernel_hyp=args.data == "Variable_Matern_Kernel",
                is_reuse_across_epochs=False,
                n_test=args.n_test_tasks,
            )
        return d, t, v, True
    train, test = get_img_datasets([args.data])
    return train, test, dict(), False


def _raw_splitter(is_1d):
    from npf.utils.datasplit import (
        CntxtTrgtGetter, GetRandomIndcs, GridCntxtTrgtGetter, RandomMasker,
        get_all_indcs, no_masker,
    )

    if is_1d:
        return CntxtTrgtGetter(
            contexts_getter=GetRandomIndcs(a=0.0, b=50),
            targets_getter=get_all_indcs,
        )
    return GridCntxtTrgtGetter(
        context_masker=RandomMasker(a=0.0, b=0.3), target_masker=no_masker
    )


def _splitter(is_1d, img_shape=None):
    from npf.data.dataloader import cntxt_trgt_collate
    from npf.utils.datasplit import (
        CntxtTrgtGetter, GetRandomIndcs, GridCntxtTrgtGetter, RandomMasker,
        get_all_indcs, no_masker,
    )

    if is_1d:
        getter = CntxtTrgtGetter(
            contexts_getter=GetRandomIndcs(a=0.0, b=50), targets_getter=get_all_indcs
        )
        return cntxt_trgt_collate(getter)
    getter = GridCntxtTrgtGetter(
        context_masker=RandomMasker(a=0.0, b=0.3), target_masker=no_masker
    )
    return cntxt_trgt_collate(getter)


def cmd_list(_args):
    print("1D models:", ", ".join(MODELS_1D))
    print("2D models:", ", ".join(MODELS_2D))
    print("GP datasets:", ", ".join(GP_DATASETS))
    from npf.data.imgs import DATASETS_DICT

    print("image datasets:", ", ".join(DATASETS_DICT))


def _common(args, is_retrain):
    from npf import zoo
    from functools import partial
    from npf.train import train_models

    train, test, valid, is_1d = _build_data(args)
    table = MODELS_1D if is_1d else MODELS_2D
    if args.model not in table:
        raise SystemExit(
            f"model {args.model} not available for {'1D' if is_1d else '2D'} data "
            f"(choose from {list(table)})"
        )
    builder = getattr(zoo, table[args.model])
    if not is_1d:
        y_dim = next(iter(train.values())).shape[0]
        builder = partial(builder, y_dim=y_dim)

    # loss-ablation knobs (the 24-model grid of the reference Losses.ipynb):
    # lower bounds on predictive / latent std via the transformer kwargs
    if args.min_sigma_pred != 0.01 or args.min_lat is not None:
        from npf.recipes import get_std_processing_kwargs

        std_kwargs = get_std_processing_kwargs(
            min_sigma_pred=args.min_sigma_pred, min_lat=args.min_lat
        )
        builder = partial(_apply_std_kwargs, builder, std_kwargs)

    criterion = _build_loss(args.model, args.loss)
    collate = _splitter(is_1d)

    trainers = train_models(
        train,
        {args.model: builder},
        criterion,
        test_datasets=test,
        valid_datasets=valid,
        chckpnt_dirname=args.chckpnt_dir,
        is_retrain=is_retrain,
        train_split=None if is_1d else None,
        device=args.device,
        max_epochs=args.epochs,
        batch_size=args.batch_size,
        lr=args.lr,
        decay_lr=args.decay_lr,
        seed=args.seed,
        runs=args.runs,
        is_reeval=not is_retrain,
        iterator_train__collate_fn=collate,
        iterator_valid__collate_fn=collate,
        amp_dtype="bfloat16" if args.bf16 else None,
        device_episodes=_raw_splitter(is_1d) if args.device_episodes else None,
        hipgraphs=args.hipgraphs,
    )
    return trainers


def _apply_std_kwargs(builder, std_kwargs):
    model = builder()
    import torch.nn as nn

    if "p_y_scale_transformer" in std_kwargs:
        model.p_y_scale_transformer = std_kwargs["p_y_scale_transformer"]
    if "q_z_scale_transformer" in std_kwargs and hasattr(
        model, "q_z_scale_transformer"
    ):
        model.q_z_scale_transformer = std_kwargs["q_z_scale_transformer"]
    return model


def cmd_train(args):
    _common(args, is_retrain=True)


def cmd_eval(args):
    _common(args, is_retrain=False)


def main(argv=None):
    p = argparse.ArgumentParser(prog="npf", description=__doc__)
    sub = p.add_subparsers(dest="cmd", required=True)

    sub.add_parser("list", help="list models/datasets").set_defaults(fn=cmd_list)

    for name, fn in (("train", cmd_train), ("eval", cmd_eval)):
        q = sub.add_parser(name)
        q.add_argument("--model", required=True)
        q.add_argument("--data", required=True)
        q.add_argument("--loss", default="auto",
                       choices=["auto", "cnpf", "elbo", "nll", "sumo"])
        q.add_argument("--epochs", type=int, default=100)
        q.add_argument("--batch-size", type=int, default=32)
        q.add_argument("--lr", type=float, default=1e-3)
        q.add_argument("--decay-lr", type=float, default=10)
        q.add_argument("--seed", type=int, default=123)
        q.add_argument("--runs", type=int, default=1)
        q.add_argument("--n-tasks", type=int, default=50000)
        q.add_argument("--n-test-tasks", type=int, default=10000)
        q.add_argument("--chckpnt-dir", default="results/")
        q.add_argument("--data-cache", default="data/gp_dataset.npz")
        q.add_argument("--device", default=None)
        q.add_argument("--bf16", action="store_true")
        q.add_argument("--device-episodes", action="store_true",
                       help="GPU-resident episodes (no DataLoader/collate)")
        q.add_argument("--hipgraphs", action="store_true",
                       help="capture optimization steps per episode shape")
        q.add_argument("--min-sigma-pred", type=float, default=0.01,
                       help="predictive-std lower bound (ablation grid knob)")
        q.add_argument("--min-lat", type=float, default=None,
                       help="latent-std lower bound (ablation grid knob)")
        q.set_defaults(fn=fn)

    args = p.parse_args(argv)
    args.fn(args)


if __name__ == "__main__":
    main()
