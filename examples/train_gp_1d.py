#!/usr/bin/env python3
"""1D GP-regression experiments — the reference's reproducibility notebooks
(CNP.ipynb .. ConvLNP.ipynb) as one script.

    python examples/train_gp_1d.py --models CNP AttnCNP ConvCNP \
        --datasets RBF_Kernel --epochs 100

Multi-GPU: `python -m torch.distributed.run --nproc-per-node 8
examples/train_gp_1d.py ...` — rank-sharded tasks, RCCL all-reduce.
"""

import argparse
import sys
from functools import partial
from pathlib import Path

sys.path.insert(0, str(Path(__file__).parent.parent))

from npf import zoo
from npf import CNPFLoss, ELBOLossLNPF, NLLLossLNPF
from npf.data.dataloader import cntxt_trgt_collate
from npf.recipes import get_datasets_single_gp
from npf.train import train_models
from npf.utils.datasplit import CntxtTrgtGetter, GetRandomIndcs, get_all_indcs

# (builder, criterion, kwargs) per model — the notebook training cells
MODELS = {
    "CNP": (zoo.cnp_1d, CNPFLoss, {}),
    "LNP": (zoo.lnp_1d, ELBOLossLNPF, {}),
    "AttnCNP": (zoo.attncnp_1d, CNPFLoss, {}),
    "AttnLNP": (zoo.attnlnp_1d, ELBOLossLNPF, {}),
    "ConvCNP": (zoo.convcnp_1d, CNPFLoss, {}),
    "ConvLNP": (zoo.convlnp_1d, NLLLossLNPF,
                {"batch_size": 16, "grad_clip_norm": 1.0}),
}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--models", nargs="+", default=["CNP"], choices=MODELS)
    p.add_argument("--datasets", nargs="+",
                   default=["RBF_Kernel", "Periodic_Kernel", "Noisy_Matern_Kernel"])
    p.add_argument("--epochs", type=int, default=100)
    p.add_argument("--n-tasks", type=int, default=50000)
    p.add_argument("--chckpnt-dir", default="results/pretrained/")
    p.add_argument("--bf16", action="store_true")
    p.add_argument("--device-episodes", action="store_true",
                   help="GPU-resident tasks + on-device splitting (fast path)")
    p.add_argument("--hipgraphs", action="store_true",
                   help="capture whole optimization steps per episode shape")
    p.add_argument("--runs-suffix", default="")
    args = p.parse_args()

    train, test, valid = get_datasets_single_gp(
        n_samples=args.n_tasks, defer_generation=args.device_episodes
    )
    train = {k: v for k, v in train.items() if k in args.datasets}

    # reference 1D splitter: U(0, 50) contexts, all 128 points as targets
    splitter = CntxtTrgtGetter(
        contexts_getter=GetRandomIndcs(a=0.0, b=50),
        targets_getter=get_all_indcs,
    )
    collate = cntxt_trgt_collate(splitter)

    for name in args.models:
        builder, criterion, extra = MODELS[name]
        train_models(
            train,
            {name: builder},
            criterion(),
            test_datasets=test,
            chckpnt_dirname=args.chckpnt_dir,
            is_retrain=True,
            train_split=None,
            max_epochs=args.epochs,
            batch_size=extra.get("batch_size", 32),
            lr=1e-3,
            decay_lr=10,
            seed=123,
            iterator_train__collate_fn=collate,
            iterator_valid__collate_fn=collate,
            amp_dtype="bfloat16" if args.bf16 else None,
            grad_clip_norm=extra.get("grad_clip_norm"),
            device_episodes=splitter if args.device_episodes else None,
            hipgraphs=args.hipgraphs,
            is_progressbar=True,
        )


if __name__ == "__main__":
    main()
