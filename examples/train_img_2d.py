#!/usr/bin/env python3
"""2D image-completion experiments — the reference's 2D notebook configs.

    python examples/train_img_2d.py --models GridConvCNP --datasets mnist
"""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).parent.parent))

from npf import zoo
from npf import CNPFLoss, ELBOLossLNPF, NLLLossLNPF
from npf.data.dataloader import cntxt_trgt_collate
from npf.recipes import add_y_dim, get_img_datasets
from npf.train import CVSplit, train_models
from npf.utils.datasplit import GridCntxtTrgtGetter, RandomMasker, no_masker

MODELS = {
    "CNP": (zoo.cnp_2d, CNPFLoss, False),
    "AttnCNP": (zoo.attncnp_2d, CNPFLoss, False),
    "AttnLNP": (zoo.attnlnp_2d, ELBOLossLNPF, False),
    "GridConvCNP": (zoo.gridconvcnp_2d, CNPFLoss, True),
    "GridConvLNP": (zoo.gridconvlnp_2d, NLLLossLNPF, True),
}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--models", nargs="+", default=["GridConvCNP"], choices=MODELS)
    p.add_argument("--datasets", nargs="+", default=["mnist"])
    p.add_argument("--epochs", type=int, default=50)
    p.add_argument("--chckpnt-dir", default="results/pretrained/")
    p.add_argument("--bf16", action="store_true")
    p.add_argument("--device-episodes", action="store_true")
    p.add_argument("--hipgraphs", action="store_true")
    p.add_argument("--batch-size", type=int, default=32)
    args = p.parse_args()

    train, test = get_img_datasets(args.datasets)

    for name in args.models:
        builder, criterion, is_grid = MODELS[name]
        models = add_y_dim({name: builder}, train)
        # reference 2D splitter: U(0, 30%) of pixels as context
        splitter = GridCntxtTrgtGetter(
            context_masker=RandomMasker(a=0.0, b=0.3),
            target_masker=no_masker,
        )
        collate = cntxt_trgt_collate(splitter, is_return_masks=is_grid)
        from functools import partial

        dev_split = partial(splitter, is_return_masks=is_grid)
        train_models(
            train,
            models,
            criterion(),
            test_datasets=test,
            chckpnt_dirname=args.chckpnt_dir,
            is_retrain=True,
            train_split=CVSplit(0.1),
            max_epochs=args.epochs,
            batch_size=args.batch_size,
            lr=1e-3,
            decay_lr=10,
            seed=123,
            iterator_train__collate_fn=collate,
            iterator_valid__collate_fn=collate,
            amp_dtype="bfloat16" if args.bf16 else None,
            device_episodes=dev_split if args.device_episodes else None,
            hipgraphs=args.hipgraphs,
            is_progressbar=True,
        )


if __name__ == "__main__":
    main()
