"""Collate: dataset batches -> NPF model kwargs (the batch ABI).

Parity with /root/reference/utils/data/dataloader.py:6-37.
"""

import torch

__all__ = ["cntxt_trgt_collate"]


def cntxt_trgt_collate(get_cntxt_trgt, is_duplicate_batch=False, **kwargs):
    """Wrap a context/target splitter into a DataLoader collate_fn.

    Returns batches as `({X_cntxt, Y_cntxt, X_trgt, Y_trgt}, Y_trgt)` — the
    model-input ABI every trainer/evaluator in this framework uses.
    `is_duplicate_batch` repeats the batch so each function appears with two
    different context/target draws (used with UNet bottleneck forcing).
    """

    def collate(batch):
        collated = torch.utils.data.dataloader.default_collate(batch)
        X, y = collated[0], collated[1]

        if is_duplicate_batch:
            X = torch.cat([X, X], dim=0)
            if y is not None:
                y = torch.cat([y, y], dim=0)

        X_cntxt, Y_cntxt, X_trgt, Y_trgt = get_cntxt_trgt(X, y, **kwargs)
        inputs = dict(X_cntxt=X_cntxt, Y_cntxt=Y_cntxt, X_trgt=X_trgt, Y_trgt=Y_trgt)
        return inputs, Y_trgt

    return collate
