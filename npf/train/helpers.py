"""Harness helpers: seeding, LR schedules, result aggregation.

Capability parity with /root/reference/utils/helpers.py (load_all_results
:22-32, get_exponential_decay_gamma :35-46, set_seed :49-55, FixRandomSeed
:58-74, count_parameters :113-115, make_Xy_input :134-158, parallelize
:77-93).
"""

import glob
import logging
import os
import random
from multiprocessing import Pool, cpu_count

import numpy as np
import torch

__all__ = [
    "mean",
    "load_all_results",
    "get_exponential_decay_gamma",
    "set_seed",
    "fix_random_seed",
    "parallelize",
    "count_parameters",
    "make_Xy_input",
    "DisableLogger",
]

logger = logging.getLogger(__name__)


def mean(l):
    return sum(l) / len(l)


def load_all_results(folder):
    """Aggregate every `{data}/{model}/run_*/eval.csv` under `folder` into a
    DataFrame with the mean test log-likelihood per run."""
    import pandas as pd

    pattern = "*/*/run_*/eval.csv"
    rows = []
    for f in glob.glob(os.path.join(folder, pattern)):
        rows.append(f.split("/")[-4:-1] + [pd.read_csv(f, header=None).mean()[0]])
    df = pd.DataFrame(rows)
    df.columns = ["Data", "Model", "Runs", "LogLike"]
    return df


def get_exponential_decay_gamma(scheduling_factor, max_epochs):
    """Per-epoch gamma so the LR decays by `scheduling_factor` over training."""
    return (1 / scheduling_factor) ** (1 / max_epochs)


def set_seed(seed):
    """Seed torch (+cuda), random and numpy."""
    if seed is not None:
        torch.manual_seed(seed)
        torch.cuda.manual_seed(seed)
        random.seed(seed)
        np.random.seed(seed)


def fix_random_seed(seed=123, is_cudnn_deterministic=False):
    """One-call deterministic setup (the reference wrapped this in a skorch
    callback, utils/helpers.py:58-74)."""
    set_seed(seed)
    torch.backends.cudnn.deterministic = is_cudnn_deterministic


def parallelize(data, func, axis_split=0, n_chunks=None, cores=None):
    """Apply `func` to chunks of a numpy array with a process pool."""
    cores = cores or cpu_count()
    if n_chunks is None:
        n_chunks = cores * 2
    data_split = np.array_split(data, n_chunks, axis=axis_split)
    with Pool(cores) as pool:
        outs = pool.map(func, data_split)
    if isinstance(outs[0], tuple):
        outs = tuple(zip(*outs))
        return tuple(np.concatenate(o, axis=axis_split) for o in outs)
    return np.concatenate(outs, axis=axis_split)


class DisableLogger:
    def __enter__(self):
        logging.disable(50)

    def __exit__(self, a, b, c):
        logging.disable(logging.NOTSET)


def count_parameters(model):
    """Number of parameters in a model."""
    return sum(p.numel() for p in model.parameters())


def _first_item_view(dataset):
    class FirstIndex:
        def __init__(self, to_index):
            self.to_index = to_index

        def __getitem__(self, i):
            return self.to_index[i][0]

        def __len__(self):
            return len(self.to_index)

    return FirstIndex(dataset)


def make_Xy_input(dataset, y=None):
    """Adapt a dataset to ({"X": ..., "y": ...}, y) trainer inputs."""
    if isinstance(dataset, dict):
        y = dataset["y"]
        X = dataset["X"]
    elif isinstance(dataset, torch.utils.data.Dataset):
        if y is None:
            y = dataset.targets
        X = _first_item_view(dataset)
    else:
        X = dataset
    return ({"X": X, "y": y}, y)
