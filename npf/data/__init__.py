"""Data layer: GP meta-tasks, image datasets, collate."""

from .dataloader import cntxt_trgt_collate  # noqa: F401
from .gp import GPDataset  # noqa: F401
from .helpers import DIR_DATA, DatasetMerger, train_dev_split  # noqa: F401
from . import kernels  # noqa: F401
