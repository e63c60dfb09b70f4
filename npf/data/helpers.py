"""Dataset plumbing: merging, train/dev splitting, image preprocessing.

Capability parity with /root/reference/utils/data/helpers.py (DatasetMerger
:18-39, _DatasetSubset :42-85, train_dev_split :88-113, preprocess :116-154,
random_translation :157-171).  The chunk cache lives in npf/data/gp.py
(npz-based; this image has no h5py).
"""

import glob
import logging
import os

import numpy as np
from PIL import Image
from sklearn.model_selection import train_test_split
from torch.utils.data import Dataset

DIR_DATA = os.path.abspath(
    os.path.join(os.path.dirname(__file__), "../../data/")
)
logger = logging.getLogger(__name__)

__all__ = [
    "DIR_DATA",
    "DatasetMerger",
    "train_dev_split",
    "preprocess",
    "random_translation",
]


class DatasetMerger(Dataset):
    """Concatenate datasets; attribute lookups go to the first one."""

    def __init__(self, datasets):
        self.datasets = list(datasets)
        self.cumul_len = np.cumsum([len(d) for d in self.datasets])

    def __getitem__(self, index):
        idx_dataset = self.cumul_len.searchsorted(index + 1)
        idx_in_dataset = index
        if idx_dataset > 0:
            idx_in_dataset -= self.cumul_len[idx_dataset - 1]
        return self.datasets[idx_dataset][idx_in_dataset]

    def __len__(self):
        return int(self.cumul_len[-1])

    def __getattr__(self, attr):
        return getattr(self.datasets[0], attr)


class _DatasetSubset(Dataset):
    """Index-mapped view of a dataset (keeps `targets`/`data` accessors)."""

    def __init__(self, to_split, idx_mapping):
        self.idx_mapping = idx_mapping
        self.length = len(idx_mapping)
        self.to_split = to_split

    def __getitem__(self, index):
        return self.to_split[self.idx_mapping[index]]

    def __len__(self):
        return self.length

    @property
    def targets(self):
        return self.to_split.targets[self.idx_mapping]

    @targets.setter
    def targets(self, values):
        self.to_split.targets[self.idx_mapping] = values

    @property
    def data(self):
        return self.to_split.data[self.idx_mapping]

    def __getattr__(self, attr):
        return getattr(self.to_split, attr)


def train_dev_split(to_split, dev_size=0.1, seed=123, is_stratify=True):
    """Split a dataset into train/dev subsets (optionally label-stratified)."""
    idcs_all = list(range(len(to_split)))
    stratify = to_split.targets if is_stratify else None
    idcs_train, idcs_val = train_test_split(
        idcs_all, stratify=stratify, test_size=dev_size, random_state=seed
    )
    return _DatasetSubset(to_split, idcs_train), _DatasetSubset(to_split, idcs_val)


def preprocess(root, size=(64, 64), img_format="JPEG", center_crop=None):
    """Resize / center-crop every image file under `root` in place."""
    imgs = []
    for ext in (".png", ".jpg", ".jpeg"):
        imgs += glob.glob(os.path.join(root, "*" + ext))

    for img_path in imgs:
        img = Image.open(img_path)
        width, height = img.size
        if size is not None and (width != size[1] or height != size[0]):
            img = img.resize(size, Image.LANCZOS)
        if center_crop is not None:
            new_w, new_h = center_crop
            img = img.crop(
                ((width - new_w) // 2, (height - new_h) // 2,
                 (width + new_w) // 2, (height + new_h) // 2)
            )
        img.save(img_path, img_format)


def random_translation(img, max_pix):
    """Random reflect-padded translation of up to `max_pix` pixels (HWC)."""
    is_pil = not isinstance(img, np.ndarray)
    if is_pil:
        img = np.atleast_3d(np.asarray(img))
    img = np.pad(img, [[max_pix, max_pix], [max_pix, max_pix], [0, 0]], mode="reflect")
    shifts = np.random.randint(-max_pix, max_pix + 1, size=[2])
    processed = np.roll(img, shifts, (0, 1))
    return processed[max_pix:-max_pix, max_pix:-max_pix, :]
