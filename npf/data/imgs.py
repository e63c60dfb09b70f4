"""Image meta-datasets.

Capability parity with /root/reference/utils/data/imgs.py (DATASETS_DICT
:26-38, MNIST :148-187, SVHN :83-145, ZeroShotMultiMNIST family :191-439,
ExternalDataset/CelebA :442-616, SingleImage :485-505).

This image has neither torchvision nor network egress, so:
- MNIST loads the raw IDX files directly (pure numpy) from `root/MNIST/raw/`.
- SVHN loads the published `.mat` files via scipy.io from `root/SVHN/`.
- CelebA* load a local `img_align_celeba/` folder via PIL.
- All datasets raise a clear `DatasetNotAvailable` when local files are
  absent (no silent downloads), and `SyntheticImages` provides a
  deterministic random-image stand-in with the same interface for benches
  and tests (bench data is synthetic by contract — BASELINE.json).
"""

import abc
import glob
import gzip
import logging
import os
import struct

import numpy as np
import torch
import torch.nn.functional as F
from PIL import Image
from torch.utils.data import Dataset

from npf.train.helpers import set_seed

from .helpers import DIR_DATA, random_translation, train_dev_split

logger = logging.getLogger(__name__)

COLOUR_BLACK = torch.tensor([0.0, 0.0, 0.0])
COLOUR_WHITE = torch.tensor([1.0, 1.0, 1.0])
COLOUR_BLUE = torch.tensor([0.0, 0.0, 1.0])

DATASETS_DICT = {
    "mnist": "MNIST",
    "svhn": "SVHN",
    "celeba32": "CelebA32",
    "celeba64": "CelebA64",
    "zs-multi-mnist": "ZeroShotMultiMNIST",
    "zsmm": "ZeroShotMultiMNIST",
    "zsmmt": "ZeroShotMultiMNISTtrnslt",
    "zsmms": "ZeroShotMultiMNISTscale",
    "zs-mnist": "ZeroShotMNIST",
    "celeba": "CelebA",
    "celeba128": "CelebA128",
    "synthetic32": "SyntheticImages32",
    "synthetic64": "SyntheticImages64",
}
DATASETS = list(DATASETS_DICT.keys())

__all__ = [
    "DATASETS_DICT",
    "DATASETS",
    "DatasetNotAvailable",
    "get_dataset",
    "get_train_test_img_dataset",
    "get_img_size",
    "get_test_upscale_factor",
    "MNIST",
    "SVHN",
    "CelebA32",
    "CelebA64",
    "CelebA128",
    "CelebA",
    "ZeroShotMultiMNIST",
    "ZeroShotMultiMNISTtrnslt",
    "ZeroShotMultiMNISTscale",
    "ZeroShotMNIST",
    "SingleImage",
    "SyntheticImages",
]


class DatasetNotAvailable(RuntimeError):
    """Raised when a dataset's local files are absent (no network egress)."""


def get_dataset(dataset):
    """Name -> dataset class (registry mirror of reference imgs.py:58-65)."""
    try:
        return globals()[DATASETS_DICT[dataset.lower()]]
    except KeyError:
        raise ValueError(f"Unknown dataset: {dataset}")


def get_train_test_img_dataset(dataset):
    """Instantiate (train, test); datasets without native splits get a 90/10
    split (reference imgs.py:45-54)."""
    cls = get_dataset(dataset)
    try:
        return cls(split="train"), cls(split="test")
    except TypeError:
        return train_dev_split(cls(), dev_size=0.1, is_stratify=False)


def get_img_size(dataset):
    return get_dataset(dataset).shape


def get_test_upscale_factor(dataset):
    """shape_test/shape ratio for zero-shot scale extrapolation evals."""
    try:
        cls = get_dataset(dataset)
        return cls.shape_test[-1] / cls.shape[-1]
    except (AttributeError, ValueError):
        return 1


def _to_chw_float(img_hw_or_hwc):
    """uint8 HW / HWC numpy or tensor -> float CHW in [0,1]."""
    t = torch.as_tensor(np.ascontiguousarray(img_hw_or_hwc))
    if t.dim() == 2:
        t = t.unsqueeze(-1)
    t = t.permute(2, 0, 1).float()
    if t.max() > 1.5:
        t = t / 255.0
    return t


def _resize_chw(t, size):
    return F.interpolate(
        t.unsqueeze(0), size=size, mode="bilinear", align_corners=False
    ).squeeze(0)


# --------------------------------------------------------------------------- #
# MNIST (raw IDX loader, no torchvision)
# --------------------------------------------------------------------------- #


def _read_idx(path):
    opener = gzip.open if path.endswith(".gz") else open
    with opener(path, "rb") as f:
        zeros, dtype, ndim = struct.unpack(">HBB", f.read(4))
        shape = struct.unpack(f">{ndim}I", f.read(4 * ndim))
        assert dtype == 8, "only uint8 IDX supported"
        return np.frombuffer(f.read(), dtype=np.uint8).reshape(shape)


def _find_idx(root, names):
    for n in names:
        for cand in (os.path.join(root, n), os.path.join(root, n + ".gz")):
            if os.path.exists(cand):
                return cand
    raise DatasetNotAvailable(
        f"MNIST raw files not found under {root} (no network egress; place "
        f"train-images-idx3-ubyte[.gz] etc. there)."
    )


def _load_mnist_raw(root, train):
    raw = os.path.join(root, "MNIST", "raw")
    prefix = "train" if train else "t10k"
    imgs = _read_idx(_find_idx(raw, [f"{prefix}-images-idx3-ubyte",
                                     f"{prefix}-images.idx3-ubyte"]))
    labels = _read_idx(_find_idx(raw, [f"{prefix}-labels-idx1-ubyte",
                                       f"{prefix}-labels.idx1-ubyte"]))
    return torch.from_numpy(imgs.copy()), labels.copy()


class MNIST(Dataset):
    """MNIST resized to 32x32 (reference imgs.py:148-187)."""

    shape = (1, 32, 32)
    n_classes = 10
    missing_px_color = COLOUR_BLUE
    name = "MNIST"

    def __init__(self, root=DIR_DATA, split="train", **kwargs):
        if split not in ("train", "test"):
            raise ValueError(f"Unknown `split = {split}`")
        self.data, self.targets = _load_mnist_raw(root, split == "train")

    def __len__(self):
        return self.data.size(0)

    def __getitem__(self, idx):
        img = self.data[idx].unsqueeze(0).float() / 255.0
        img = _resize_chw(img, (32, 32))
        return img, int(self.targets[idx])


class SVHN(Dataset):
    """SVHN from the published .mat files (reference imgs.py:83-145)."""

    shape = (3, 32, 32)
    missing_px_color = COLOUR_BLACK
    n_classes = 10
    name = "SVHN"

    def __init__(self, root=DIR_DATA, split="train", **kwargs):
        if split not in ("train", "test", "extra"):
            raise ValueError(f"Unknown `split = {split}`")
        path = os.path.join(root, "SVHN", f"{split}_32x32.mat")
        if not os.path.exists(path):
            raise DatasetNotAvailable(
                f"SVHN file {path} not found (no network egress)."
            )
        import scipy.io as sio

        mat = sio.loadmat(path)
        self.data = np.transpose(mat["X"], (3, 0, 1, 2))  # N,H,W,C
        self.targets = mat["y"].astype(np.int64).squeeze() % 10

    def __len__(self):
        return len(self.data)

    def __getitem__(self, idx):
        return _to_chw_float(self.data[idx]), int(self.targets[idx])


# --------------------------------------------------------------------------- #
# generated zero-shot MNIST variants
# --------------------------------------------------------------------------- #


class ZeroShotMultiMNIST(Dataset):
    """Train: 28px digits centered on a larger black canvas; test: several
    digits shifted around the canvas (reference imgs.py:191-355)."""

    missing_px_color = COLOUR_BLUE
    n_classes = 0
    shape = (1, 56, 56)
    files = {"train": "train", "test": "test"}
    name = "ZeroShotMultiMNIST"

    def __init__(
        self, root=DIR_DATA, split="train", n_test_digits=2, final_size=None,
        seed=123, translation=0, **kwargs,
    ):
        if split not in ("train", "test"):
            raise ValueError(f"Unknown `split = {split}`")
        self.translation = translation
        self.split = split
        self.dir = os.path.join(root, self.name)
        self.n_test_digits = n_test_digits
        self.seed = seed
        self.final_size = final_size
        self._init_size = 28

        saved = os.path.join(
            self.dir, f"{self.files[split]}_seed{seed}_digits{n_test_digits}.pt"
        )
        if os.path.exists(saved):
            self.data = torch.load(saved)
        else:
            os.makedirs(self.dir, exist_ok=True)
            source, _ = _load_mnist_raw(root, split == "train")
            logger.info(f"Generating {self.name} {split} split.")
            if split == "train":
                self.data = self.make_multi_mnist_train(source)
            else:
                self.data = self.make_multi_mnist_test(source)
            torch.save(self.data, saved)

        self.data = self.data.float() / 255
        if self.final_size is not None:
            self.data = F.interpolate(
                self.data.unsqueeze(1).float(), size=self.final_size,
                mode="bilinear", align_corners=True,
            ).squeeze(1)

    def __len__(self):
        return self.data.size(0)

    def make_multi_mnist_train(self, train_dataset):
        """Center digits on a (28*n)^2 black canvas."""
        set_seed(self.seed)
        fin = self._init_size * self.n_test_digits
        init = train_dataset.shape[1:]
        bg = np.zeros((train_dataset.size(0), fin, fin), dtype=np.uint8)
        b = (np.array((fin, fin)) - init) // 2
        bg[:, b[0] : -b[0], b[1] : -b[1]] = train_dataset
        return torch.from_numpy(bg)

    def make_multi_mnist_test(self, test_dataset, varying_axis=None, n_test_digits=None):
        """Shift digits along `varying_axis` and stack along the other
        (both axes mixed when None)."""
        set_seed(self.seed)
        n_test = test_dataset.size(0)
        if n_test_digits is None:
            n_test_digits = self.n_test_digits

        if varying_axis is None:
            out0 = self.make_multi_mnist_test(
                test_dataset[: n_test // 2], varying_axis=0, n_test_digits=n_test_digits
            )
            out1 = self.make_multi_mnist_test(
                test_dataset[: n_test // 2], varying_axis=1, n_test_digits=n_test_digits
            )
            return torch.cat((out0, out1), dim=0)[torch.randperm(n_test)]

        fin = self._init_size * self.n_test_digits
        n_tmp = self.n_test_digits * n_test
        init = test_dataset.shape[1:]
        tmp_size = list(init)
        tmp_size[varying_axis] = fin
        tmp_bg = torch.from_numpy(np.zeros((n_tmp, *tmp_size), dtype=np.uint8))

        max_shift = fin - init[varying_axis]
        shifts = np.random.randint(max_shift, size=n_tmp)
        test_dataset = test_dataset.repeat(self.n_test_digits, 1, 1)[torch.randperm(n_tmp)]

        for i, shift in enumerate(shifts):
            slices = [slice(None), slice(None)]
            slices[varying_axis] = slice(shift, shift + self._init_size)
            tmp_bg[i, slices[0], slices[1]] = test_dataset[i]

        return torch.cat(tmp_bg.split(n_test, 0), dim=1 + 1 - varying_axis)

    def _transform(self, img_hw):
        if self.split == "train" and self.translation:
            # reflect-padded random roll (reference random_translation,
            # utils/data/helpers.py:157-171)
            arr = np.atleast_3d(img_hw.numpy())
            arr = random_translation(arr, self.translation)
            return torch.from_numpy(arr[..., 0].copy()).float().unsqueeze(0)
        return img_hw.unsqueeze(0).float()

    def __getitem__(self, idx):
        return self._transform(self.data[idx]), 0


class ZeroShotMultiMNISTtrnslt(ZeroShotMultiMNIST):
    def __init__(self, *args, **kwargs):
        super().__init__(*args, translation=14, **kwargs)


class ZeroShotMultiMNISTscale(ZeroShotMultiMNIST):
    """Train on plain 32px MNIST, test on 56px multi-digit canvases
    (zero-shot scale extrapolation)."""

    name = "ZeroShotMultiMNISTscale"
    shape = (1, 32, 32)
    shape_test = (1, 56, 56)

    def __init__(self, *args, **kwargs):
        super().__init__(*args, translation=5, **kwargs)
        if self.split == "test":
            self.shape = self.shape_test

    def make_multi_mnist_train(self, train_dataset):
        return train_dataset


class ZeroShotMNIST(ZeroShotMultiMNIST):
    """Single translated digit on the large canvas (reference imgs.py:397-439)."""

    missing_px_color = COLOUR_BLUE
    n_classes = 0
    shape = (1, 56, 56)
    files = {"train": "train", "test": "test"}
    name = "ZeroShotMNIST"

    def make_multi_mnist_test(self, test_dataset, varying_axis=None, n_test_digits=None):
        return super().make_multi_mnist_test(
            test_dataset, varying_axis=varying_axis, n_test_digits=1
        )


# --------------------------------------------------------------------------- #
# external (folder-based) datasets
# --------------------------------------------------------------------------- #


class ExternalDataset(Dataset, abc.ABC):
    """Folder-based dataset; requires the files locally (no downloads here)."""

    def __init__(self, root, **kwargs):
        self.dir = os.path.join(root, self.name)
        self.train_data = os.path.join(self.dir, type(self).files["train"])
        if not os.path.isdir(self.dir):
            raise DatasetNotAvailable(
                f"{type(self).__name__}: expected local data at {self.dir} "
                f"(no network egress in this environment)."
            )

    def __len__(self):
        return len(self.imgs)


class CelebA64(ExternalDataset):
    """CelebA resized to 64x64 from a local `img_align_celeba/` folder
    (reference imgs.py:509-599; download/md5 pipeline requires egress)."""

    files = {"train": "img_align_celeba"}
    shape = (3, 64, 64)
    missing_px_color = COLOUR_BLACK
    n_classes = 0
    name = "celeba64"

    def __init__(self, root=DIR_DATA, **kwargs):
        super().__init__(root, **kwargs)
        self.imgs = sorted(glob.glob(self.train_data + "/*"))

    def __getitem__(self, idx):
        img = Image.open(self.imgs[idx])
        if img.size != (self.shape[2], self.shape[1]):
            img = img.resize((self.shape[2], self.shape[1]), Image.LANCZOS)
        return _to_chw_float(np.asarray(img)), 0


class CelebA32(CelebA64):
    shape = (3, 32, 32)
    name = "celeba32"


class CelebA128(CelebA64):
    shape = (3, 128, 128)
    name = "celeba128"


class CelebA(CelebA64):
    shape = (3, 218, 178)
    name = "celeba"


class SingleImage(Dataset):
    """Dataset of one image (reference imgs.py:485-505)."""

    def __init__(self, img, resize=None, missing_px_color=COLOUR_BLACK):
        self.missing_px_color = missing_px_color
        t = torch.as_tensor(img).float()
        if t.max() > 1.5:
            t = t / 255.0
        if t.dim() == 2:
            t = t.unsqueeze(0)
        elif t.size(-1) in (1, 3):
            t = t.permute(2, 0, 1)
        if resize is not None:
            t = _resize_chw(t, resize)
        self.img = t
        self.shape = tuple(t.shape)

    def __getitem__(self, i):
        return self.img.clone(), 0

    def __len__(self):
        return 1


# --------------------------------------------------------------------------- #
# synthetic stand-in (benches / tests; no local data needed)
# --------------------------------------------------------------------------- #


class SyntheticImages(Dataset):
    """Deterministic random images with the standard (img, label) interface.

    Smooth random fields in [0,1] so context/target completion is non-trivial.
    """

    missing_px_color = COLOUR_BLACK
    n_classes = 0
    name = "synthetic"

    def __init__(self, shape=(3, 32, 32), n_samples=1024, split="train", seed=123):
        self.shape = shape
        g = torch.Generator().manual_seed(seed + (0 if split == "train" else 1))
        c, h, w = shape
        low = torch.rand(n_samples, c, max(h // 4, 1), max(w // 4, 1), generator=g)
        self.data = F.interpolate(low, size=(h, w), mode="bilinear", align_corners=False)
        self.targets = np.zeros(n_samples, dtype=np.int64)

    def __len__(self):
        return self.data.size(0)

    def __getitem__(self, idx):
        return self.data[idx], 0


class SyntheticImages32(SyntheticImages):
    shape = (3, 32, 32)
    name = "synthetic32"

    def __init__(self, split="train", **kw):
        super().__init__(shape=(3, 32, 32), split=split, **kw)


class SyntheticImages64(SyntheticImages):
    shape = (3, 64, 64)
    name = "synthetic64"

    def __init__(self, split="train", **kw):
        super().__init__(shape=(3, 64, 64), split=split, **kw)
