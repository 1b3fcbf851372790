"""Vision datasets.

Reference: VGG/datasets.py DatasetHDF5 (ImageNet packed in HDF5, per-process
chunked reads) and torchvision CIFAR-10 (VGG/dl_trainer.py:286).  Offline
this image has neither torchvision datasets nor h5py, so:

* cifar_like_dataset: deterministic synthetic tensors with CIFAR shapes for
  the training loop (data path parity for the bench contract), or real
  CIFAR-10 python-pickle batches when a downloaded copy exists on disk.
* Hdf5ImagenetDataset: the reference's HDF5 layout, gated on h5py.
"""
from __future__ import annotations

import os
import pickle
from typing import Optional, Tuple

import numpy as np
import torch
from torch.utils.data import Dataset, TensorDataset


def cifar_like_dataset(
    root: Optional[str] = None, train: bool = True, n_synthetic: int = 1024,
    seed: int = 0
) -> Dataset:
    """Real CIFAR-10 if the standard `cifar-10-batches-py` pickles exist
    under `root`, else a seeded synthetic stand-in of identical shape."""
    if root:
        d = os.path.join(root, "cifar-10-batches-py")
        names = (
            [f"data_batch_{i}" for i in range(1, 6)] if train else ["test_batch"]
        )
        if all(os.path.exists(os.path.join(d, n)) for n in names):
            xs, ys = [], []
            for n in names:
                with open(os.path.join(d, n), "rb") as f:
                    entry = pickle.load(f, encoding="latin1")
                xs.append(np.asarray(entry["data"], dtype=np.uint8))
                ys.extend(entry["labels"])
            x = torch.from_numpy(np.concatenate(xs)).view(-1, 3, 32, 32).float() / 255.0
            mean = torch.tensor([0.4914, 0.4822, 0.4465]).view(1, 3, 1, 1)
            std = torch.tensor([0.2023, 0.1994, 0.2010]).view(1, 3, 1, 1)
            x = (x - mean) / std
            return TensorDataset(x, torch.tensor(ys, dtype=torch.long))
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n_synthetic, 3, 32, 32, generator=g)
    y = torch.randint(0, 10, (n_synthetic,), generator=g)
    return TensorDataset(x, y)


def digits_dataset(split: str = "train", image_size: int = 32,
                   channels: int = 3, test_fraction: float = 0.2,
                   seed: int = 0) -> Dataset:
    """REAL image data available offline: the UCI handwritten-digits set
    bundled with scikit-learn (1797 8x8 grayscale images, 10 classes).

    Serves the role of the reference's real CIFAR-10 accuracy runs
    (VGG/dl_trainer.py:286,709-784) in an offline image: images are
    bilinearly upsampled to `image_size` and channel-repeated so the
    CIFAR-shaped model zoo (resnet20/vgg16/caffe_cifar) trains unchanged.
    Deterministic seeded train/test split; per-channel standardization
    from TRAIN statistics only.
    """
    from sklearn.datasets import load_digits

    X, y = load_digits(return_X_y=True)
    x = torch.from_numpy(X).float().view(-1, 1, 8, 8) / 16.0
    y = torch.from_numpy(y).long()
    g = torch.Generator().manual_seed(seed)
    perm = torch.randperm(x.shape[0], generator=g)
    n_test = int(x.shape[0] * test_fraction)
    idx = perm[n_test:] if split == "train" else perm[:n_test]
    x, y = x[idx], y[idx]
    if image_size != 8:
        x = torch.nn.functional.interpolate(
            x, size=(image_size, image_size), mode="bilinear",
            align_corners=False)
    if channels != 1:
        x = x.repeat(1, channels, 1, 1)
    tr_idx = perm[n_test:]
    x_all = torch.from_numpy(X).float().view(-1, 1, 8, 8) / 16.0
    mean = x_all[tr_idx].mean()
    std = x_all[tr_idx].std().clamp_min(1e-6)
    x = (x - mean) / std
    return TensorDataset(x, y)


class Hdf5ImagenetDataset(Dataset):
    """The reference's HDF5-packed ImageNet (VGG/datasets.py DatasetHDF5:
    one 'data'/'label' pair per split, read in chunks)."""

    def __init__(self, path: str, split: str = "train"):
        try:
            import h5py  # noqa: F401
        except ImportError as e:
            raise RuntimeError(
                "Hdf5ImagenetDataset requires h5py (not installed in the "
                "offline image); use cifar_like_dataset or synthetic batches"
            ) from e
        import h5py

        self.f = h5py.File(path, "r")
        self.data = self.f[f"{split}_data"]
        self.labels = self.f[f"{split}_labels"]

    def __len__(self) -> int:
        return self.data.shape[0]

    def __getitem__(self, idx: int) -> Tuple[torch.Tensor, int]:
        x = torch.from_numpy(np.asarray(self.data[idx], dtype=np.float32))
        return x, int(self.labels[idx])
