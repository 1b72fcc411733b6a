"""paddle.vision.datasets (reference: python/paddle/vision/datasets/).

Folder-backed datasets work from local files; download-backed ones
(MNIST/Cifar/...) accept a local `data_file` and raise a clear error if
asked to download (no network egress in this environment).
"""
from __future__ import annotations

import gzip
import os
import pickle
import struct

import numpy as np

from ..io import Dataset


def _no_download(name, path):
    raise RuntimeError(
        f"{name}: '{path}' not found and downloading is impossible here "
        "(no network egress).  Pass data_file=/path/to/local/archive.")


class DatasetFolder(Dataset):
    """Samples arranged as root/class_x/*.ext (reference: folder.py)."""

    def __init__(self, root, loader=None, extensions=None, transform=None,
                 is_valid_file=None):
        self.root = root
        self.transform = transform
        self.loader = loader or self._default_loader
        exts = extensions or (".jpg", ".jpeg", ".png", ".bmp", ".npy")
        classes = sorted(d for d in os.listdir(root)
                         if os.path.isdir(os.path.join(root, d)))
        self.classes = classes
        self.class_to_idx = {c: i for i, c in enumerate(classes)}
        self.samples = []
        for c in classes:
            cdir = os.path.join(root, c)
            for fn in sorted(os.listdir(cdir)):
                ok = (is_valid_file(fn) if is_valid_file
                      else fn.lower().endswith(exts))
                if ok:
                    self.samples.append((os.path.join(cdir, fn),
                                         self.class_to_idx[c]))

    @staticmethod
    def _default_loader(path):
        if path.endswith(".npy"):
            return np.load(path)
        from . import image_load
        return image_load(path)

    def __getitem__(self, i):
        path, label = self.samples[i]
        img = self.loader(path)
        if self.transform:
            img = self.transform(img)
        return img, label

    def __len__(self):
        return len(self.samples)


class ImageFolder(DatasetFolder):
    """Flat folder of images, no labels (reference: folder.py:ImageFolder)."""

    def __init__(self, root, loader=None, extensions=None, transform=None,
                 is_valid_file=None):
        self.root = root
        self.transform = transform
        self.loader = loader or self._default_loader
        exts = extensions or (".jpg", ".jpeg", ".png", ".bmp", ".npy")
        self.samples = [os.path.join(root, f) for f in sorted(os.listdir(root))
                        if f.lower().endswith(exts)]

    def __getitem__(self, i):
        img = self.loader(self.samples[i])
        if self.transform:
            img = self.transform(img)
        return [img]

    def __len__(self):
        return len(self.samples)


class MNIST(Dataset):
    """IDX-format MNIST from local files (reference: mnist.py)."""

    NAME = "MNIST"

    def __init__(self, image_path=None, label_path=None, mode="train",
                 transform=None, download=True, backend=None):
        self.transform = transform
        if image_path is None or not os.path.exists(image_path):
            _no_download(self.NAME, image_path)
        self.images = self._read_images(image_path)
        self.labels = self._read_labels(label_path)

    @staticmethod
    def _open(path):
        return gzip.open(path, "rb") if path.endswith(".gz") else open(path, "rb")

    def _read_images(self, path):
        with self._open(path) as f:
            magic, n, rows, cols = struct.unpack(">IIII", f.read(16))
            return np.frombuffer(f.read(), dtype=np.uint8).reshape(n, rows, cols)

    def _read_labels(self, path):
        with self._open(path) as f:
            magic, n = struct.unpack(">II", f.read(8))
            return np.frombuffer(f.read(), dtype=np.uint8)

    def __getitem__(self, i):
        img = self.images[i].astype("float32")[None]
        if self.transform:
            img = self.transform(img)
        return img, int(self.labels[i])

    def __len__(self):
        return len(self.images)


class FashionMNIST(MNIST):
    NAME = "FashionMNIST"


class Cifar10(Dataset):
    """CIFAR pickle batches from a local archive dir (reference: cifar.py)."""

    NUM_CLASSES = 10

    def __init__(self, data_file=None, mode="train", transform=None,
                 download=True, backend=None):
        self.transform = transform
        if data_file is None or not os.path.exists(data_file):
            _no_download(type(self).__name__, data_file)
        xs, ys = [], []
        names = ([f"data_batch_{i}" for i in range(1, 6)]
                 if mode == "train" else ["test_batch"])
        if self.NUM_CLASSES == 100:
            names = ["train"] if mode == "train" else ["test"]
        for n in names:
            with open(os.path.join(data_file, n), "rb") as f:
                d = pickle.load(f, encoding="bytes")
            xs.append(d[b"data"])
            ys += list(d.get(b"labels", d.get(b"fine_labels", [])))
        self.data = np.concatenate(xs).reshape(-1, 3, 32, 32)
        self.labels = ys

    def __getitem__(self, i):
        img = self.data[i].astype("float32")
        if self.transform:
            img = self.transform(img)
        return img, int(self.labels[i])

    def __len__(self):
        return len(self.data)


class Cifar100(Cifar10):
    NUM_CLASSES = 100


class Flowers(Dataset):
    def __init__(self, data_file=None, label_file=None, setid_file=None,
                 mode="train", transform=None, download=True, backend=None):
        _no_download("Flowers", data_file)


class VOC2012(Dataset):
    def __init__(self, data_file=None, mode="train", transform=None,
                 download=True, backend=None):
        _no_download("VOC2012", data_file)
