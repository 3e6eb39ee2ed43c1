"""CV dataset wrappers emitting the framework's Instance contract.

Reference behavior: libai/data/datasets/{cifar,imagenet,mnist}.py — thin
wrappers whose job is the sample contract: Instance(images=DistTensorData,
labels=DistTensorData(placement_idx=-1)).  torchvision is not available in
this image, so these read the standard on-disk formats directly (CIFAR/MNIST
binary archives, ImageFolder via PIL).
"""

import os
import pickle
import struct

import numpy as np
import torch

from ..structures import DistTensorData, Instance

__all__ = ["CIFAR10Dataset", "MNISTDataset", "ImageFolderDataset"]

_CIFAR_MEAN = np.array([0.4914, 0.4822, 0.4465], dtype=np.float32)
_CIFAR_STD = np.array([0.2470, 0.2435, 0.2616], dtype=np.float32)


class CIFAR10Dataset(torch.utils.data.Dataset):
    """Reads the python-version CIFAR-10 batches (cifar-10-batches-py)."""

    def __init__(self, root, train=True, transform=None):
        base = os.path.join(root, "cifar-10-batches-py")
        files = (
            [f"data_batch_{i}" for i in range(1, 6)] if train else ["test_batch"]
        )
        xs, ys = [], []
        for f in files:
            with open(os.path.join(base, f), "rb") as fh:
                d = pickle.load(fh, encoding="bytes")
            xs.append(d[b"data"])
            ys.extend(d[b"labels"])
        self.data = np.concatenate(xs).reshape(-1, 3, 32, 32)
        self.labels = np.asarray(ys, dtype=np.int64)
        self.transform = transform

    def __len__(self):
        return len(self.labels)

    def __getitem__(self, i):
        img = self.data[i].astype(np.float32) / 255.0
        img = (img - _CIFAR_MEAN[:, None, None]) / _CIFAR_STD[:, None, None]
        t = torch.from_numpy(img)
        if self.transform is not None:
            t = self.transform(t)
        return Instance(
            images=DistTensorData(t),
            labels=DistTensorData(torch.tensor(self.labels[i]), placement_idx=-1),
        )


class MNISTDataset(torch.utils.data.Dataset):
    """Reads the idx-format MNIST files."""

    def __init__(self, root, train=True, transform=None):
        prefix = "train" if train else "t10k"
        with open(os.path.join(root, f"{prefix}-images-idx3-ubyte"), "rb") as f:
            _, n, rows, cols = struct.unpack(">IIII", f.read(16))
            self.images = np.frombuffer(f.read(), dtype=np.uint8).reshape(
                n, 1, rows, cols
            )
        with open(os.path.join(root, f"{prefix}-labels-idx1-ubyte"), "rb") as f:
            struct.unpack(">II", f.read(8))
            self.labels = np.frombuffer(f.read(), dtype=np.uint8).astype(np.int64)
        self.transform = transform

    def __len__(self):
        return len(self.labels)

    def __getitem__(self, i):
        img = (self.images[i].astype(np.float32) / 255.0 - 0.1307) / 0.3081
        t = torch.from_numpy(img)
        if self.transform is not None:
            t = self.transform(t)
        return Instance(
            images=DistTensorData(t),
            labels=DistTensorData(torch.tensor(self.labels[i]), placement_idx=-1),
        )


class ImageFolderDataset(torch.utils.data.Dataset):
    """class-per-subdirectory image tree read via PIL; resize+center-crop to
    `img_size`, ImageNet normalization."""

    MEAN = np.array([0.485, 0.456, 0.406], dtype=np.float32)
    STD = np.array([0.229, 0.224, 0.225], dtype=np.float32)

    def __init__(self, root, img_size=224, transform=None):
        self.root = root
        self.img_size = img_size
        self.transform = transform
        classes = sorted(
            d for d in os.listdir(root) if os.path.isdir(os.path.join(root, d))
        )
        self.class_to_idx = {c: i for i, c in enumerate(classes)}
        self.samples = []
        for c in classes:
            cdir = os.path.join(root, c)
            for f in sorted(os.listdir(cdir)):
                if f.lower().endswith((".jpg", ".jpeg", ".png", ".bmp")):
                    self.samples.append((os.path.join(cdir, f), self.class_to_idx[c]))

    def __len__(self):
        return len(self.samples)

    def __getitem__(self, i):
        from PIL import Image

        path, label = self.samples[i]
        img = Image.open(path).convert("RGB")
        s = self.img_size
        w, h = img.size
        scale = s / min(w, h)
        img = img.resize((max(s, int(round(w * scale))),
                          max(s, int(round(h * scale)))))
        w, h = img.size
        left, top = (w - s) // 2, (h - s) // 2
        img = img.crop((left, top, left + s, top + s))
        arr = np.asarray(img, dtype=np.float32).transpose(2, 0, 1) / 255.0
        arr = (arr - self.MEAN[:, None, None]) / self.STD[:, None, None]
        t = torch.from_numpy(arr)
        if self.transform is not None:
            t = self.transform(t)
        return Instance(
            images=DistTensorData(t),
            labels=DistTensorData(torch.tensor(label), placement_idx=-1),
        )
