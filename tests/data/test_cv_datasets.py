import os
import pickle
import struct

import numpy as np
import torch

from libai_amd.data.datasets import CIFAR10Dataset, MNISTDataset


def test_cifar10_reader(tmp_path):
    base = tmp_path / "cifar-10-batches-py"
    base.mkdir()
    rng = np.random.RandomState(0)
    for f in [f"data_batch_{i}" for i in range(1, 6)] + ["test_batch"]:
        data = {b"data": rng.randint(0, 255, (10, 3072), dtype=np.uint8),
                b"labels": rng.randint(0, 10, 10).tolist()}
        with open(base / f, "wb") as fh:
            pickle.dump(data, fh)
    ds = CIFAR10Dataset(str(tmp_path), train=True)
    assert len(ds) == 50
    inst = ds[0]
    assert inst.get("images").tensor.shape == (3, 32, 32)
    assert inst.get("images").tensor.dtype == torch.float32


def test_mnist_reader(tmp_path):
    rng = np.random.RandomState(0)
    imgs = rng.randint(0, 255, (6, 28, 28), dtype=np.uint8)
    labels = rng.randint(0, 10, 6, dtype=np.uint8)
    with open(tmp_path / "train-images-idx3-ubyte", "wb") as f:
        f.write(struct.pack(">IIII", 2051, 6, 28, 28))
        f.write(imgs.tobytes())
    with open(tmp_path / "train-labels-idx1-ubyte", "wb") as f:
        f.write(struct.pack(">II", 2049, 6))
        f.write(labels.tobytes())
    ds = MNISTDataset(str(tmp_path), train=True)
    assert len(ds) == 6
    assert ds[2].get("labels").tensor.item() == int(labels[2])
