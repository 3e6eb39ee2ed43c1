"""Megatron-style mmap indexed dataset (.bin token store + .idx index).

Reference behavior: libai/data/data_utils/indexed_dataset.py:28-603 —
same on-disk format (MMIDIDX magic) so corpora preprocessed by either
framework interchange.  dtype is chosen by vocab size.
"""

import os
import struct

import numpy as np
import torch

__all__ = ["MMapIndexedDataset", "MMapIndexedDatasetBuilder", "make_dataset",
           "best_fitting_dtype", "data_file_path", "index_file_path"]

_INDEX_MAGIC = b"MMIDIDX\x00\x00"

_DTYPES = {
    1: np.uint8,
    2: np.int8,
    3: np.int16,
    4: np.int32,
    5: np.int64,
    6: np.float64,
    7: np.float32,
    8: np.uint16,
}
_DTYPE_CODES = {np.dtype(v): k for k, v in _DTYPES.items()}


def best_fitting_dtype(vocab_size=None):
    if vocab_size is not None and vocab_size < 65500:
        return np.uint16
    return np.int32


def data_file_path(prefix):
    return prefix + ".bin"


def index_file_path(prefix):
    return prefix + ".idx"


class _Index:
    def __init__(self, path):
        with open(path, "rb") as f:
            magic = f.read(9)
            assert magic == _INDEX_MAGIC, f"bad index magic in {path}"
            (version,) = struct.unpack("<Q", f.read(8))
            assert version == 1
            (code,) = struct.unpack("<B", f.read(1))
            self.dtype = np.dtype(_DTYPES[code])
            (self._len,) = struct.unpack("<Q", f.read(8))
            (self._doc_count,) = struct.unpack("<Q", f.read(8))
            offset = f.tell()
        buf = np.memmap(path, mode="r", order="C")
        self.sizes = np.frombuffer(buf, dtype=np.int32, count=self._len,
                                   offset=offset)
        offset += self.sizes.nbytes
        self.pointers = np.frombuffer(buf, dtype=np.int64, count=self._len,
                                      offset=offset)
        offset += self.pointers.nbytes
        self.doc_idx = np.frombuffer(buf, dtype=np.int64, count=self._doc_count,
                                     offset=offset)

    def __len__(self):
        return self._len

    @staticmethod
    def write(path, sizes, doc_idx, dtype):
        with open(path, "wb") as f:
            f.write(_INDEX_MAGIC)
            f.write(struct.pack("<Q", 1))
            f.write(struct.pack("<B", _DTYPE_CODES[np.dtype(dtype)]))
            f.write(struct.pack("<Q", len(sizes)))
            f.write(struct.pack("<Q", len(doc_idx)))
            sizes32 = np.array(sizes, dtype=np.int32)
            f.write(sizes32.tobytes(order="C"))
            pointers = np.zeros(len(sizes), dtype=np.int64)
            itemsize = np.dtype(dtype).itemsize
            if len(sizes) > 1:
                np.cumsum(sizes32[:-1].astype(np.int64) * itemsize,
                          out=pointers[1:])
            f.write(pointers.tobytes(order="C"))
            f.write(np.array(doc_idx, dtype=np.int64).tobytes(order="C"))


class MMapIndexedDataset(torch.utils.data.Dataset):
    def __init__(self, path):
        self._path = path
        self._index = _Index(index_file_path(path))
        self._bin = np.memmap(data_file_path(path), mode="r", order="C")

    def __len__(self):
        return len(self._index)

    def __getitem__(self, idx):
        if isinstance(idx, slice):
            return [self[i] for i in range(*idx.indices(len(self)))]
        ptr = self._index.pointers[idx]
        size = self._index.sizes[idx]
        return np.frombuffer(self._bin, dtype=self._index.dtype, count=size,
                             offset=ptr)

    def get(self, idx, offset=0, length=None):
        ptr = self._index.pointers[idx] + offset * self._index.dtype.itemsize
        size = self._index.sizes[idx] - offset
        if length is not None:
            size = min(size, length)
        return np.frombuffer(self._bin, dtype=self._index.dtype, count=size,
                             offset=ptr)

    @property
    def sizes(self):
        return self._index.sizes

    @property
    def doc_idx(self):
        return self._index.doc_idx

    @property
    def dtype(self):
        return self._index.dtype

    @staticmethod
    def exists(path):
        return os.path.exists(index_file_path(path)) and os.path.exists(
            data_file_path(path)
        )


class MMapIndexedDatasetBuilder:
    def __init__(self, out_file, dtype=np.int32):
        self._data_file = open(out_file, "wb")
        self._dtype = np.dtype(dtype)
        self._sizes = []
        self._doc_idx = [0]

    def add_item(self, tensor):
        arr = np.asarray(tensor, dtype=self._dtype)
        self._data_file.write(arr.tobytes(order="C"))
        self._sizes.append(len(arr))

    def end_document(self):
        self._doc_idx.append(len(self._sizes))

    def merge_file_(self, another_prefix):
        other = MMapIndexedDataset(another_prefix)
        assert other.dtype == self._dtype
        base = len(self._sizes)
        for i in range(len(other)):
            self.add_item(other[i])
        for d in other.doc_idx[1:]:
            self._doc_idx.append(base + int(d))

    def finalize(self, index_file):
        self._data_file.close()
        _Index.write(index_file, self._sizes, self._doc_idx, self._dtype)


def make_dataset(path, impl="mmap", skip_warmup=False):
    assert impl == "mmap", "only the mmap indexed dataset is supported"
    return MMapIndexedDataset(path)
