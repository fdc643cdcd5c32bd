"""Binary indexed dataset — same on-disk format as the reference
(megatron/data/indexed_dataset.py:341-584, fairseq-derived):

  .idx: magic b'MMIDIDX\\x00\\x00' + version + dtype code + counts +
        sizes(int32) + pointers(int64) + doc_idx(int64)
  .bin: raw token payload

Only the mmap variant is implemented (the reference's lazy/cached variants
exist for pre-mmap files; data produced by preprocess tools is always mmap).
"""

from __future__ import annotations

import os
import shutil
import struct
from functools import lru_cache

import numpy as np
import torch

_INDEX_HEADER = b"MMIDIDX\x00\x00"

dtypes = {
    1: np.uint8,
    2: np.int8,
    3: np.int16,
    4: np.int32,
    5: np.int64,
    6: np.float64,
    7: np.float32,
    8: np.uint16,
}


def code(dtype):
    for k in dtypes:
        if dtypes[k] == dtype:
            return k
    raise ValueError(dtype)


def index_file_path(prefix_path):
    return prefix_path + ".idx"


def data_file_path(prefix_path):
    return prefix_path + ".bin"


def infer_dataset_impl(path):
    if not os.path.exists(index_file_path(path)):
        return None
    with open(index_file_path(path), "rb") as f:
        magic = f.read(9)
        if magic == _INDEX_HEADER:
            return "mmap"
    return "cached"


def make_dataset(path, impl="infer", skip_warmup=False):
    if impl == "infer":
        impl = infer_dataset_impl(path)
    if impl in ("mmap", "cached", "lazy", None):
        if not MMapIndexedDataset.exists(path):
            raise FileNotFoundError(
                f"indexed dataset not found: {path}(.idx/.bin)"
            )
        return MMapIndexedDataset(path, skip_warmup)
    raise ValueError(f"unknown dataset impl {impl}")


def dataset_exists(path, impl=None):
    return MMapIndexedDataset.exists(path)


class MMapIndexedDataset(torch.utils.data.Dataset):
    class Index:
        @classmethod
        def writer(cls, path, dtype):
            class _Writer:
                def __enter__(self):
                    self._file = open(path, "wb")
                    self._file.write(_INDEX_HEADER)
                    self._file.write(struct.pack("<Q", 1))
                    self._file.write(struct.pack("<B", code(dtype)))
                    return self

                @staticmethod
                def _get_pointers(sizes):
                    dtype_size = dtype().itemsize
                    address = 0
                    pointers = []
                    for size in sizes:
                        pointers.append(address)
                        address += size * dtype_size
                    return pointers

                def write(self, sizes, doc_idx):
                    pointers = self._get_pointers(sizes)
                    self._file.write(struct.pack("<Q", len(sizes)))
                    self._file.write(struct.pack("<Q", len(doc_idx)))
                    sizes32 = np.array(sizes, dtype=np.int32)
                    self._file.write(sizes32.tobytes(order="C"))
                    pointers64 = np.array(pointers, dtype=np.int64)
                    self._file.write(pointers64.tobytes(order="C"))
                    doc_idx64 = np.array(doc_idx, dtype=np.int64)
                    self._file.write(doc_idx64.tobytes(order="C"))

                def __exit__(self, exc_type, exc_val, exc_tb):
                    self._file.close()

            return _Writer()

        def __init__(self, path, skip_warmup=False):
            with open(path, "rb") as stream:
                magic_test = stream.read(9)
                assert magic_test == _INDEX_HEADER, (
                    "Index file doesn't match expected format."
                )
                version = struct.unpack("<Q", stream.read(8))
                assert version == (1,)
                (dtype_code,) = struct.unpack("<B", stream.read(1))
                self._dtype = dtypes[dtype_code]
                self._dtype_size = self._dtype().itemsize
                self._len = struct.unpack("<Q", stream.read(8))[0]
                self._doc_count = struct.unpack("<Q", stream.read(8))[0]
                offset = stream.tell()

            self._bin_buffer_mmap = np.memmap(path, mode="r", order="C")
            self._bin_buffer = memoryview(self._bin_buffer_mmap)
            self._sizes = np.frombuffer(
                self._bin_buffer, dtype=np.int32, count=self._len, offset=offset
            )
            self._pointers = np.frombuffer(
                self._bin_buffer, dtype=np.int64, count=self._len,
                offset=offset + self._sizes.nbytes,
            )
            self._doc_idx = np.frombuffer(
                self._bin_buffer, dtype=np.int64, count=self._doc_count,
                offset=offset + self._sizes.nbytes + self._pointers.nbytes,
            )

        def __del__(self):
            if hasattr(self, "_bin_buffer_mmap"):
                self._bin_buffer_mmap._mmap.close()
                del self._bin_buffer_mmap

        @property
        def dtype(self):
            return self._dtype

        @property
        def sizes(self):
            return self._sizes

        @property
        def doc_idx(self):
            return self._doc_idx

        @lru_cache(maxsize=8)
        def __getitem__(self, i):
            return self._pointers[i], self._sizes[i]

        def __len__(self):
            return self._len

    def __init__(self, path, skip_warmup=False):
        super().__init__()
        self._path = None
        self._index = None
        self._bin_buffer = None
        self._do_init(path, skip_warmup)

    def __getstate__(self):
        return self._path

    def __setstate__(self, state):
        self._do_init(state, skip_warmup=True)

    def _do_init(self, path, skip_warmup):
        self._path = path
        self._index = self.Index(index_file_path(self._path), skip_warmup)
        self._bin_buffer_mmap = np.memmap(
            data_file_path(self._path), mode="r", order="C"
        )
        self._bin_buffer = memoryview(self._bin_buffer_mmap)

    def __del__(self):
        if hasattr(self, "_bin_buffer_mmap"):
            self._bin_buffer_mmap._mmap.close()
            del self._bin_buffer_mmap
        del self._index

    def __len__(self):
        return len(self._index)

    def __getitem__(self, idx):
        if isinstance(idx, (int, np.integer)):
            ptr, size = self._index[idx]
            np_array = np.frombuffer(
                self._bin_buffer, dtype=self._index.dtype, count=size,
                offset=ptr,
            )
            return np_array
        elif isinstance(idx, slice):
            start, stop, step = idx.indices(len(self))
            if step != 1:
                raise ValueError("Slices into indexed_dataset must be contiguous")
            ptr = self._index._pointers[start]
            sizes = self._index._sizes[idx]
            offsets = list(np.cumsum(sizes))
            total_size = sum(sizes)
            np_array = np.frombuffer(
                self._bin_buffer, dtype=self._index.dtype, count=total_size,
                offset=ptr,
            )
            return np.split(np_array, offsets[:-1])
        raise TypeError(f"unsupported index type {type(idx)}")

    def get(self, idx, offset=0, length=None):
        ptr, size = self._index[idx]
        if length is None:
            length = size - offset
        ptr += offset * np.dtype(self._index.dtype).itemsize
        np_array = np.frombuffer(
            self._bin_buffer, dtype=self._index.dtype, count=length, offset=ptr
        )
        return np_array

    @property
    def sizes(self):
        return self._index.sizes

    @property
    def doc_idx(self):
        return self._index.doc_idx

    def get_doc_idx(self):
        return self._index._doc_idx

    def set_doc_idx(self, doc_idx):
        self._index._doc_idx = doc_idx

    @property
    def supports_prefetch(self):
        return False

    @staticmethod
    def exists(path):
        return os.path.exists(index_file_path(path)) and os.path.exists(
            data_file_path(path)
        )


class MMapIndexedDatasetBuilder:
    def __init__(self, out_file, dtype=np.int64):
        self._data_file = open(out_file, "wb")
        self._dtype = dtype
        self._sizes = []
        self._doc_idx = [0]

    @property
    def dtype(self):
        return self._dtype

    def add_item(self, tensor):
        np_array = np.array(tensor.numpy() if torch.is_tensor(tensor) else tensor,
                            dtype=self._dtype)
        self._data_file.write(np_array.tobytes(order="C"))
        self._sizes.append(np_array.size)

    def end_document(self):
        self._doc_idx.append(len(self._sizes))

    def merge_file_(self, another_file):
        index = MMapIndexedDataset.Index(index_file_path(another_file))
        assert index.dtype == self._dtype
        offset = len(self._sizes)
        self._sizes.extend(index.sizes)
        self._doc_idx.extend((offset + index.doc_idx)[1:])
        with open(data_file_path(another_file), "rb") as f:
            shutil.copyfileobj(f, self._data_file)

    def finalize(self, index_file):
        self._data_file.close()
        with MMapIndexedDataset.Index.writer(index_file, self._dtype) as index:
            index.write(self._sizes, self._doc_idx)


def make_builder(out_file, impl="mmap", dtype=np.int64):
    return MMapIndexedDatasetBuilder(out_file, dtype=dtype)
