"""Binary token storage: MMapIndexedDataset + builder.

Reference behavior: paddlenlp/data/indexed_dataset.py:360 (MMapIndexedDataset
with struct-packed .idx header + .bin payload) — the standard Megatron-LM
mmap format, kept bit-compatible so corpora preprocessed for the reference
load directly:

  .idx: magic b"MMIDIDX\\x00\\x00" | u64 version=1 | u8 dtype_code |
        u64 n_sequences | u64 n_docs | i32 sizes[n_seq] |
        i64 pointers[n_seq] | i64 doc_idx[n_docs]
  .bin: raw token array
"""
from __future__ import annotations

import struct
from typing import List, Optional

import numpy as np

_INDEX_MAGIC = b"MMIDIDX\x00\x00"

DTYPES = {
    1: np.uint8,
    2: np.int8,
    3: np.int16,
    4: np.int32,
    5: np.int64,
    6: np.float64,
    7: np.float32,
    8: np.uint16,
}
DTYPE_CODES = {np.dtype(v): k for k, v in DTYPES.items()}


def data_file_path(prefix):
    return prefix + ".bin"


def index_file_path(prefix):
    return prefix + ".idx"


class MMapIndexedDataset:
    class Index:
        def __init__(self, path: str):
            with open(path, "rb") as f:
                magic = f.read(9)
                assert magic == _INDEX_MAGIC, f"bad index magic in {path}"
                (version,) = struct.unpack("<Q", f.read(8))
                assert version == 1
                (dtype_code,) = struct.unpack("<B", f.read(1))
                self.dtype = DTYPES[dtype_code]
                (self._len,) = struct.unpack("<Q", f.read(8))
                (self._doc_count,) = struct.unpack("<Q", f.read(8))
                offset = f.tell()
            self._buffer = np.memmap(path, mode="r")
            self.sizes = np.frombuffer(self._buffer, dtype=np.int32,
                                       count=self._len, offset=offset)
            offset += self.sizes.nbytes
            self.pointers = np.frombuffer(self._buffer, dtype=np.int64,
                                          count=self._len, offset=offset)
            offset += self.pointers.nbytes
            self.doc_idx = np.frombuffer(self._buffer, dtype=np.int64,
                                         count=self._doc_count, offset=offset)

        def __len__(self):
            return self._len

    def __init__(self, path_prefix: str):
        self._path = path_prefix
        self._index = self.Index(index_file_path(path_prefix))
        self._bin_buffer = np.memmap(data_file_path(path_prefix), mode="r")

    def __len__(self):
        return len(self._index)

    @property
    def sizes(self):
        return self._index.sizes

    @property
    def doc_idx(self):
        return self._index.doc_idx

    @property
    def dtype(self):
        return self._index.dtype

    def get(self, idx: int, offset: int = 0, length: Optional[int] = None) -> np.ndarray:
        size = self._index.sizes[idx]
        ptr = self._index.pointers[idx]
        if length is None:
            length = size - offset
        itemsize = np.dtype(self._index.dtype).itemsize
        return np.frombuffer(
            self._bin_buffer, dtype=self._index.dtype, count=length,
            offset=ptr + offset * itemsize,
        )

    def __getitem__(self, idx):
        return self.get(idx)


class MMapIndexedDatasetBuilder:
    """Streaming writer used by the preprocessing tools."""

    def __init__(self, out_prefix: str, dtype=np.uint16):
        self._prefix = out_prefix
        self._dtype = np.dtype(dtype)
        self._bin = open(data_file_path(out_prefix), "wb")
        self._sizes: List[int] = []
        self._pointers: List[int] = []
        self._doc_idx: List[int] = [0]
        self._offset = 0

    def add_item(self, tokens: np.ndarray):
        arr = np.asarray(tokens, dtype=self._dtype)
        self._bin.write(arr.tobytes(order="C"))
        self._pointers.append(self._offset)
        self._sizes.append(len(arr))
        self._offset += arr.nbytes

    def end_document(self):
        self._doc_idx.append(len(self._sizes))

    def finalize(self):
        self._bin.close()
        with open(index_file_path(self._prefix), "wb") as f:
            f.write(_INDEX_MAGIC)
            f.write(struct.pack("<Q", 1))
            f.write(struct.pack("<B", DTYPE_CODES[self._dtype]))
            f.write(struct.pack("<Q", len(self._sizes)))
            f.write(struct.pack("<Q", len(self._doc_idx)))
            f.write(np.asarray(self._sizes, dtype=np.int32).tobytes(order="C"))
            f.write(np.asarray(self._pointers, dtype=np.int64).tobytes(order="C"))
            f.write(np.asarray(self._doc_idx, dtype=np.int64).tobytes(order="C"))


def make_indexed_dataset(path_prefix: str) -> MMapIndexedDataset:
    return MMapIndexedDataset(path_prefix)
