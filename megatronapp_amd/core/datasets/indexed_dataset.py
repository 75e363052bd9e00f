"""Memory-mapped .bin/.idx token dataset — byte-compatible with the
reference's IndexedDataset format (core/datasets/indexed_dataset.py, 857
LoC) so corpora tokenized by either toolchain interchange.

Format:
  <path>.idx : _INDEX_HEADER  b"MMIDIDX\\x00\\x00"
               <u64 version=1> <u8 dtype_code> <u64 sequence_count>
               <u64 document_count>
               sequence_lengths  i32[sequence_count]
               sequence_pointers u64[sequence_count]   (byte offsets)
               document_indices  u64[document_count]
  <path>.bin : raw token values (dtype per code)
"""

from __future__ import annotations

import os
import struct
from typing import List, Optional

import numpy as np
import torch

_INDEX_HEADER = b"MMIDIDX\x00\x00"

DTYPES = {
    1: np.uint8, 2: np.int8, 3: np.int16, 4: np.int32, 5: np.int64,
    6: np.float64, 7: np.float32, 8: np.uint16,
}
DTYPE_CODES = {v: k for k, v in DTYPES.items()}


def _index_path(prefix: str) -> str:
    return prefix + ".idx"


def _bin_path(prefix: str) -> str:
    return prefix + ".bin"


class IndexedDatasetBuilder:
    def __init__(self, path_prefix: str, dtype=np.int32):
        self.path_prefix = path_prefix
        self.dtype = np.dtype(dtype).type
        self._bin = open(_bin_path(path_prefix), "wb")
        self.sequence_lengths: List[int] = []
        self.document_indices: List[int] = [0]

    def add_item(self, tensor) -> None:
        arr = np.asarray(tensor, dtype=self.dtype)
        self._bin.write(arr.tobytes(order="C"))
        self.sequence_lengths.append(len(arr))

    def add_document(self, tensor, lengths: List[int]) -> None:
        arr = np.asarray(tensor, dtype=self.dtype)
        self._bin.write(arr.tobytes(order="C"))
        self.sequence_lengths.extend(lengths)
        self.document_indices.append(len(self.sequence_lengths))

    def end_document(self) -> None:
        self.document_indices.append(len(self.sequence_lengths))

    def finalize(self) -> None:
        self._bin.close()
        itemsize = np.dtype(self.dtype).itemsize
        with open(_index_path(self.path_prefix), "wb") as f:
            f.write(_INDEX_HEADER)
            f.write(struct.pack("<Q", 1))
            f.write(struct.pack("<B", DTYPE_CODES[self.dtype]))
            f.write(struct.pack("<Q", len(self.sequence_lengths)))
            f.write(struct.pack("<Q", len(self.document_indices)))
            lengths = np.array(self.sequence_lengths, dtype=np.int32)
            pointers = np.zeros(len(lengths), dtype=np.int64)
            if len(lengths) > 1:
                np.cumsum(lengths[:-1] * itemsize, out=pointers[1:])
            f.write(lengths.tobytes(order="C"))
            f.write(pointers.tobytes(order="C"))
            f.write(np.array(self.document_indices,
                             dtype=np.int64).tobytes(order="C"))


class IndexedDataset(torch.utils.data.Dataset):
    def __init__(self, path_prefix: str, mmap: bool = True):
        self.path_prefix = path_prefix
        with open(_index_path(path_prefix), "rb") as f:
            header = f.read(len(_INDEX_HEADER))
            assert header == _INDEX_HEADER, f"bad index header in {path_prefix}"
            (version,) = struct.unpack("<Q", f.read(8))
            assert version == 1
            (code,) = struct.unpack("<B", f.read(1))
            self.dtype = DTYPES[code]
            (self.sequence_count,) = struct.unpack("<Q", f.read(8))
            (self.document_count,) = struct.unpack("<Q", f.read(8))
            offset = f.tell()
        buf = np.memmap(_index_path(path_prefix), mode="r", offset=offset)
        n = self.sequence_count
        self.sequence_lengths = np.frombuffer(buf, dtype=np.int32, count=n)
        self.sequence_pointers = np.frombuffer(
            buf, dtype=np.int64, count=n, offset=n * 4)
        self.document_indices = np.frombuffer(
            buf, dtype=np.int64, count=self.document_count, offset=n * 12)
        self.bin = np.memmap(_bin_path(path_prefix), dtype=self.dtype,
                             mode="r")

    def __len__(self):
        return self.sequence_count

    def get(self, idx: int, offset: int = 0, length: Optional[int] = None):
        ptr = self.sequence_pointers[idx] // np.dtype(self.dtype).itemsize
        ln = self.sequence_lengths[idx] - offset
        if length is not None:
            ln = min(ln, length)
        return np.asarray(self.bin[ptr + offset:ptr + offset + ln])

    def __getitem__(self, idx):
        return self.get(idx)
