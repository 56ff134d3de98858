"""Memory-mapped indexed token dataset (Megatron .bin/.idx on-disk format).

Reader + writer for the standard ``MMapIndexedDataset`` layout so existing
preprocessed corpora work unchanged (reference dependency: NeMo/Megatron
indexed_dataset used by data/datasets/gpt_dataset_patch.py):

  <prefix>.idx: magic "MMIDIDX\\x00\\x00" | u64 version=1 | u8 dtype_code |
                i64 count | i64 doc_count | i32 sizes[count] |
                i64 pointers[count] | i64 doc_idx[doc_count+? ]
  <prefix>.bin: raw token array
"""

from __future__ import annotations

import struct
from typing import List

import numpy as np

_MAGIC = b"MMIDIDX\x00\x00"

_DTYPES = {
    1: np.uint8, 2: np.int8, 3: np.int16, 4: np.int32,
    5: np.int64, 6: np.float64, 7: np.float32, 8: np.uint16, 9: np.uint32,
}
_DTYPE_CODES = {np.dtype(v): k for k, v in _DTYPES.items()}


class MMapIndexedDataset:
    def __init__(self, prefix: str):
        idx_path = prefix + ".idx"
        bin_path = prefix + ".bin"
        with open(idx_path, "rb") as f:
            magic = f.read(9)
            assert magic == _MAGIC, f"bad index magic in {idx_path}"
            (version,) = struct.unpack("<Q", f.read(8))
            assert version == 1
            (code,) = struct.unpack("<B", f.read(1))
            self.dtype = np.dtype(_DTYPES[code])
            (count,) = struct.unpack("<q", f.read(8))
            (doc_count,) = struct.unpack("<q", f.read(8))
            offset = f.tell()
        buf = np.memmap(idx_path, mode="r")
        self.sizes = np.frombuffer(buf, dtype=np.int32, count=count, offset=offset)
        offset += count * 4
        self.pointers = np.frombuffer(buf, dtype=np.int64, count=count, offset=offset)
        offset += count * 8
        self.doc_idx = np.frombuffer(buf, dtype=np.int64, count=doc_count, offset=offset)
        self.bin = np.memmap(bin_path, dtype=self.dtype, mode="r")

    def __len__(self):
        return len(self.sizes)

    def get(self, doc_id: int, offset: int = 0, length: int = None):
        ptr = self.pointers[doc_id] // self.dtype.itemsize + offset
        if length is None:
            length = self.sizes[doc_id] - offset
        return np.asarray(self.bin[ptr : ptr + length])

    def __getitem__(self, i):
        return self.get(i)


class MMapIndexedDatasetBuilder:
    def __init__(self, prefix: str, dtype=np.int32):
        self.prefix = prefix
        self.dtype = np.dtype(dtype)
        self._bin = open(prefix + ".bin", "wb")
        self.sizes: List[int] = []
        self.doc_idx: List[int] = [0]

    def add_document(self, tokens):
        arr = np.asarray(tokens, dtype=self.dtype)
        self._bin.write(arr.tobytes())
        self.sizes.append(len(arr))
        self.doc_idx.append(len(self.sizes))

    def finalize(self):
        self._bin.close()
        sizes = np.asarray(self.sizes, dtype=np.int32)
        pointers = np.zeros(len(sizes), dtype=np.int64)
        np.cumsum(sizes[:-1] * self.dtype.itemsize, out=pointers[1:])
        with open(self.prefix + ".idx", "wb") as f:
            f.write(_MAGIC)
            f.write(struct.pack("<Q", 1))
            f.write(struct.pack("<B", _DTYPE_CODES[self.dtype]))
            f.write(struct.pack("<q", len(sizes)))
            f.write(struct.pack("<q", len(self.doc_idx)))
            f.write(sizes.tobytes())
            f.write(pointers.tobytes())
            f.write(np.asarray(self.doc_idx, dtype=np.int64).tobytes())
