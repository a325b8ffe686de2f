"""Memory-mapped token dataset (.bin tokens + .idx metadata).

Reference behavior: nemo_automodel/components/datasets/llm/megatron/
indexed_dataset.py (.bin/.idx mmap pair). Own on-disk format:
  .idx: magic b"AMDIDX01" | dtype code u8 | pad[7] | n_docs u64 |
        sizes int32[n_docs] | pointers int64[n_docs]
  .bin: concatenated token arrays.
"""

from __future__ import annotations

import struct

import numpy as np

_MAGIC = b"AMDIDX01"
_DTYPES = {1: np.uint16, 2: np.int32, 3: np.int64}
_DTYPE_CODES = {np.dtype(v): k for k, v in _DTYPES.items()}


class IndexedDatasetWriter:
    def __init__(self, path_prefix: str, dtype=np.int32):
        self.prefix = path_prefix
        self.dtype = np.dtype(dtype)
        self._bin = open(path_prefix + ".bin", "wb")
        self.sizes: list[int] = []
        self.pointers: list[int] = []
        self._offset = 0

    def add_document(self, tokens) -> None:
        arr = np.asarray(tokens, dtype=self.dtype)
        self._bin.write(arr.tobytes())
        self.pointers.append(self._offset)
        self.sizes.append(len(arr))
        self._offset += arr.nbytes

    def finalize(self) -> None:
        self._bin.close()
        with open(self.prefix + ".idx", "wb") as f:
            f.write(_MAGIC)
            f.write(struct.pack("<B7x", _DTYPE_CODES[self.dtype]))
            f.write(struct.pack("<Q", len(self.sizes)))
            f.write(np.asarray(self.sizes, dtype=np.int32).tobytes())
            f.write(np.asarray(self.pointers, dtype=np.int64).tobytes())


class IndexedDataset:
    def __init__(self, path_prefix: str):
        idx_path = path_prefix + ".idx"
        with open(idx_path, "rb") as f:
            assert f.read(8) == _MAGIC, f"bad magic in {idx_path}"
            (code,) = struct.unpack("<B7x", f.read(8))
            (n_docs,) = struct.unpack("<Q", f.read(8))
            self.dtype = np.dtype(_DTYPES[code])
            self.sizes = np.frombuffer(f.read(4 * n_docs), dtype=np.int32)
            self.pointers = np.frombuffer(f.read(8 * n_docs), dtype=np.int64)
        self._data = np.memmap(path_prefix + ".bin", dtype=self.dtype, mode="r")

    def __len__(self) -> int:
        return len(self.sizes)

    def get(self, doc_idx: int, offset: int = 0, length: int | None = None) -> np.ndarray:
        start = self.pointers[doc_idx] // self.dtype.itemsize + offset
        n = (self.sizes[doc_idx] - offset) if length is None else length
        return np.asarray(self._data[start : start + n])

    def __getitem__(self, doc_idx: int) -> np.ndarray:
        return self.get(doc_idx)

    @property
    def total_tokens(self) -> int:
        return int(self.sizes.sum())
