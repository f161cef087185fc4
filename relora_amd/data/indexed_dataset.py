"""Binary token storage: fairseq/Megatron-compatible ``.idx`` + ``.bin`` pairs.

On-disk formats are bit-compatible with the reference
(peft_pretraining/megatron_dataset/indexed_dataset.py) so existing
pretokenized corpora load unchanged:

* ``MMapIndexedDataset`` — magic ``MMIDIDX\\x00\\x00``; header = version u64,
  dtype-code u8, sequence count u64, document count u64; then int32 sizes,
  int64 byte pointers, int64 doc_idx.  Reads are zero-copy
  ``np.frombuffer`` views over one shared mmap (reference :348-565).
* legacy ``IndexedDataset`` — magic ``TNTIDX\\x00\\x00`` with int64
  dim-offset/data-offset tables and seek/read access (reference :133-221).

Builders write both formats; ``merge_file_`` appends a finished shard.
"""

import os
import shutil
import struct

import numpy as np
import torch

_MMAP_MAGIC = b"MMIDIDX\x00\x00"
_LEGACY_MAGIC = b"TNTIDX\x00\x00"

# dtype codes shared with the reference format (indexed_dataset.py:98-107)
DTYPES = {
    1: np.uint8,
    2: np.int8,
    3: np.int16,
    4: np.int32,
    5: np.int64,
    6: np.float32,
    7: np.float64,
    8: np.uint16,
}


def dtype_code(dtype):
    for k, v in DTYPES.items():
        if v == dtype:
            return k
    raise ValueError(f"unsupported dtype {dtype}")


def index_file_path(prefix):
    return prefix + ".idx"


def data_file_path(prefix):
    return prefix + ".bin"


def best_fitting_dtype(vocab_size=None):
    """uint16 when the vocab fits (reference :28-32), else int32."""
    if vocab_size is not None and vocab_size < 65500:
        return np.uint16
    return np.int32


def create_doc_idx(sizes):
    """Document boundaries from a sizes list where 0-length sentinels split docs."""
    doc_idx = [0]
    for i, s in enumerate(sizes):
        if s == 0:
            doc_idx.append(i + 1)
    return doc_idx


def exists(path):
    return os.path.exists(index_file_path(path)) and os.path.exists(data_file_path(path))


def infer_dataset_impl(path):
    if not exists(path):
        return None
    with open(index_file_path(path), "rb") as f:
        magic = f.read(8)
    if magic == _LEGACY_MAGIC:
        return "cached"
    if magic == _MMAP_MAGIC[:8]:
        return "mmap"
    return None


def make_dataset(path, impl, skip_warmup=False):
    """Open a dataset by impl name ('mmap' | 'lazy' | 'cached' | 'infer')."""
    if not exists(path):
        raise FileNotFoundError(
            f"indexed dataset not found: {path} (.idx/.bin pair expected)")
    if impl == "infer":
        impl = infer_dataset_impl(path)
    if impl == "mmap":
        return MMapIndexedDataset(path, skip_warmup=skip_warmup)
    if impl in ("lazy", "cached"):
        return IndexedDataset(path)
    raise ValueError(f"unknown dataset impl: {impl}")


def make_builder(out_file, impl, vocab_size=None):
    if impl == "mmap":
        return MMapIndexedDatasetBuilder(out_file, dtype=best_fitting_dtype(vocab_size))
    return IndexedDatasetBuilder(out_file)


# ---------------------------------------------------------------------------
# mmap implementation (the production path)
# ---------------------------------------------------------------------------


class _MMapIndex:
    """Reader/writer of the ``.idx`` sidecar for MMapIndexedDataset."""

    def __init__(self, path, skip_warmup=False):
        with open(path, "rb") as f:
            magic = f.read(9)
            if magic != _MMAP_MAGIC:
                raise ValueError(f"{path}: bad magic {magic!r}, not an MMIDIDX index")
            (version,) = struct.unpack("<Q", f.read(8))
            if version != 1:
                raise ValueError(f"{path}: unsupported index version {version}")
            (code,) = struct.unpack("<B", f.read(1))
            self.dtype = DTYPES[code]
            (self._len,) = struct.unpack("<Q", f.read(8))
            (self._doc_count,) = struct.unpack("<Q", f.read(8))
            offset = f.tell()

        if not skip_warmup:
            _warmup(path)
        self._mmap = np.memmap(path, mode="r", order="C")
        buf = memoryview(self._mmap)
        self.sizes = np.frombuffer(buf, dtype=np.int32, count=self._len, offset=offset)
        offset += self.sizes.nbytes
        self.pointers = np.frombuffer(buf, dtype=np.int64, count=self._len, offset=offset)
        offset += self.pointers.nbytes
        self.doc_idx = np.frombuffer(buf, dtype=np.int64, count=self._doc_count, offset=offset)

    def __len__(self):
        return self._len

    @staticmethod
    def write(path, dtype, sizes, doc_idx):
        itemsize = dtype().itemsize
        pointers = np.zeros(len(sizes), dtype=np.int64)
        if len(sizes):
            np.cumsum(np.asarray(sizes[:-1], dtype=np.int64) * itemsize, out=pointers[1:])
        with open(path, "wb") as f:
            f.write(_MMAP_MAGIC)
            f.write(struct.pack("<Q", 1))
            f.write(struct.pack("<B", dtype_code(dtype)))
            f.write(struct.pack("<Q", len(sizes)))
            f.write(struct.pack("<Q", len(doc_idx)))
            f.write(np.asarray(sizes, dtype=np.int32).tobytes(order="C"))
            f.write(pointers.tobytes(order="C"))
            f.write(np.asarray(doc_idx, dtype=np.int64).tobytes(order="C"))


def _warmup(path):
    """Touch the file sequentially so later random mmap reads hit page cache."""
    with open(path, "rb") as f:
        while f.read(32 * 1024 * 1024):
            pass


class MMapIndexedDataset(torch.utils.data.Dataset):
    def __init__(self, path, skip_warmup=False):
        super().__init__()
        self._path = path
        self._index = _MMapIndex(index_file_path(path), skip_warmup=skip_warmup)
        if not skip_warmup:
            _warmup(data_file_path(path))
        self._bin = np.memmap(data_file_path(path), mode="r", order="C")
        self._buf = memoryview(self._bin)

    def __len__(self):
        return len(self._index)

    def __getstate__(self):
        return self._path

    def __setstate__(self, path):
        self.__init__(path, skip_warmup=True)

    def __getitem__(self, idx):
        if isinstance(idx, slice):
            start, stop, step = idx.indices(len(self))
            if step != 1:
                raise ValueError("slices must be contiguous")
            sizes = self._index.sizes[start:stop]
            arr = np.frombuffer(
                self._buf, dtype=self._index.dtype,
                count=int(sizes.sum()),
                offset=int(self._index.pointers[start]))
            return np.split(arr, np.cumsum(sizes)[:-1])
        ptr = int(self._index.pointers[idx])
        size = int(self._index.sizes[idx])
        return np.frombuffer(self._buf, dtype=self._index.dtype, count=size, offset=ptr)

    def get(self, idx, offset=0, length=None):
        """Read ``length`` tokens of sequence ``idx`` starting at ``offset``
        as a zero-copy view (reference :528-541)."""
        ptr = int(self._index.pointers[idx])
        size = int(self._index.sizes[idx])
        if length is None:
            length = size - offset
        ptr += offset * self._index.dtype().itemsize
        return np.frombuffer(self._buf, dtype=self._index.dtype, count=length, offset=ptr)

    @property
    def sizes(self):
        return self._index.sizes

    @property
    def doc_idx(self):
        return self._index.doc_idx

    def get_doc_idx(self):
        return self._index.doc_idx

    def set_doc_idx(self, doc_idx):
        self._index.doc_idx = doc_idx

    @property
    def supports_prefetch(self):
        return False

    @staticmethod
    def exists(path):
        return exists(path)


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
        arr = np.asarray(tensor.numpy() if torch.is_tensor(tensor) else tensor,
                         dtype=self._dtype)
        self._data_file.write(arr.tobytes(order="C"))
        self._sizes.append(arr.size)

    def end_document(self):
        self._doc_idx.append(len(self._sizes))

    def merge_file_(self, another_file):
        index = _MMapIndex(index_file_path(another_file), skip_warmup=True)
        if index.dtype != self._dtype:
            raise ValueError("dtype mismatch while merging shards")
        base = len(self._sizes)
        self._sizes.extend(int(s) for s in index.sizes)
        self._doc_idx.extend(base + int(d) for d in index.doc_idx[1:])
        with open(data_file_path(another_file), "rb") as f:
            shutil.copyfileobj(f, self._data_file)

    def finalize(self, index_file):
        self._data_file.close()
        _MMapIndex.write(index_file, self._dtype, self._sizes, self._doc_idx)


# ---------------------------------------------------------------------------
# legacy TNTIDX implementation (read + build, for old corpora)
# ---------------------------------------------------------------------------


class IndexedDataset(torch.utils.data.Dataset):
    _HDR_MAGIC = _LEGACY_MAGIC

    def __init__(self, path):
        super().__init__()
        self._path = path
        with open(index_file_path(path), "rb") as f:
            magic = f.read(8)
            if magic != _LEGACY_MAGIC:
                raise ValueError(f"{path}: not a TNTIDX index")
            (version,) = struct.unpack("<Q", f.read(8))
            if version != 1:
                raise ValueError(f"unsupported version {version}")
            code, self.element_size = struct.unpack("<QQ", f.read(16))
            self.dtype = DTYPES[code]
            self._len, self.s = struct.unpack("<QQ", f.read(16))
            (self.doc_count,) = struct.unpack("<Q", f.read(8))
            self.dim_offsets = np.fromfile(f, dtype=np.int64, count=self._len + 1)
            self.data_offsets = np.fromfile(f, dtype=np.int64, count=self._len + 1)
            self.sizes_all = np.fromfile(f, dtype=np.int64, count=self.s)
            self.doc_idx = np.fromfile(f, dtype=np.int64, count=self.doc_count)
        self._data = open(data_file_path(path), "rb", buffering=0)

    def __len__(self):
        return self._len

    def __del__(self):
        if getattr(self, "_data", None) is not None:
            self._data.close()

    def _read(self, start_el, count):
        out = np.empty(count, dtype=self.dtype)
        self._data.seek(start_el * self.element_size)
        self._data.readinto(out)
        return out

    def __getitem__(self, idx):
        if isinstance(idx, slice):
            raise TypeError("legacy IndexedDataset does not support slicing")
        start = self.data_offsets[idx]
        count = self.data_offsets[idx + 1] - start
        return self._read(int(start), int(count))

    def get(self, idx, offset=0, length=None):
        start = int(self.data_offsets[idx])
        size = int(self.data_offsets[idx + 1]) - start
        if length is None:
            length = size - offset
        return self._read(start + offset, length)

    @property
    def sizes(self):
        return self.sizes_all

    @staticmethod
    def exists(path):
        return exists(path)


class IndexedDatasetBuilder:
    element_sizes = {
        np.uint8: 1, np.int8: 1, np.int16: 2, np.int32: 4,
        np.int64: 8, np.float32: 4, np.float64: 8, np.uint16: 2,
    }

    def __init__(self, out_file, dtype=np.int32):
        self.out_file = open(out_file, "wb")
        self.dtype = dtype
        self.data_offsets = [0]
        self.dim_offsets = [0]
        self.sizes = []
        self.doc_idx = [0]
        self.element_size = self.element_sizes[dtype]

    def add_item(self, tensor):
        arr = np.asarray(tensor.numpy() if torch.is_tensor(tensor) else tensor,
                         dtype=self.dtype)
        nbytes = self.out_file.write(arr.tobytes(order="C"))
        self.data_offsets.append(self.data_offsets[-1] + nbytes // self.element_size)
        self.sizes.extend(arr.shape)
        self.dim_offsets.append(self.dim_offsets[-1] + len(arr.shape))

    def end_document(self):
        self.doc_idx.append(len(self.sizes))

    def finalize(self, index_file):
        self.out_file.close()
        with open(index_file, "wb") as f:
            f.write(_LEGACY_MAGIC)
            f.write(struct.pack("<Q", 1))
            f.write(struct.pack("<QQ", dtype_code(self.dtype), self.element_size))
            f.write(struct.pack("<QQ", len(self.data_offsets) - 1, len(self.sizes)))
            f.write(struct.pack("<Q", len(self.doc_idx)))
            np.asarray(self.dim_offsets, dtype=np.int64).tofile(f)
            np.asarray(self.data_offsets, dtype=np.int64).tofile(f)
            np.asarray(self.sizes, dtype=np.int64).tofile(f)
            np.asarray(self.doc_idx, dtype=np.int64).tofile(f)
