"""Megatron-style pretraining data pipeline (thin, faithful):

  - MMapIndexedDataset / MMapIndexedDatasetBuilder: the reference's on-disk
    .bin/.idx format (data/megatron/indexed_dataset.py — header MMIDIDX,
    version 1, dtype code, int32 sequence lengths, int64 pointers, int64
    document indices). Byte-compatible with files the reference writes.
  - GPTDataset: document/sample/shuffle indices
    (data/megatron/gpt_dataset.py:241-401) with the sample index built by
    the native C-ABI builder (csrc/data_index.cpp, the reference's
    helpers.cpp:74-225 equivalent).
  - BlendedDataset: greedy max-error dataset interleave (helpers.cpp:17-69).
  - MegatronBatchSampler-lite: rank-sharded sequential sampling with
    consumed_samples resume (data/megatron/sampler.py).

Samples are {"text": int64 (seq_length+1,)} windows crossing document
boundaries, exactly what ModelWrapperForPretraining.forward consumes.
"""

import ctypes
import struct
from pathlib import Path

import numpy
import torch

from .ops import hip
from .utils import get_rank, get_world_size

_INDEX_HEADER = b"MMIDIDX\x00\x00"

_DTYPE_CODES = {
    numpy.uint8: 1,
    numpy.int8: 2,
    numpy.int16: 3,
    numpy.int32: 4,
    numpy.int64: 5,
    numpy.float64: 6,
    numpy.float32: 7,
    numpy.uint16: 8,
}
_CODE_DTYPES = {v: k for k, v in _DTYPE_CODES.items()}


class MMapIndexedDatasetBuilder:
    """Writer for the .bin/.idx pair (indexed_dataset.py:103-186)."""

    def __init__(self, path_prefix: str, dtype=numpy.int32):
        self.path_prefix = str(path_prefix)
        self.dtype = dtype
        self._bin = open(self.path_prefix + ".bin", "wb")
        self.sequence_lengths = []
        self.document_indices = [0]

    def add_document(self, tokens) -> None:
        arr = numpy.asarray(tokens, dtype=self.dtype)
        self._bin.write(arr.tobytes(order="C"))
        self.sequence_lengths.append(len(arr))
        self.document_indices.append(len(self.sequence_lengths))

    def finalize(self) -> None:
        self._bin.close()
        with open(self.path_prefix + ".idx", "wb") as f:
            f.write(_INDEX_HEADER)
            f.write(struct.pack("<Q", 1))
            f.write(struct.pack("<B", _DTYPE_CODES[self.dtype]))
            f.write(struct.pack("<Q", len(self.sequence_lengths)))
            f.write(struct.pack("<Q", len(self.document_indices)))
            lengths = numpy.array(self.sequence_lengths, dtype=numpy.int32)
            itemsize = numpy.dtype(self.dtype).itemsize
            pointers = numpy.zeros(len(lengths), dtype=numpy.int64)
            if len(lengths) > 1:
                numpy.cumsum(lengths[:-1] * itemsize, out=pointers[1:])
            f.write(lengths.tobytes(order="C"))
            f.write(pointers.tobytes(order="C"))
            f.write(numpy.array(self.document_indices, dtype=numpy.int64).tobytes(order="C"))


class MMapIndexedDataset:
    """mmap reader for the .bin/.idx pair (indexed_dataset.py:228-523)."""

    def __init__(self, path_prefix: str):
        self.path_prefix = str(path_prefix)
        with open(self.path_prefix + ".idx", "rb") as f:
            header = f.read(9)
            assert header == _INDEX_HEADER, f"bad header, cannot read: {self.path_prefix}.idx"
            (version,) = struct.unpack("<Q", f.read(8))
            assert version == 1
            (code,) = struct.unpack("<B", f.read(1))
            self.dtype = _CODE_DTYPES[code]
            (self.sequence_count,) = struct.unpack("<Q", f.read(8))
            (self.document_count,) = struct.unpack("<Q", f.read(8))
            offset = f.tell()
        idx_buf = numpy.memmap(self.path_prefix + ".idx", mode="r", order="C")
        self.sequence_lengths = numpy.frombuffer(idx_buf, dtype=numpy.int32, count=self.sequence_count, offset=offset)
        offset += self.sequence_lengths.nbytes
        self.sequence_pointers = numpy.frombuffer(idx_buf, dtype=numpy.int64, count=self.sequence_count, offset=offset)
        offset += self.sequence_pointers.nbytes
        self.document_indices = numpy.frombuffer(idx_buf, dtype=numpy.int64, count=self.document_count, offset=offset)
        self._bin = numpy.memmap(self.path_prefix + ".bin", mode="r", order="C")

    def __len__(self):
        return self.sequence_count

    def get(self, idx: int, offset: int = 0, length: int | None = None) -> numpy.ndarray:
        ptr = self.sequence_pointers[idx]
        size = self.sequence_lengths[idx] - offset
        if length is not None:
            size = length
        itemsize = numpy.dtype(self.dtype).itemsize
        return numpy.frombuffer(self._bin, dtype=self.dtype, count=size, offset=int(ptr + offset * itemsize))

    def __getitem__(self, idx: int) -> numpy.ndarray:
        return self.get(idx)


def build_sample_idx(sizes: numpy.ndarray, doc_idx: numpy.ndarray, seq_length: int, num_epochs: int,
                     tokens_per_epoch: int) -> numpy.ndarray:
    """Native sample-index builder (helpers.cpp:74-225 semantics)."""
    num_samples = (num_epochs * tokens_per_epoch - 1) // seq_length
    sizes = numpy.ascontiguousarray(sizes, dtype=numpy.int32)
    doc_idx = numpy.ascontiguousarray(doc_idx, dtype=numpy.int32)
    use64 = num_epochs * tokens_per_epoch > numpy.iinfo(numpy.int32).max // 2
    out_dtype = numpy.int64 if use64 else numpy.int32
    out = numpy.empty((num_samples + 1, 2), dtype=out_dtype)
    lib = hip.lib()
    fn = lib.dolomite_build_sample_idx_i64 if use64 else lib.dolomite_build_sample_idx_i32
    rc = fn(
        ctypes.c_void_p(sizes.ctypes.data), ctypes.c_void_p(doc_idx.ctypes.data),
        seq_length, num_epochs, tokens_per_epoch, ctypes.c_void_p(out.ctypes.data), num_samples,
    )
    hip.check(rc, "build_sample_idx")
    return out


def build_blending_indices(weights: numpy.ndarray, size: int) -> tuple[numpy.ndarray, numpy.ndarray]:
    """Native blending builder (helpers.cpp:17-69 semantics)."""
    weights = numpy.ascontiguousarray(weights, dtype=numpy.float64)
    dataset_index = numpy.empty(size, dtype=numpy.int16)
    dataset_sample_index = numpy.empty(size, dtype=numpy.int64)
    rc = hip.lib().dolomite_build_blending_indices(
        ctypes.c_void_p(dataset_index.ctypes.data), ctypes.c_void_p(dataset_sample_index.ctypes.data),
        ctypes.c_void_p(weights.ctypes.data), len(weights), size,
    )
    hip.check(rc, "build_blending_indices")
    return dataset_index, dataset_sample_index


class GPTDataset:
    """Doc/sample/shuffle-index dataset over an MMapIndexedDataset
    (gpt_dataset.py:241-401; indices built in memory, not cached to disk)."""

    def __init__(self, indexed_dataset: MMapIndexedDataset, num_samples: int, seq_length: int, seed: int = 1234,
                 documents: numpy.ndarray | None = None):
        self.indexed = indexed_dataset
        self.seq_length = seq_length
        if documents is None:
            documents = numpy.arange(len(indexed_dataset), dtype=numpy.int32)
        self.documents = documents

        tokens_per_epoch = int(numpy.sum(self.indexed.sequence_lengths[documents]))
        # _get_num_epochs (gpt_dataset.py:417-439): epochs until >= the
        # requested sample count (with the -1 window overlap)
        num_epochs = 0
        tokens = 0
        while True:
            num_epochs += 1
            tokens += tokens_per_epoch
            if (tokens - 1) // seq_length >= num_samples:
                break

        rng = numpy.random.RandomState(seed)
        # document index: epochs-replicated then shuffled (gpt_dataset.py:442-476)
        doc_idx = numpy.tile(documents, num_epochs).astype(numpy.int32)
        rng.shuffle(doc_idx)
        self.doc_idx = doc_idx
        self.sample_idx = build_sample_idx(
            self.indexed.sequence_lengths, doc_idx, seq_length, num_epochs, tokens_per_epoch
        )
        total = self.sample_idx.shape[0] - 1
        shuffle_idx = numpy.arange(total, dtype=numpy.uint32 if total < 2**32 - 2 else numpy.int64)
        rng.shuffle(shuffle_idx)
        self.shuffle_idx = shuffle_idx
        self.num_samples = num_samples

    def __len__(self):
        return self.sample_idx.shape[0] - 1

    def __getitem__(self, idx: int) -> dict:
        """(gpt_dataset.py:83-160): gather seq_length+1 tokens across the
        documents the window spans."""
        idx = int(self.shuffle_idx[idx % len(self)])
        doc_index_beg, doc_index_beg_offset = self.sample_idx[idx]
        doc_index_end, doc_index_end_offset = self.sample_idx[idx + 1]
        if doc_index_beg == doc_index_end:
            text = self.indexed.get(
                int(self.doc_idx[doc_index_beg]), offset=int(doc_index_beg_offset),
                length=int(doc_index_end_offset - doc_index_beg_offset + 1),
            )
        else:
            parts = [self.indexed.get(int(self.doc_idx[doc_index_beg]), offset=int(doc_index_beg_offset))]
            for i in range(int(doc_index_beg) + 1, int(doc_index_end)):
                parts.append(self.indexed.get(int(self.doc_idx[i])))
            parts.append(self.indexed.get(int(self.doc_idx[doc_index_end]), length=int(doc_index_end_offset) + 1))
            text = numpy.concatenate(parts)
        assert len(text) == self.seq_length + 1, (len(text), self.seq_length + 1)
        return {"text": torch.from_numpy(text.astype(numpy.int64))}


class MegatronDataLoader:
    """Rank-sharded sequential loader with consumed_samples resume
    (data/megatron/sampler.py + ResumableDataLoader semantics)."""

    def __init__(self, dataset: GPTDataset, micro_batch_size: int, consumed_samples: int = 0):
        self.dataset = dataset
        self.micro_batch_size = micro_batch_size
        self.consumed_samples = consumed_samples

    def state_dict(self):
        return {"consumed_samples": self.consumed_samples}

    def load_state_dict(self, sd):
        self.consumed_samples = sd["consumed_samples"]

    def __iter__(self):
        return self

    def __next__(self):
        world, rank = get_world_size(), get_rank()
        start = self.consumed_samples + rank * self.micro_batch_size
        rows = [self.dataset[start + i]["text"] for i in range(self.micro_batch_size)]
        self.consumed_samples += world * self.micro_batch_size
        return {"text": torch.stack(rows)}
