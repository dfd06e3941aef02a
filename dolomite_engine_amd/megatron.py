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


def _build_document_index(documents, num_epochs, rng, separate_final_epoch: bool):
    """Epochs-replicated shuffled document order (gpt_dataset.py:442-476).

    With separate_final_epoch the last epoch's documents are shuffled on
    their own and appended, so a partially-consumed final epoch samples a
    contiguous (separately shuffled) tail instead of leaking into the
    global shuffle."""
    if not separate_final_epoch or num_epochs == 1:
        doc_idx = numpy.tile(documents, num_epochs).astype(documents.dtype)
        rng.shuffle(doc_idx)
        return doc_idx
    first = _build_document_index(documents, num_epochs - 1, rng, False)
    last = _build_document_index(documents, 1, rng, False)
    return numpy.concatenate((first, last))


def _build_shuffle_index(num_samples: int, total_size: int, rng):
    """[0, num_samples) and [num_samples, total_size) shuffled separately
    (gpt_dataset.py:478-510)."""
    dtype_ = numpy.uint32 if total_size < numpy.iinfo(numpy.uint32).max - 1 else numpy.int64
    first = numpy.arange(0, num_samples, dtype=dtype_)
    rng.shuffle(first)
    if num_samples == total_size:
        return first
    last = numpy.arange(num_samples, total_size, dtype=dtype_)
    rng.shuffle(last)
    return numpy.concatenate((first, last))




# FIM (fill-in-the-middle) augmentation, gpt_dataset.py:16-20, 513-600.
FIM_PREFIX = "<fim_prefix>"
FIM_MIDDLE = "<fim_middle>"
FIM_SUFFIX = "<fim_suffix>"
FIM_PAD = "<fim_pad>"


def fim_permute(sample, np_rng, fim_rate, fim_spm_rate, tokenizer,
                suffix_tok_id, prefix_tok_id, middle_tok_id, pad_tok_id):
    """One document segment -> PSM/SPM FIM rearrangement with probability
    fim_rate (gpt_dataset.py:513-600, truncate_or_pad=False call form)."""
    if not np_rng.binomial(1, fim_rate):
        return sample
    contents = tokenizer.detokenize(sample)
    boundaries = list(np_rng.randint(low=0, high=len(contents) + 1, size=2))
    boundaries.sort()
    prefix = numpy.array(tokenizer.tokenize(contents[: boundaries[0]]), dtype=numpy.int64)
    middle = numpy.array(tokenizer.tokenize(contents[boundaries[0] : boundaries[1]]), dtype=numpy.int64)
    suffix = numpy.array(tokenizer.tokenize(contents[boundaries[1] :]), dtype=numpy.int64)
    if np_rng.binomial(1, fim_spm_rate):  # SPM (FIM paper variant 2)
        return numpy.concatenate([[prefix_tok_id, suffix_tok_id], suffix, [middle_tok_id], prefix, middle])
    return numpy.concatenate([[prefix_tok_id], prefix, [suffix_tok_id], suffix, [middle_tok_id], middle])


def fim_transform(sample, np_rng, fim_rate, fim_spm_rate, tokenizer,
                  suffix_tok_id, prefix_tok_id, middle_tok_id, pad_tok_id):
    """Apply FIM per document segment (split on eod), then truncate/pad back
    to the original window length (gpt_dataset.py:170-239)."""
    sample_len = sample.shape[0]
    eod = tokenizer.eod
    breaks = numpy.argwhere(sample == eod)
    kw = dict(fim_rate=fim_rate, fim_spm_rate=fim_spm_rate, tokenizer=tokenizer,
              suffix_tok_id=suffix_tok_id, prefix_tok_id=prefix_tok_id,
              middle_tok_id=middle_tok_id, pad_tok_id=pad_tok_id)
    if breaks.shape != (0, 1):
        cur = 0
        parts = []
        for loc in numpy.nditer(breaks):
            if loc - cur > 0:
                parts += [fim_permute(sample[cur:loc], np_rng, **kw), numpy.array([eod], dtype=numpy.int64)]
            cur = int(loc) + 1
        parts.append(fim_permute(sample[cur:], np_rng, **kw))
        sample = numpy.concatenate(parts)
    else:
        sample = fim_permute(sample, np_rng, **kw)
    diff = sample.shape[0] - sample_len
    if diff > 0:
        sample = sample[:sample_len]
    elif diff < 0:
        sample = numpy.concatenate([sample, numpy.full(-diff, pad_tok_id, dtype=numpy.int64)])
    assert sample.shape[0] == sample_len
    return sample


class GPTDataset:
    """Doc/sample/shuffle-index dataset over an MMapIndexedDataset
    (gpt_dataset.py:241-401; indices built in memory, not cached to disk)."""

    def __init__(self, indexed_dataset: MMapIndexedDataset, num_samples: int, seq_length: int, seed: int = 1234,
                 documents: numpy.ndarray | None = None,
                 fim_rate: float = 0.0, fim_spm_rate: float = 0.5, tokenizer=None):
        self.indexed = indexed_dataset
        self.seq_length = seq_length
        self.fim_rate = fim_rate
        self.fim_spm_rate = fim_spm_rate
        self.tokenizer = tokenizer
        if fim_rate != 0:
            assert 0 <= fim_rate <= 1 and tokenizer is not None, "FIM needs a tokenizer and 0<=rate<=1"
            self.np_rng = numpy.random.RandomState(seed=seed)
            self.suffix_tok_id, self.prefix_tok_id, self.middle_tok_id, self.pad_tok_id = (
                tokenizer.convert_tokens_to_ids(t) for t in (FIM_SUFFIX, FIM_PREFIX, FIM_MIDDLE, FIM_PAD)
            )
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

        # separate_final_epoch decision (gpt_dataset.py:300-317): shuffle the
        # final epoch separately when the run consumes <80% of it
        if num_epochs == 1:
            separate_final_epoch = False
            num_samples_sans_final_epoch = 0
        else:
            num_samples_sans_final_epoch = ((num_epochs - 1) * tokens_per_epoch - 1) // seq_length
            num_samples_from_final_epoch = num_samples - num_samples_sans_final_epoch
            num_samples_per_epoch = (tokens_per_epoch - 1) // seq_length
            assert 0 <= num_samples_from_final_epoch <= num_samples_per_epoch + 1
            separate_final_epoch = num_samples_from_final_epoch < int(0.80 * num_samples_per_epoch)

        rng = numpy.random.RandomState(seed)
        doc_idx = _build_document_index(documents.astype(numpy.int32), num_epochs, rng, separate_final_epoch)
        self.doc_idx = doc_idx
        self.sample_idx = build_sample_idx(
            self.indexed.sequence_lengths, doc_idx, seq_length, num_epochs, tokens_per_epoch
        )
        total = self.sample_idx.shape[0] - 1
        if separate_final_epoch:
            self.shuffle_idx = _build_shuffle_index(num_samples_sans_final_epoch, total, rng)
        else:
            self.shuffle_idx = _build_shuffle_index(total, total, rng)
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
        text = text.astype(numpy.int64)
        if self.fim_rate != 0:
            text = fim_transform(
                text, self.np_rng, self.fim_rate, self.fim_spm_rate, self.tokenizer,
                self.suffix_tok_id, self.prefix_tok_id, self.middle_tok_id, self.pad_tok_id,
            )
        return {"text": torch.from_numpy(text)}


def parse_and_normalize_split(split: str) -> list:
    """"949,50,1" -> [0.949, 0.05, 0.001] (blended_megatron_dataset_config.py:98-115)."""
    import re

    vals = list(map(float, re.findall(r"[.0-9]+", split)))
    vals = vals + [0.0] * (3 - len(vals))
    assert len(vals) == 3 and all(v >= 0.0 for v in vals)
    total = sum(vals)
    return [v / total for v in vals]


def get_split_indices(split: list, num_elements: int) -> list:
    """Document-index bounds per split (blended_megatron_dataset_builder.py:376-398)."""
    bounds = [0]
    for pct in split:
        bounds.append(bounds[-1] + int(round(pct * float(num_elements))))
    bounds[1:] = [b - (bounds[-1] - num_elements) for b in bounds[1:]]
    assert bounds[-1] == num_elements
    return bounds


class BlendedDataset:
    """Weighted interleave of GPTDatasets (blended_dataset.py:22-120): the
    greedy max-error assignment comes from the native builder
    (helpers.cpp:17-69 semantics via csrc)."""

    def __init__(self, datasets: list, weights: list, size: int):
        assert len(datasets) == len(weights)
        total = sum(weights)
        self.weights = [w / total for w in weights]
        self.datasets = datasets
        self.size = size
        self.dataset_index, self.dataset_sample_index = build_blending_indices(
            numpy.array(self.weights, dtype=numpy.float64), size
        )

    def __len__(self):
        return self.size

    def __getitem__(self, idx: int) -> dict:
        if idx >= self.size:
            raise IndexError(idx)
        return self.datasets[int(self.dataset_index[idx])][int(self.dataset_sample_index[idx])]


def train_val_test_samples(num_training_steps: int, micro_batch_size: int, gradient_accumulation_steps: int,
                           eval_interval: int | None, eval_steps: int, world_size: int) -> tuple:
    """Per-split sample targets (data/megatron/__init__.py:215-234)."""
    per_step = micro_batch_size * gradient_accumulation_steps * world_size
    train = num_training_steps * per_step
    val = (num_training_steps // eval_interval + 1) * eval_steps * per_step if eval_interval else 0
    test = eval_steps * per_step
    return train, val, test


def build_train_val_test_datasets(data_path, split: str, sizes: tuple, seq_length: int, seed: int,
                                  fim_rate: float = 0.0, fim_spm_rate: float = 0.5, tokenizer=None):
    """Build the train/val/test GPTDatasets, blended across weighted paths
    (data/megatron/__init__.py:18-110 options 1 and 2).

    data_path: "prefix" | ["prefix"] | [w0, "prefix0", w1, "prefix1", ...]
    split: "949,50,1"-style ratio string over each dataset's documents.
    sizes: (train_samples, val_samples, test_samples).
    """
    if isinstance(data_path, (str, Path)):
        blend = [1.0, str(data_path)]
    elif len(data_path) == 1:
        blend = [1.0, str(data_path[0])]
    else:
        assert len(data_path) % 2 == 0, "weighted blend must be [w0, path0, w1, path1, ...]"
        blend = [float(data_path[i]) if i % 2 == 0 else str(data_path[i]) for i in range(len(data_path))]

    weights = [blend[i] for i in range(0, len(blend), 2)]
    prefixes = [blend[i + 1] for i in range(0, len(blend), 2)]
    total_w = sum(weights)
    weights = [w / total_w for w in weights]
    # 0.5% margin per dataset (blended_megatron_dataset_builder.py:421-426)
    import math

    sizes_per_dataset = [[int(math.ceil(s * w * 1.005)) for s in sizes] for w in weights]

    split_vec = parse_and_normalize_split(split)
    per_dataset_splits = []  # [dataset][split] -> GPTDataset | None
    for prefix, dsizes in zip(prefixes, sizes_per_dataset):
        indexed = MMapIndexedDataset(prefix)
        ndocs = len(indexed)
        bounds = get_split_indices(split_vec, ndocs)
        per_split = []
        for si in range(3):
            if split_vec[si] == 0.0 or sizes[si] == 0:
                per_split.append(None)
            else:
                docs = numpy.arange(bounds[si], bounds[si + 1], dtype=numpy.int32)
                per_split.append(GPTDataset(indexed, dsizes[si], seq_length, seed=seed, documents=docs,
                                            fim_rate=fim_rate, fim_spm_rate=fim_spm_rate, tokenizer=tokenizer))
        per_dataset_splits.append(per_split)

    out = []
    for si in range(3):
        live = [ds[si] for ds in per_dataset_splits if ds[si] is not None]
        if not live:
            out.append(None)
        elif len(per_dataset_splits) == 1:
            out.append(live[0])
        else:
            out.append(BlendedDataset(live, weights, sizes[si]))
    return tuple(out)


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
