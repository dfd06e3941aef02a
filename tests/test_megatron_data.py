"""Megatron data pipeline: .bin/.idx round trip, native index builders vs
pure-python restatements, GPT window extraction correctness, blending, and
consumed_samples resume (reference tests/data/megatron_data_test.py model)."""

import numpy
import pytest
import torch

from dolomite_engine_amd.megatron import (
    GPTDataset,
    MegatronDataLoader,
    MMapIndexedDataset,
    MMapIndexedDatasetBuilder,
    build_blending_indices,
    build_sample_idx,
)


def _write_corpus(tmp_path, docs, dtype=numpy.int32):
    b = MMapIndexedDatasetBuilder(tmp_path / "corpus", dtype=dtype)
    for d in docs:
        b.add_document(d)
    b.finalize()
    return MMapIndexedDataset(tmp_path / "corpus")


def test_bin_idx_round_trip(tmp_path):
    g = numpy.random.RandomState(0)
    docs = [g.randint(0, 1000, size=g.randint(1, 50)).astype(numpy.int32) for _ in range(20)]
    ds = _write_corpus(tmp_path, docs)
    assert len(ds) == 20
    for i, d in enumerate(docs):
        numpy.testing.assert_array_equal(ds[i], d)
        numpy.testing.assert_array_equal(ds.get(i, offset=1), d[1:])
        if len(d) > 2:
            numpy.testing.assert_array_equal(ds.get(i, offset=1, length=len(d) - 2), d[1:-1])


def _sample_idx_python(sizes, doc_idx, seq_length, num_epochs, tokens_per_epoch):
    """Pure-python restatement of helpers.cpp:74-148 (the pin)."""
    num_samples = (num_epochs * tokens_per_epoch - 1) // seq_length
    out = numpy.zeros((num_samples + 1, 2), dtype=numpy.int64)
    sample_index, doc_idx_index, doc_offset = 1, 0, 0
    out[0] = (0, 0)
    while sample_index <= num_samples:
        remaining = seq_length + 1
        while remaining != 0:
            doc_id = doc_idx[doc_idx_index]
            doc_length = sizes[doc_id] - doc_offset
            remaining -= doc_length
            if remaining <= 0:
                doc_offset += remaining + doc_length - 1
                remaining = 0
            else:
                doc_idx_index += 1
                doc_offset = 0
        out[sample_index] = (doc_idx_index, doc_offset)
        sample_index += 1
    return out


def test_native_sample_idx_matches_python():
    g = numpy.random.RandomState(1)
    sizes = g.randint(3, 40, size=30).astype(numpy.int32)
    num_epochs = 3
    doc_idx = numpy.tile(numpy.arange(30, dtype=numpy.int32), num_epochs)
    g.shuffle(doc_idx)
    tokens_per_epoch = int(sizes.sum())
    native = build_sample_idx(sizes, doc_idx, 16, num_epochs, tokens_per_epoch)
    ref = _sample_idx_python(sizes, doc_idx, 16, num_epochs, tokens_per_epoch)
    numpy.testing.assert_array_equal(native.astype(numpy.int64), ref)


def test_native_blending_matches_python():
    weights = numpy.array([0.5, 0.3, 0.2])
    size = 500
    di, dsi = build_blending_indices(weights, size)
    # python restatement of helpers.cpp:17-69
    cur = [0, 0, 0]
    for s in range(size):
        sd = max(float(s), 1.0)
        errors = [weights[d] * sd - cur[d] for d in range(3)]
        d = int(numpy.argmax(errors))
        assert di[s] == d, s
        assert dsi[s] == cur[d], s
        cur[d] += 1
    achieved = numpy.bincount(di, minlength=3) / size
    numpy.testing.assert_allclose(achieved, weights, atol=0.01)


def test_gpt_dataset_windows_correct(tmp_path):
    g = numpy.random.RandomState(2)
    docs = [g.randint(0, 100, size=g.randint(4, 30)).astype(numpy.int32) for _ in range(12)]
    ds = _write_corpus(tmp_path, docs)
    S = 10
    gpt = GPTDataset(ds, num_samples=40, seq_length=S, seed=7)

    # every sample must equal the corresponding slice of the epoch-replicated
    # shuffled document token stream
    stream = numpy.concatenate([docs[d] for d in gpt.doc_idx])
    for i in range(len(gpt)):
        row = gpt[i]["text"].numpy()
        sidx = int(gpt.shuffle_idx[i])
        expected = stream[sidx * S : sidx * S + S + 1]
        numpy.testing.assert_array_equal(row, expected)


def test_megatron_dataloader_resume(tmp_path):
    g = numpy.random.RandomState(3)
    docs = [g.randint(0, 100, size=20).astype(numpy.int32) for _ in range(10)]
    ds = _write_corpus(tmp_path, docs)
    gpt = GPTDataset(ds, num_samples=64, seq_length=8, seed=1)

    dl = MegatronDataLoader(gpt, micro_batch_size=2)
    batches = [next(dl) for _ in range(4)]
    state = dl.state_dict()
    more = [next(dl) for _ in range(2)]

    dl2 = MegatronDataLoader(gpt, micro_batch_size=2)
    dl2.load_state_dict(state)
    again = [next(dl2) for _ in range(2)]
    for a, b in zip(more, again):
        torch.testing.assert_close(a["text"], b["text"], rtol=0, atol=0)


def test_separate_final_epoch_matches_reference_semantics(tmp_path):
    """Multi-epoch run consuming a small slice of the last epoch: the
    document index must keep the final epoch contiguous (separately
    shuffled) and the shuffle index must shuffle the two ranges
    independently (reference gpt_dataset.py:300-317, 442-510)."""
    g = numpy.random.RandomState(3)
    docs = [g.randint(0, 100, size=20).astype(numpy.int32) for _ in range(10)]
    ds = _write_corpus(tmp_path, docs)
    seq_length = 16
    tokens_per_epoch = 200
    samples_per_epoch = (tokens_per_epoch - 1) // seq_length  # 12
    # 3 epochs with barely any samples from the last -> separate_final_epoch
    num_samples = 2 * samples_per_epoch + 1
    gpt = GPTDataset(ds, num_samples=num_samples, seq_length=seq_length, seed=5)

    # reference construction with the same RandomState sequence
    rng = numpy.random.RandomState(5)
    documents = numpy.arange(10, dtype=numpy.int32)
    first = numpy.tile(documents, 2).astype(numpy.int32)
    rng.shuffle(first)
    last = documents.copy()
    rng.shuffle(last)
    expect_doc = numpy.concatenate([first, last])
    numpy.testing.assert_array_equal(gpt.doc_idx, expect_doc)

    total = gpt.sample_idx.shape[0] - 1
    ns_sans_final = (2 * tokens_per_epoch - 1) // seq_length
    f = numpy.arange(0, ns_sans_final, dtype=numpy.uint32)
    rng.shuffle(f)
    l = numpy.arange(ns_sans_final, total, dtype=numpy.uint32)
    rng.shuffle(l)
    numpy.testing.assert_array_equal(gpt.shuffle_idx, numpy.concatenate([f, l]))
    # windows still well-formed
    for i in (0, num_samples - 1):
        assert len(gpt[i]["text"]) == seq_length + 1


def test_blended_train_val_test_split(tmp_path):
    """build_train_val_test_datasets: weighted two-corpus blend with a
    document split; per-split datasets draw only from their document range
    and the blend follows the greedy max-error assignment."""
    from dolomite_engine_amd.megatron import BlendedDataset, build_train_val_test_datasets

    g = numpy.random.RandomState(11)
    for name, base in (("c0", 0), ("c1", 50000)):
        b = MMapIndexedDatasetBuilder(tmp_path / name, dtype=numpy.int32)
        for i in range(40):
            b.add_document(base + g.randint(0, 1000, size=30).astype(numpy.int32))
        b.finalize()

    sizes = (64, 16, 8)
    train, val, test = build_train_val_test_datasets(
        [0.3, str(tmp_path / "c0"), 0.7, str(tmp_path / "c1")],
        split="8,1,1", sizes=sizes, seq_length=16, seed=9,
    )
    assert isinstance(train, BlendedDataset) and isinstance(val, BlendedDataset)
    assert len(train) == sizes[0] and len(val) == sizes[1] and len(test) == sizes[2]
    # blend ratio of the underlying assignment approximately follows weights
    frac1 = float(numpy.mean(train.dataset_index == 1))
    assert 0.5 < frac1 < 0.9
    # val documents come from the val split range only: tokens of c0 are
    # <50000, c1 >=50000; both corpora contribute across the blend
    seen = {int(train[i]["text"][0] >= 50000) for i in range(len(train))}
    assert seen == {0, 1}
    for i in range(len(val)):
        assert len(val[i]["text"]) == 17


def test_single_path_split_val_loader(tmp_path):
    """Single un-weighted corpus path: split gives disjoint document
    ranges for train/val."""
    from dolomite_engine_amd.megatron import build_train_val_test_datasets

    b = MMapIndexedDatasetBuilder(tmp_path / "c", dtype=numpy.int32)
    for i in range(100):
        b.add_document(numpy.full(21, i, dtype=numpy.int32))
    b.finalize()
    train, val, test = build_train_val_test_datasets(
        str(tmp_path / "c"), split="90,5,5", sizes=(50, 10, 10), seq_length=20, seed=1,
    )
    train_docs = {int(train[i]["text"][0]) for i in range(len(train))}
    val_docs = {int(val[i]["text"][0]) for i in range(len(val))}
    test_docs = {int(test[i]["text"][0]) for i in range(len(test))}
    assert train_docs and val_docs and test_docs
    assert max(train_docs) < 90 and min(val_docs) >= 90 and val_docs.isdisjoint(test_docs)


class _FakeFimTokenizer:
    """Identity 'tokenizer' over comma-separated token strings, with the
    FIM sentinel vocabulary (tests the transform, not BPE)."""

    eod = 0
    _special = {"<fim_suffix>": 9001, "<fim_prefix>": 9002, "<fim_middle>": 9003, "<fim_pad>": 9004}

    def detokenize(self, arr):
        return ",".join(str(int(x)) for x in arr)

    def tokenize(self, s):
        return [int(x) for x in s.split(",") if x != ""]

    def convert_tokens_to_ids(self, t):
        return self._special[t]


def test_fim_transform_semantics():
    """gpt_dataset.py:513-600: PSM/SPM rearrangement per eod-split segment,
    window length preserved, sentinels present, rate=0 is identity."""
    from dolomite_engine_amd.megatron import fim_transform

    tok = _FakeFimTokenizer()
    rng = numpy.random.RandomState(0)
    sample = numpy.array([5, 6, 7, 0, 8, 9, 10, 11], dtype=numpy.int64)

    out0 = fim_transform(sample.copy(), rng, 0.0, 0.5, tok, 9001, 9002, 9003, 9004)
    numpy.testing.assert_array_equal(out0, sample)

    rng = numpy.random.RandomState(1)
    out = fim_transform(sample.copy(), rng, 1.0, 0.5, tok, 9001, 9002, 9003, 9004)
    assert out.shape == sample.shape
    assert (out >= 9001).any(), out  # sentinels inserted
    # eod-separated structure: eod still present (first segment + separator)
    assert (out == 0).sum() >= 1
    # the multiset of original tokens in the output is a subset of the input
    kept = [t for t in out if t < 9000 and t != 0]
    assert set(kept) <= set(sample.tolist())


def test_gpt_dataset_fim_rate(tmp_path):
    from dolomite_engine_amd.megatron import GPTDataset

    docs = [numpy.arange(1, 30, dtype=numpy.int32) for _ in range(6)]
    ds = _write_corpus(tmp_path, docs)
    tok = _FakeFimTokenizer()
    gpt = GPTDataset(ds, num_samples=8, seq_length=16, seed=3, fim_rate=1.0, tokenizer=tok)
    for i in range(4):
        t = gpt[i]["text"]
        assert t.shape == (17,)
    # rate 0 path unchanged vs plain dataset
    a = GPTDataset(ds, num_samples=8, seq_length=16, seed=3)
    b = GPTDataset(ds, num_samples=8, seq_length=16, seed=3, fim_rate=0.0, tokenizer=tok)
    torch.testing.assert_close(a[0]["text"], b[0]["text"])


import os as _os


@pytest.mark.skipif(not _os.path.isdir("/root/reference"), reason="reference checkout not present")
def test_fim_permute_matches_reference():
    """Our fim_permute vs the reference's permute (gpt_dataset.py:513-600)
    on the same RandomState stream and tokenizer, across rates."""
    import importlib.util
    import sys
    import types

    # gpt_dataset imports heavy siblings; load just the permute function by
    # executing the module source with stubbed imports
    src = open("/root/reference/dolomite_engine/data/megatron/gpt_dataset.py").read()
    start = src.find("def permute(")
    assert start > 0
    # isolate the function body (ends at the next top-level def/EOF)
    import re

    m = re.search(r"\ndef \w+\(", src[start + 10 :])
    fn_src = src[start : start + 10 + m.start()] if m else src[start:]
    ns = {"numpy": numpy, "AutoTokenizer": object}  # annotation-only name
    exec(fn_src, ns)
    ref_permute = ns["permute"]

    from dolomite_engine_amd.megatron import fim_permute

    tok = _FakeFimTokenizer()
    for seed in range(6):
        for rate, spm in ((1.0, 0.0), (1.0, 1.0), (0.5, 0.5), (0.0, 0.5)):
            sample = numpy.arange(10, 30, dtype=numpy.int64)
            r1 = numpy.random.RandomState(seed)
            ours = fim_permute(sample.copy(), r1, rate, spm, tok, 9001, 9002, 9003, 9004)
            r2 = numpy.random.RandomState(seed)
            theirs, _ = ref_permute(sample.copy(), r2, rate, spm, tok, truncate_or_pad=False,
                                    suffix_tok_id=9001, prefix_tok_id=9002, middle_tok_id=9003,
                                    pad_tok_id=9004)
            numpy.testing.assert_array_equal(ours, numpy.asarray(theirs, dtype=numpy.int64))
