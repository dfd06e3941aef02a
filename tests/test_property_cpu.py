"""Property-based tests (hypothesis) for the pure-Python parity logic:
randomized shapes/contents beyond the fixed golden cases.

  - QKVLayout addressing: the stride arithmetic the HIP kernels use to read
    q/k/v inside the packed c_attn output must agree with the reference's
    view/split semantics for every (H, Hkv, D, head_type).
  - build_sample_idx: the native C builder vs a direct Python restatement
    of helpers.cpp for arbitrary document length mixes.
  - document-boundary reset: eos scan -> cu_seqlens vs a brute-force split.
"""

import numpy
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from dolomite_engine_amd.ops import QKVLayout

settings.register_profile("ci", deadline=None, max_examples=40)
settings.load_profile("ci")


@st.composite
def head_configs(draw):
    head_type = draw(st.sampled_from(["mha", "gqa", "mqa"]))
    D = draw(st.sampled_from([8, 16, 64, 80, 128]))
    if head_type == "mha":
        H = draw(st.integers(1, 8))
        Hkv = H
    elif head_type == "mqa":
        H = draw(st.integers(1, 8))
        Hkv = 1
    else:
        Hkv = draw(st.integers(2, 4))
        H = Hkv * draw(st.integers(2, 4))
    return head_type, H, Hkv, D


@given(head_configs(), st.integers(1, 5))
def test_qkv_layout_strides_match_unpack(cfg, T):
    """The (t_stride, group_stride, offsets) addressing the HIP kernels use
    must pick out exactly the elements unpack_cpu (the reference split
    semantics, attention/padding_free.py:79-116) yields."""
    head_type, H, Hkv, D = cfg
    lo = QKVLayout.make(H, Hkv, D, head_type)
    qkv = torch.arange(T * lo.row_len, dtype=torch.float32).reshape(T, lo.row_len)
    q, k, v = lo.unpack_cpu(qkv)
    for t in range(T):
        for h in range(H):
            base = t * lo.row_len + (h // lo.G) * lo.q_gstride + (h % lo.G) * D
            expect = qkv.reshape(-1)[base : base + D]
            torch.testing.assert_close(q[t, h], expect, rtol=0, atol=0)
        for j in range(Hkv):
            kbase = t * lo.row_len + lo.k_off + j * lo.kv_hstride
            vbase = t * lo.row_len + lo.v_off + j * lo.kv_hstride
            torch.testing.assert_close(k[t, j], qkv.reshape(-1)[kbase : kbase + D], rtol=0, atol=0)
            torch.testing.assert_close(v[t, j], qkv.reshape(-1)[vbase : vbase + D], rtol=0, atol=0)


@given(
    st.lists(st.integers(1, 50), min_size=1, max_size=12),
    st.integers(2, 33),
    st.integers(1, 3),
)
def test_build_sample_idx_property(doc_lens, seq_length, num_epochs):
    """Native C builder vs the pure-python helpers.cpp restatement for
    arbitrary document length mixes."""
    from dolomite_engine_amd.megatron import build_sample_idx
    from tests.test_megatron_data import _sample_idx_python

    sizes = numpy.array(doc_lens, dtype=numpy.int32)
    doc_idx = numpy.tile(numpy.arange(len(sizes), dtype=numpy.int32), num_epochs)
    tokens_per_epoch = int(sizes.sum())
    if (num_epochs * tokens_per_epoch - 1) // seq_length <= 0:
        return
    got = build_sample_idx(sizes, doc_idx, seq_length, num_epochs, tokens_per_epoch)
    ref = _sample_idx_python(sizes, doc_idx, seq_length, num_epochs, tokens_per_epoch)
    numpy.testing.assert_array_equal(numpy.asarray(got, dtype=numpy.int64), ref)


@given(st.lists(st.integers(0, 30), min_size=2, max_size=40), st.integers(0, 30))
def test_document_boundary_reset_property(tokens, eos):
    """reset_attention_mask semantics (model_wrapper/pretraining.py:136-158):
    cu_seqlens from the eos scan == brute-force document split. Single row
    (B=1, S=len(tokens)); rows always close a document window."""
    from dolomite_engine_amd.model_wrapper import ModelWrapperForPretraining

    S = len(tokens)
    w = ModelWrapperForPretraining.__new__(ModelWrapperForPretraining)
    w.eos_token_id = eos
    w.reset_position_ids = True
    w._buffers_device = None
    flat = torch.tensor(tokens, dtype=torch.int64)
    cu, max_len, pos = w._document_boundaries(flat, 1, S)

    ref_cu = [0] + [i + 1 for i, tok in enumerate(tokens) if tok == eos or i == S - 1]
    ref_cu = sorted(set(ref_cu))
    assert cu.tolist() == ref_cu
    seg_lens = [b - a for a, b in zip(ref_cu, ref_cu[1:])]
    assert max_len == max(seg_lens)
    assert pos.tolist() == [i for n in seg_lens for i in range(n)]
