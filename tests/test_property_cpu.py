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
    try:
        from tests.test_megatron_data import _sample_idx_python
    except ModuleNotFoundError:  # subset invocations without the package path
        from test_megatron_data import _sample_idx_python

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


@given(
    st.integers(1, 6),       # rows
    st.sampled_from([8, 64, 96, 2560]),
    st.sampled_from(["fp32", "bf16"]),
)
def test_rmsnorm_cpu_branch_matches_torch(T, H, dtype):
    """The CPU branch of FusedRMSNorm (the branch the gloo training tests
    run) vs torch's own rms_norm at random shapes, fwd and bwd."""
    dt = torch.float32 if dtype == "fp32" else torch.bfloat16
    g = torch.Generator().manual_seed(T * H)
    x = torch.randn(T, H, generator=g).to(dt).requires_grad_(True)
    w = (torch.randn(H, generator=g) * 0.1 + 1.0).to(dt).requires_grad_(True)

    from dolomite_engine_amd.ops import fused_rmsnorm

    y, _ = fused_rmsnorm(x, w, 1e-6)
    dy = torch.randn(T, H, generator=g).to(dt)
    y.backward(dy)

    xr = x.detach().clone().requires_grad_(True)
    wr = w.detach().clone().requires_grad_(True)
    # reference semantics (rmsnorm/base.py:18-25): fp32 accum, cast BEFORE w
    x32 = xr.float()
    yr = wr * (x32 * torch.rsqrt(x32.pow(2).mean(-1, keepdim=True) + 1e-6)).to(dt)
    yr.backward(dy)

    tol = dict(rtol=1e-6, atol=1e-6) if dtype == "fp32" else dict(rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(y, yr, **tol)
    torch.testing.assert_close(x.grad, xr.grad, **tol)
    torch.testing.assert_close(w.grad, wr.grad, **tol)


@given(head_configs(), st.integers(2, 12))
def test_rope_forward_inverse_roundtrip(cfg, T):
    """RoPE packed apply then inverse apply is the identity (the backward
    rotation used by RoPEPackedQKV.backward is exact: R^T = -R, duplicated
    table halves commute)."""
    from dolomite_engine_amd.ops.functional import RoPEPackedQKV

    head_type, H, Hkv, D = cfg
    if D % 2 != 0:
        return
    lo = QKVLayout.make(H, Hkv, D, head_type)
    g = torch.Generator().manual_seed(T)
    qkv = torch.randn(T, lo.row_len, generator=g)
    pos = torch.arange(T, dtype=torch.float32)
    inv = 1.0 / (10000.0 ** (torch.arange(0, D, 2, dtype=torch.float32) / D))
    fr = torch.outer(pos, inv)
    emb = torch.cat((fr, fr), dim=-1)
    cos, sin = emb.cos(), emb.sin()

    rotated = RoPEPackedQKV.apply(qkv.clone(), cos, sin, lo)
    # inverse rotation = apply with negated sin
    restored = RoPEPackedQKV.apply(rotated.clone(), cos, -sin, lo)
    torch.testing.assert_close(restored, qkv, rtol=1e-5, atol=1e-5)
    # v slots must be untouched by the forward rotation
    _, _, v0 = lo.unpack_cpu(qkv)
    _, _, v1 = lo.unpack_cpu(rotated)
    torch.testing.assert_close(v0, v1, rtol=0, atol=0)


@given(st.integers(2, 40), st.sampled_from([17, 512, 1000]), st.integers(0, 3))
def test_fused_ce_cpu_matches_torch(T, V, n_ignore):
    """CPU branch of FusedCrossEntropy vs F.cross_entropy(mean, -100),
    forward and dlogits, at random shapes with ignored labels."""
    import torch.nn.functional as F

    from dolomite_engine_amd.ops import fused_cross_entropy

    g = torch.Generator().manual_seed(T * V + n_ignore)
    logits = torch.randn(T, V, generator=g).requires_grad_(True)
    labels = torch.randint(0, V, (T,), generator=g)
    labels[:n_ignore] = -100
    if (labels != -100).sum() == 0:
        return
    loss = fused_cross_entropy(logits, labels)
    loss.backward()

    lr = logits.detach().clone().requires_grad_(True)
    ref = F.cross_entropy(lr, labels, ignore_index=-100)
    ref.backward()
    torch.testing.assert_close(loss, ref, rtol=1e-6, atol=1e-6)
    torch.testing.assert_close(logits.grad, lr.grad, rtol=1e-5, atol=1e-7)
