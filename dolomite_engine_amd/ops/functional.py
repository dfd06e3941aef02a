"""Autograd ops over the dolomite_hip C-ABI.

Each torch.autograd.Function dispatches per device:
  - CUDA (= ROCm/HIP here): the hand-written gfx950 kernels via ctypes.
    A missing extension raises — no eager fallback on GPU.
  - CPU: a plain-torch restatement of the SAME math, so the identical
    padding-free graph (including the ZeRO-2 training loop) is exercisable
    in CPU tests. The reference itself cannot run its padding-free path on
    CPU (flash-attn gated, attention/padding_free.py:10); the CPU branch
    here is a test vehicle, never what bench.py or smoke() measures.

Reference anchors for the math are in include/dolomite_hip.h and
oracle/model.py.
"""

from dataclasses import dataclass

import torch

from . import hip


# ---------------------------------------------------------------------------
# Packed-QKV layout bookkeeping (attention/base.py:72-81 fused c_attn output)
# ---------------------------------------------------------------------------


@dataclass(frozen=True)
class QKVLayout:
    H: int      # query heads
    Hkv: int    # kv heads
    D: int      # head dim
    G: int      # query heads per kv group
    row_len: int
    q_gstride: int  # elements between q groups within a row
    k_off: int      # element offset of kv-head-0 K within a row
    kv_hstride: int # elements between kv heads (K->K and V->V)
    v_off: int      # element offset of kv-head-0 V within a row

    @staticmethod
    def make(H: int, Hkv: int, D: int, attention_head_type: str) -> "QKVLayout":
        """Mirrors the reference's per-head-type QKV split on the packed
        (T, h + 2*Hkv*D) c_attn output (attention/padding_free.py:79-116)."""
        row_len = H * D + 2 * Hkv * D
        if attention_head_type == "mha":
            # view (T, H, 3D): [q | k | v] per head
            return QKVLayout(H, Hkv, D, 1, row_len, 3 * D, D, 3 * D, 2 * D)
        if attention_head_type == "gqa":
            G = H // Hkv
            # view (T, Hkv, (G+2)D): [q*G | k | v] per group
            return QKVLayout(H, Hkv, D, G, row_len, (G + 2) * D, G * D, (G + 2) * D, (G + 1) * D)
        if attention_head_type == "mqa":
            # split (H*D, D, D)
            return QKVLayout(H, 1, D, H, row_len, 0, H * D, 0, H * D + D)
        raise ValueError(attention_head_type)

    def unpack_cpu(self, qkv: torch.Tensor):
        """CPU view/reshape of packed qkv -> q (T,H,D), k/v (T,Hkv,D)
        (copies where the reference needs a reshape too)."""
        T = qkv.shape[0]
        H, Hkv, D, G = self.H, self.Hkv, self.D, self.G
        if G == 1:  # mha
            hs = qkv.view(T, Hkv, 3 * D)
            q, k, v = hs.chunk(3, dim=-1)
        elif self.Hkv > 1:  # gqa
            hs = qkv.view(T, Hkv, (G + 2) * D)
            q, k, v = hs.split((G * D, D, D), dim=-1)
            q = q.reshape(T, H, D)
        else:  # mqa
            q, k, v = qkv.split((H * D, D, D), dim=-1)
            q = q.view(T, H, D)
            k = k.unsqueeze(1)
            v = v.unsqueeze(1)
        return q.reshape(T, H, D), k.reshape(T, Hkv, D), v.reshape(T, Hkv, D)


# ---------------------------------------------------------------------------
# Fused (residual-add +) RMSNorm
# ---------------------------------------------------------------------------


def _rmsnorm_fwd_cpu(x, res, w, eps):
    s = x if res is None else x + res
    s32 = s.float()
    rstd = torch.rsqrt(s32.pow(2).mean(-1, keepdim=True) + eps)
    y = w * (s32 * rstd).to(x.dtype)
    return y, s, rstd.squeeze(-1)


def _rmsnorm_bwd_cpu(dy, s, w, rstd, dtype):
    s32 = s.float()
    r = rstd.unsqueeze(-1)
    shat = s32 * r
    wdy = (w * dy).float()
    dot = (wdy * shat).mean(-1, keepdim=True)
    dx = (r * (wdy - shat * dot)).to(dtype)
    dw = (dy.float() * shat.to(dtype).float()).sum(0).to(dtype)
    return dx, dw


class FusedRMSNorm(torch.autograd.Function):
    """y, s = rmsnorm(x [+ residual]); s is the pre-norm sum (next residual).
    Semantics: rmsnorm/base.py:18-25 (fp32 accum, cast before weight)."""

    @staticmethod
    def forward(ctx, x, weight, eps, residual):
        T, H = x.shape[0], x.shape[-1]
        x2 = x.reshape(-1, H)
        if x2.is_cuda:
            x2 = x2.contiguous()
            res2 = residual.reshape(-1, H).contiguous() if residual is not None else None
            y = torch.empty_like(x2)
            s = torch.empty_like(x2) if residual is not None else x2
            rstd = torch.empty(x2.shape[0], dtype=torch.float32, device=x2.device)
            with hip.prof("rmsnorm_fwd"):
             hip.check(
                hip.lib().dolomite_rmsnorm_fwd(
                    hip.stream(), hip.ptr(x2), hip.ptr(res2), hip.ptr(weight),
                    hip.ptr(y), hip.ptr(s) if residual is not None else hip.ptr(None),
                    hip.ptr(rstd), x2.shape[0], H, float(eps), hip.dt(x2),
                ),
                "rmsnorm_fwd",
            )
        else:
            y, s, rstd = _rmsnorm_fwd_cpu(x2, residual.reshape(-1, H) if residual is not None else None, weight, eps)
        ctx.save_for_backward(s, weight, rstd)
        ctx.has_residual = residual is not None
        ctx.shape = x.shape
        return y.view(x.shape), s.view(x.shape)

    @staticmethod
    def backward(ctx, dy, ds):
        s, weight, rstd = ctx.saved_tensors
        H = s.shape[-1]
        dy2 = dy.reshape(-1, H)
        if s.is_cuda:
            dy2 = dy2.contiguous()
            ds2 = ds.reshape(-1, H).contiguous() if ds is not None else None
            dx = torch.empty_like(s)
            nb = hip.lib().dolomite_rmsnorm_bwd_nblocks(s.shape[0])
            dw_partial = torch.empty(nb, H, dtype=torch.float32, device=s.device)
            with hip.prof("rmsnorm_bwd"):
             hip.check(
                hip.lib().dolomite_rmsnorm_bwd(
                    hip.stream(), hip.ptr(dy2), hip.ptr(s), hip.ptr(weight),
                    hip.ptr(rstd), hip.ptr(dx), hip.ptr(dw_partial),
                    hip.ptr(ds2),  # residual-stream grad folded into dx in-kernel
                    s.shape[0], H, hip.dt(s),
                ),
                "rmsnorm_bwd",
            )
            dw32 = torch.zeros(H, dtype=torch.float32, device=s.device)
            hip.check(
                hip.lib().dolomite_reduce_partials(hip.stream(), hip.ptr(dw_partial), hip.ptr(dw32), nb, H),
                "reduce_partials",
            )
            dw = dw32.to(weight.dtype)
            dx = dx.view(ctx.shape)
        else:
            dx, dw = _rmsnorm_bwd_cpu(dy2, s, weight, rstd, dy.dtype)
            dx = dx.view(ctx.shape)
            if ds is not None:
                dx = dx + ds
        dres = dx if ctx.has_residual else None
        return dx, dw, None, dres


def fused_rmsnorm(x, weight, eps, residual=None):
    return FusedRMSNorm.apply(x, weight, eps, residual)


# ---------------------------------------------------------------------------
# Fused (residual-add +) LayerNorm
# ---------------------------------------------------------------------------


def _layernorm_fwd_cpu(x, res, w, b, eps):
    s = x if res is None else x + res
    s32 = s.float()
    mu = s32.mean(-1, keepdim=True)
    var = (s32 - mu).pow(2).mean(-1, keepdim=True)
    rstd = torch.rsqrt(var + eps)
    y = ((s32 - mu) * rstd * w.float() + b.float()).to(x.dtype)
    return y, s, mu.squeeze(-1), rstd.squeeze(-1)


class FusedLayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps, residual):
        H = x.shape[-1]
        x2 = x.reshape(-1, H)
        if x2.is_cuda:
            x2 = x2.contiguous()
            res2 = residual.reshape(-1, H).contiguous() if residual is not None else None
            y = torch.empty_like(x2)
            s = torch.empty_like(x2) if residual is not None else x2
            mean = torch.empty(x2.shape[0], dtype=torch.float32, device=x2.device)
            rstd = torch.empty_like(mean)
            hip.check(
                hip.lib().dolomite_layernorm_fwd(
                    hip.stream(), hip.ptr(x2), hip.ptr(res2), hip.ptr(weight), hip.ptr(bias),
                    hip.ptr(y), hip.ptr(s) if residual is not None else hip.ptr(None),
                    hip.ptr(mean), hip.ptr(rstd), x2.shape[0], H, float(eps), hip.dt(x2),
                ),
                "layernorm_fwd",
            )
        else:
            y, s, mean, rstd = _layernorm_fwd_cpu(
                x2, residual.reshape(-1, H) if residual is not None else None, weight, bias, eps
            )
        ctx.save_for_backward(s, weight, mean, rstd)
        ctx.has_residual = residual is not None
        ctx.shape = x.shape
        return y.view(x.shape), s.view(x.shape)

    @staticmethod
    def backward(ctx, dy, ds):
        s, weight, mean, rstd = ctx.saved_tensors
        H = s.shape[-1]
        dy2 = dy.reshape(-1, H)
        if s.is_cuda:
            dy2 = dy2.contiguous()
            ds2 = ds.reshape(-1, H).contiguous() if ds is not None else None
            dx = torch.empty_like(s)
            nb = hip.lib().dolomite_rmsnorm_bwd_nblocks(s.shape[0])
            dwdb = torch.empty(2 * nb, H, dtype=torch.float32, device=s.device)
            hip.check(
                hip.lib().dolomite_layernorm_bwd(
                    hip.stream(), hip.ptr(dy2), hip.ptr(s), hip.ptr(weight),
                    hip.ptr(mean), hip.ptr(rstd), hip.ptr(dx), hip.ptr(dwdb),
                    hip.ptr(ds2), s.shape[0], H, hip.dt(s),
                ),
                "layernorm_bwd",
            )
            # dw partials are rows [0, nb), db rows [nb, 2nb) — reduce both
            # halves separately (out buffers zeroed: atomic join)
            dw32 = torch.zeros(H, dtype=torch.float32, device=s.device)
            db32 = torch.zeros(H, dtype=torch.float32, device=s.device)
            hip.check(hip.lib().dolomite_reduce_partials(hip.stream(), hip.ptr(dwdb), hip.ptr(dw32), nb, H), "rp")
            hip.check(
                hip.lib().dolomite_reduce_partials(hip.stream(), hip.ptr(dwdb, nb * H), hip.ptr(db32), nb, H), "rp"
            )
            dw = dw32.to(weight.dtype)
            db = db32.to(weight.dtype)
            dx = dx.view(ctx.shape)
        else:
            s32 = s.float()
            xhat = (s32 - mean.unsqueeze(-1)) * rstd.unsqueeze(-1)
            wdy = dy2.float() * weight.float()
            m1 = wdy.mean(-1, keepdim=True)
            m2 = (wdy * xhat).mean(-1, keepdim=True)
            dx = (rstd.unsqueeze(-1) * (wdy - m1 - xhat * m2)).to(dy.dtype)
            dw = (dy2.float() * xhat).sum(0).to(weight.dtype)
            db = dy2.float().sum(0).to(weight.dtype)
            dx = dx.view(ctx.shape)
            if ds is not None:
                dx = dx + ds
        dres = dx if ctx.has_residual else None
        return dx, dw, db, None, dres


def fused_layernorm(x, weight, bias, eps, residual=None):
    return FusedLayerNorm.apply(x, weight, bias, eps, residual)


# ---------------------------------------------------------------------------
# RoPE on the packed QKV buffer (in-place)
# ---------------------------------------------------------------------------


def _rope_cpu(qkv, cos, sin, lo: QKVLayout, inverse: bool):
    out = qkv.clone()
    q, k, _ = lo.unpack_cpu(out)
    D = lo.D
    c = cos.to(torch.float32)  # (T, D)
    s = sin.to(torch.float32) * (-1.0 if inverse else 1.0)

    def rot(x):  # x: (T, D)
        x32 = x.float()
        x1, x2 = x32.chunk(2, dim=-1)
        rotated = torch.cat((-x2, x1), dim=-1)
        return (x32 * c + rotated * s).to(x.dtype)

    # write back through the packed layout
    for h in range(lo.H):
        off = (h // lo.G) * lo.q_gstride + (h % lo.G) * D
        out[:, off : off + D] = rot(q[:, h]).to(qkv.dtype)
    for j in range(lo.Hkv):
        off = lo.k_off + j * lo.kv_hstride
        out[:, off : off + D] = rot(k[:, j]).to(qkv.dtype)
    return out


class RoPEPackedQKV(torch.autograd.Function):
    """In-place rotary embedding of q,k heads inside the packed c_attn output
    (rope.py:104-121 at padding_free.py:38-40). cos/sin: (T, D) fp32 gathered
    per token. Backward is the inverse rotation (R^T = -R; tables duplicated)."""

    @staticmethod
    def forward(ctx, qkv, cos, sin, layout: QKVLayout):
        ctx.layout = layout
        ctx.save_for_backward(cos, sin)
        if qkv.is_cuda:
            lo = layout
            hip.check(
                hip.lib().dolomite_rope_qkv(
                    hip.stream(), hip.ptr(qkv), hip.ptr(qkv), hip.ptr(cos), hip.ptr(sin),
                    qkv.shape[0], lo.row_len, lo.H, lo.Hkv, lo.D, lo.G,
                    lo.q_gstride, lo.k_off, lo.kv_hstride, 1, 0, hip.dt(qkv),
                ),
                "rope_qkv fwd",
            )
            ctx.mark_dirty(qkv)
            return qkv
        out = _rope_cpu(qkv, cos, sin, layout, inverse=False)
        return out

    @staticmethod
    def backward(ctx, dqkv):
        cos, sin = ctx.saved_tensors
        lo = ctx.layout
        if dqkv.is_cuda:
            dqkv = dqkv.contiguous()
            hip.check(
                hip.lib().dolomite_rope_qkv(
                    hip.stream(), hip.ptr(dqkv), hip.ptr(dqkv), hip.ptr(cos), hip.ptr(sin),
                    dqkv.shape[0], lo.row_len, lo.H, lo.Hkv, lo.D, lo.G,
                    lo.q_gstride, lo.k_off, lo.kv_hstride, -1, 0, hip.dt(dqkv),
                ),
                "rope_qkv bwd",
            )
            return dqkv, None, None, None
        return _rope_cpu(dqkv, cos, sin, lo, inverse=True), None, None, None


def rope_packed_qkv(qkv, cos, sin, layout: QKVLayout):
    return RoPEPackedQKV.apply(qkv, cos, sin, layout)


# ---------------------------------------------------------------------------
# Varlen causal flash attention on the packed QKV buffer
# ---------------------------------------------------------------------------


def _attn_fwd_cpu(qkv, cu, scale, lo: QKVLayout):
    """fp32-softmax varlen attention + lse, CPU restatement of the kernel."""
    q, k, v = lo.unpack_cpu(qkv)
    T = q.shape[0]
    o = torch.empty(T, lo.H * lo.D, dtype=qkv.dtype)
    lse = torch.empty(lo.H, T, dtype=torch.float32)
    cul = cu.tolist()
    for i in range(len(cul) - 1):
        s, e = cul[i], cul[i + 1]
        if e == s:
            continue
        qi = q[s:e].float()
        ki = k[s:e].float()
        vi = v[s:e].float()
        L = e - s
        mask = torch.ones(L, L, dtype=torch.bool).tril()
        for h in range(lo.H):
            j = h // (lo.H // lo.Hkv) if lo.Hkv > 1 else 0
            sc = (qi[:, h] @ ki[:, j].T) * scale
            sc = sc.masked_fill(~mask, float("-inf"))
            m = sc.max(-1, keepdim=True).values
            p = (sc - m).exp()
            l = p.sum(-1, keepdim=True)
            lse[h, s:e] = (m + l.log()).squeeze(-1)
            o[s:e, h * lo.D : (h + 1) * lo.D] = ((p / l) @ vi[:, j]).to(qkv.dtype)
    return o, lse


class VarlenAttention(torch.autograd.Function):
    """Causal varlen attention, the flash_attn_varlen_func replacement
    (padding_free.py:51-62). Input: packed qkv (T, row_len); output
    o (T, H*D). LSE saved for the recompute backward."""

    @staticmethod
    def forward(ctx, qkv, cu_seqlens, max_seqlen: int, layout: QKVLayout, scale: float):
        lo = layout
        T = qkv.shape[0]
        batch = cu_seqlens.shape[0] - 1
        if qkv.is_cuda:
            qkv = qkv.contiguous()
            o = torch.empty(T, lo.H * lo.D, dtype=qkv.dtype, device=qkv.device)
            lse = torch.empty(lo.H, T, dtype=torch.float32, device=qkv.device)
            with hip.prof("fa_varlen_fwd"):
             hip.check(
                hip.lib().dolomite_fa_varlen_fwd(
                    hip.stream(),
                    hip.ptr(qkv), hip.ptr(qkv, lo.k_off), hip.ptr(qkv, lo.v_off),
                    hip.ptr(o), hip.ptr(lse), hip.ptr(cu_seqlens),
                    batch, int(max_seqlen), T, lo.H, lo.Hkv, lo.D, lo.G,
                    lo.row_len, lo.q_gstride, lo.row_len, lo.kv_hstride,
                    lo.row_len, lo.kv_hstride, float(scale), hip.dt(qkv),
                ),
                "fa_varlen_fwd",
            )
        else:
            o, lse = _attn_fwd_cpu(qkv, cu_seqlens, scale, lo)
        ctx.save_for_backward(qkv, o, lse, cu_seqlens)
        ctx.layout = lo
        ctx.scale = scale
        ctx.max_seqlen = int(max_seqlen)
        return o

    @staticmethod
    def backward(ctx, dout):
        qkv, o, lse, cu = ctx.saved_tensors
        lo: QKVLayout = ctx.layout
        T = qkv.shape[0]
        batch = cu.shape[0] - 1
        if qkv.is_cuda:
            dout = dout.contiguous()
            delta = torch.empty(lo.H, T, dtype=torch.float32, device=qkv.device)
            hip.check(
                hip.lib().dolomite_fa_bwd_preprocess(
                    hip.stream(), hip.ptr(o), hip.ptr(dout), hip.ptr(delta),
                    T, lo.H, lo.D, lo.H * lo.D, lo.H * lo.D, hip.dt(qkv),
                ),
                "fa_bwd_preprocess",
            )
            # (T, H, D) per-q-head partials, written exclusively per
            # workgroup (no zero-init needed, every valid slot is stored)
            dk_acc = torch.empty(T, lo.H, lo.D, dtype=torch.float32, device=qkv.device)
            dv_acc = torch.empty(T, lo.H, lo.D, dtype=torch.float32, device=qkv.device)
            dqkv = torch.empty_like(qkv)
            with hip.prof("fa_varlen_bwd"):
             hip.check(
                hip.lib().dolomite_fa_varlen_bwd(
                    hip.stream(),
                    hip.ptr(qkv), hip.ptr(qkv, lo.k_off), hip.ptr(qkv, lo.v_off),
                    hip.ptr(dout), hip.ptr(lse), hip.ptr(delta), hip.ptr(dqkv),
                    hip.ptr(dk_acc), hip.ptr(dv_acc),
                    hip.ptr(cu), batch, ctx.max_seqlen, T, lo.H, lo.Hkv, lo.D, lo.G,
                    lo.row_len, lo.q_gstride, lo.row_len, lo.kv_hstride,
                    lo.row_len, lo.kv_hstride, lo.H * lo.D, float(ctx.scale), hip.dt(qkv),
                ),
                "fa_varlen_bwd",
            )
            hip.check(
                hip.lib().dolomite_fa_grad_finalize(
                    hip.stream(), hip.ptr(dk_acc), hip.ptr(dv_acc), hip.ptr(dqkv),
                    T, lo.Hkv, lo.D, lo.G, lo.row_len,
                    lo.k_off, lo.kv_hstride, lo.v_off, hip.dt(qkv),
                ),
                "fa_grad_finalize",
            )
            return dqkv, None, None, None, None
        # CPU: differentiate the CPU restatement with torch autograd
        qkv_l = qkv.detach().requires_grad_(True)
        with torch.enable_grad():
            o2, _ = _attn_fwd_cpu_autograd(qkv_l, cu, ctx.scale, lo)
            (dqkv,) = torch.autograd.grad(o2, qkv_l, dout)
        return dqkv, None, None, None, None


def _attn_fwd_cpu_autograd(qkv, cu, scale, lo: QKVLayout):
    q, k, v = lo.unpack_cpu(qkv)
    T = q.shape[0]
    outs = []
    cul = cu.tolist()
    pieces = []
    for i in range(len(cul) - 1):
        s, e = cul[i], cul[i + 1]
        if e == s:
            continue
        qi = q[s:e].float()
        ki = k[s:e].float()
        vi = v[s:e].float()
        L = e - s
        mask = torch.ones(L, L, dtype=torch.bool).tril()
        hs = []
        for h in range(lo.H):
            j = h // (lo.H // lo.Hkv) if lo.Hkv > 1 else 0
            sc = (qi[:, h] @ ki[:, j].T) * scale
            sc = sc.masked_fill(~mask, float("-inf"))
            p = torch.softmax(sc, dim=-1)
            hs.append(p @ vi[:, j])
        pieces.append(torch.cat(hs, dim=-1).to(qkv.dtype))
    o = torch.cat(pieces, dim=0)
    return o, None


def varlen_attention(qkv, cu_seqlens, max_seqlen, layout: QKVLayout, scale: float):
    return VarlenAttention.apply(qkv, cu_seqlens, int(max_seqlen), layout, scale)


# ---------------------------------------------------------------------------
# Fused cross entropy (mean over non-ignored rows)
# ---------------------------------------------------------------------------


class FusedCrossEntropy(torch.autograd.Function):
    """F.cross_entropy(mean, ignore_index=-100) replacement
    (model_wrapper/pretraining.py:125 / gpt_dolomite/main.py:200).
    Backward recomputes softmax from logits + saved lse (no softmax tensor
    materialized in forward)."""

    @staticmethod
    def forward(ctx, logits, labels, ignore_index: int = -100):
        T, V = logits.shape
        if logits.is_cuda:
            row_stride = logits.stride(0)
            assert logits.stride(1) == 1
            row_loss = torch.empty(T, dtype=torch.float32, device=logits.device)
            lse = torch.empty(T, dtype=torch.float32, device=logits.device)
            with hip.prof("ce_fwd"):
             hip.check(
                hip.lib().dolomite_ce_fwd(
                    hip.stream(), hip.ptr(logits), hip.ptr(labels), hip.ptr(row_loss),
                    hip.ptr(lse), T, V, row_stride, ignore_index, hip.dt(logits),
                ),
                "ce_fwd",
            )
        else:
            l32 = logits.float()
            lse = torch.logsumexp(l32, dim=-1)
            valid = labels != ignore_index
            picked = l32.gather(-1, labels.clamp_min(0).unsqueeze(-1)).squeeze(-1)
            row_loss = (lse - picked) * valid
        n_valid = (labels != ignore_index).sum()
        loss = row_loss.sum() / n_valid
        ctx.save_for_backward(logits, labels, lse, n_valid)
        ctx.ignore_index = ignore_index
        return loss

    @staticmethod
    def backward(ctx, gout):
        logits, labels, lse, n_valid = ctx.saved_tensors
        T, V = logits.shape
        if logits.is_cuda:
            # device-side scale: no host readback of gout/n_valid
            gs_dev = (gout.float() / n_valid.float()).reshape(1).contiguous()
            dlogits = torch.empty_like(logits)
            with hip.prof("ce_bwd"):
             hip.check(
                hip.lib().dolomite_ce_bwd(
                    hip.stream(), hip.ptr(logits), hip.ptr(labels), hip.ptr(lse),
                    hip.ptr(dlogits), hip.ptr(gs_dev), T, V, logits.stride(0),
                    ctx.ignore_index, hip.dt(logits),
                ),
                "ce_bwd",
            )
        else:
            gs = float(gout) / float(n_valid)
            p = torch.softmax(logits.float(), dim=-1)
            onehot = torch.zeros_like(p)
            onehot.scatter_(-1, labels.clamp_min(0).unsqueeze(-1), 1.0)
            valid = (labels != ctx.ignore_index).unsqueeze(-1)
            dlogits = ((p - onehot) * valid * gs).to(logits.dtype)
        return dlogits, None, None


def fused_cross_entropy(logits, labels, ignore_index: int = -100):
    return FusedCrossEntropy.apply(logits, labels, ignore_index)


# ---------------------------------------------------------------------------
# Fused AdamW on a flat fp32 master shard (not an autograd op)
# ---------------------------------------------------------------------------


def adamw_step_flat(master, grad, exp_avg, exp_avg_sq, step, lr, beta1, beta2, eps, weight_decay, param_out=None,
                    grad_scale=None):
    """torch.optim.AdamW semantics on flat fp32 tensors, optional bf16
    write-out of the updated params (the ZeRO-2 local shard step).
    grad may be bf16 (the flat autograd bucket) — the kernel upcasts per
    element. grad_scale: optional 0-dim DEVICE tensor (the grad-clip
    coefficient) fused into the kernel's grad read."""
    if master.is_cuda:
        hip.check(
            hip.lib().dolomite_adamw_step(
                hip.stream(), hip.ptr(master), hip.ptr(param_out), hip.ptr(grad),
                hip.dt(grad), hip.ptr(exp_avg), hip.ptr(exp_avg_sq),
                master.numel(), float(lr), float(beta1), float(beta2), float(eps),
                float(weight_decay), int(step), hip.ptr(grad_scale),
            ),
            "adamw_step",
        )
        return
    g = grad.float()
    if grad_scale is not None:
        g = g * float(grad_scale)
    master.mul_(1 - lr * weight_decay)
    exp_avg.mul_(beta1).add_(g, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    bc1 = 1 - beta1**step
    bc2 = 1 - beta2**step
    denom = (exp_avg_sq / bc2).sqrt().add_(eps)
    master.addcdiv_(exp_avg, denom, value=-(lr / bc1))
    if param_out is not None:
        param_out.copy_(master.to(param_out.dtype))


# ---------------------------------------------------------------------------
# Grouped expert GEMM (MoE) — replaces the reference's scattermoe grouped
# path (moe_dolomite/moe/scatter.py:109-138) for the SparseMoE expert
# matmuls (moe/base.py:137-156). Input rows are expert-sorted; `offsets`
# is the (E+1,) int32 cumulative group boundary array on device.
# ---------------------------------------------------------------------------


class GroupedExpertGemm(torch.autograd.Function):
    """y[t] = x[t] @ W[e(t)]^T (+ b[e(t)]); one HIP launch over all experts.

    backward: dX via the dgrad kernel, dW via the wgrad kernel, db via
    per-expert segment sums (E is small)."""

    @staticmethod
    def forward(ctx, x, weight, bias, offsets, max_rows):
        E, N, K = weight.shape
        y = torch.empty(x.shape[0], N, dtype=x.dtype, device=x.device)
        with hip.prof("moe_gemm_fwd"):
            hip.check(
                hip.lib().dolomite_moe_gemm_fwd(
                    hip.stream(), hip.ptr(x), hip.ptr(weight),
                    hip.ptr(bias) if bias is not None else None, hip.ptr(y),
                    hip.ptr(offsets), E, int(max_rows), N, K, hip.dt(x),
                ),
                "moe_gemm_fwd",
            )
        ctx.save_for_backward(x, weight, offsets)
        ctx.max_rows = int(max_rows)
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, offsets = ctx.saved_tensors
        E, N, K = weight.shape
        dy = dy.contiguous()
        dx = torch.empty_like(x)
        with hip.prof("moe_gemm_dgrad"):
            hip.check(
                hip.lib().dolomite_moe_gemm_dgrad(
                    hip.stream(), hip.ptr(dy), hip.ptr(weight), hip.ptr(dx),
                    hip.ptr(offsets), E, ctx.max_rows, N, K, hip.dt(dy),
                ),
                "moe_gemm_dgrad",
            )
        dw = torch.empty_like(weight)
        with hip.prof("moe_gemm_wgrad"):
            hip.check(
                hip.lib().dolomite_moe_gemm_wgrad(
                    hip.stream(), hip.ptr(dy), hip.ptr(x), hip.ptr(dw),
                    hip.ptr(offsets), E, N, K, hip.dt(dy),
                ),
                "moe_gemm_wgrad",
            )
        db = None
        if ctx.has_bias:
            off = offsets.tolist()
            db = torch.stack([dy[off[e]:off[e + 1]].sum(0) for e in range(E)])
        return dx, dw, db, None, None


class MoERowsGather(torch.autograd.Function):
    """expert_inputs = x[batch_index] with a deterministic, roofline-rate
    backward: torch's indexing backward for this pattern (indexing_backward
    / indexFuncLargeIndex) runs ~9x off the HBM roofline; the combine
    kernel sums each token's top-k slot rows in FIXED order instead
    (moe/base.py:108-127 semantics)."""

    @staticmethod
    def forward(ctx, x, batch_index, inv, k_top):
        ctx.save_for_backward(batch_index, inv)
        ctx.k_top = k_top
        ctx.total_q = x.shape[0]
        return x[batch_index]

    @staticmethod
    def backward(ctx, dexp):
        batch_index, inv = ctx.saved_tensors
        dx = _rows_combine(dexp, inv, batch_index, ctx.total_q, ctx.k_top)
        return dx, None, None, None


class MoERowsCombine(torch.autograd.Function):
    """out = zeros(total_q, K).index_add(0, batch_index, h)
    (moe/base.py:127) with the deterministic combine kernel on the forward
    and a plain gather backward."""

    @staticmethod
    def forward(ctx, h, inv, batch_index, total_q, k_top):
        ctx.save_for_backward(batch_index)
        return _rows_combine(h, inv, batch_index, total_q, k_top)

    @staticmethod
    def backward(ctx, dout):
        (batch_index,) = ctx.saved_tensors
        return dout[batch_index], None, None, None, None


def _rows_combine(h, inv, batch_index, total_q, k_top):
    h = h.contiguous()
    if h.is_cuda and h.dtype == torch.bfloat16 and h.shape[1] % 8 == 0:
        out = torch.empty(total_q, h.shape[1], dtype=h.dtype, device=h.device)
        hip.check(
            hip.lib().dolomite_moe_rows_combine(
                hip.stream(), hip.ptr(h), hip.ptr(inv), hip.ptr(out),
                total_q, h.shape[1], k_top, hip.dt(h),
            ),
            "moe_rows_combine",
        )
        return out
    return torch.zeros(total_q, h.shape[1], dtype=h.dtype, device=h.device).index_add(0, batch_index, h)


def grouped_expert_gemm(x, weight, bias, num_tokens_per_expert):
    """Dispatch helper: HIP grouped kernel on CUDA bf16 with 8-aligned dims
    (one launch for all experts); returns None if unsupported so the caller
    can fall back to the eager per-expert loop."""
    E, N, K = weight.shape
    if not (x.is_cuda and x.dtype == torch.bfloat16 and K % 8 == 0 and N % 8 == 0):
        return None
    # Measured crossover (tools_moe_gemm_probe.py, 1xMI355X): the one-launch
    # grouped kernel wins up to ~6x when per-expert GEMMs are small/launch-
    # bound (E=32, 512 rows/expert: 342 vs 54 TF), while the per-expert
    # rocBLAS loop wins for few large experts (E=8, 4096 rows/expert:
    # 630 vs 373 TF) — route the large-expert regime to the eager loop.
    if E <= 16 and x.shape[0] // max(E, 1) >= 2048:
        return None
    offsets = torch.zeros(E + 1, dtype=torch.int32, device=x.device)
    offsets[1:] = num_tokens_per_expert.cumsum(0).to(torch.int32)
    max_rows = int(num_tokens_per_expert.max())
    if max_rows == 0:
        return torch.zeros(x.shape[0], N, dtype=x.dtype, device=x.device)
    return GroupedExpertGemm.apply(x.contiguous(), weight, bias, offsets, max_rows)
