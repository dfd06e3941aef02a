"""CPU restatement (fp32 PyTorch) of the reference GPTDolomite hot path.

TEST INFRASTRUCTURE ONLY — see oracle/__init__.py header.

Every function cites the reference file:line (paths relative to
/root/reference/dolomite_engine/) whose algorithm it restates. This is a
restatement, not a copy: the reference's module structure (HF PreTrainedModel,
registries, TP hooks) is dropped; only the arithmetic is preserved.
"""

import math
from dataclasses import dataclass, field

import torch
import torch.nn as nn
import torch.nn.functional as F


# ---------------------------------------------------------------------------
# Per-op references
# ---------------------------------------------------------------------------


def rmsnorm_ref(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    """RMSNorm, reference semantics: fp32 upcast, rsqrt(mean(x^2)+eps),
    cast back to input dtype BEFORE multiplying by weight.
    Restates hf_models/modeling_utils/normalization/rmsnorm/base.py:18-25."""
    input_dtype = x.dtype
    x32 = x.to(torch.float32)
    variance = x32.pow(2).mean(-1, keepdim=True)
    x32 = x32 * torch.rsqrt(variance + eps)
    return weight * x32.to(input_dtype)


def layernorm_ref(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor, eps: float) -> torch.Tensor:
    """LayerNorm as the reference uses it (torch nn.LayerNorm;
    hf_models/modeling_utils/normalization/__init__.py 'layernorm'/'torch')."""
    return F.layer_norm(x, (x.shape[-1],), weight, bias, eps)


def rope_cos_sin_ref(
    head_dim: int, seq_len: int, base: float = 10000.0, dtype: torch.dtype = torch.float32
) -> tuple[torch.Tensor, torch.Tensor]:
    """RoPE cos/sin cache. Restates position_embedding/rope.py:37-56:
    inv_freq over even dims, outer product with positions, emb = cat(freqs, freqs),
    tables computed in fp32 then cast to the model dtype."""
    inv_freq = 1.0 / (base ** (torch.arange(0, head_dim, 2, dtype=torch.float32) / head_dim))
    t = torch.arange(seq_len, dtype=torch.float32)
    freqs = torch.outer(t, inv_freq)
    emb = torch.cat((freqs, freqs), dim=-1)
    return emb.cos().to(dtype), emb.sin().to(dtype)


def apply_rope_ref(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
    """x*cos + rotate_half(x)*sin, half-split (NeoX) convention.
    Restates position_embedding/rope.py:104-121."""
    x1, x2 = torch.chunk(x, 2, dim=-1)
    rotated = torch.cat((-x2, x1), dim=-1)
    return (x * cos) + (rotated * sin)


def attention_eager_dense_ref(
    q: torch.Tensor,  # (B, H, S, D)
    k: torch.Tensor,  # (B, H_kv, S, D)
    v: torch.Tensor,  # (B, H_kv, S, D)
    softmax_scale: float,
    causal: bool = True,
    attention_bias: torch.Tensor | None = None,  # (B, 1|H, S, S) additive, float
    softmax_in_fp32: bool = True,
) -> torch.Tensor:
    """Dense eager attention. Restates modeling_utils/attention/base.py:171-269:
    repeat kv heads (utils.py:109-118), scaled QK^T (baddbmm), causal mask to
    finfo.min (gpt_dolomite/base.py:553-557), softmax in fp32, cast back, PV."""
    B, H, S, D = q.shape
    H_kv = k.shape[1]
    if H_kv != H:
        if H_kv == 1:
            k = k.expand(-1, H, -1, -1)
            v = v.expand(-1, H, -1, -1)
        else:
            k = k.repeat_interleave(H // H_kv, dim=1)
            v = v.repeat_interleave(H // H_kv, dim=1)

    dtype = q.dtype
    scores = torch.matmul(q, k.transpose(-1, -2)) * softmax_scale
    if attention_bias is not None:
        scores = scores + attention_bias
    if causal:
        mask = torch.ones(S, S, dtype=torch.bool, device=q.device).tril()
        scores = scores.masked_fill(~mask, torch.finfo(scores.dtype).min)
    sm_dtype = torch.float32 if softmax_in_fp32 else dtype
    p = F.softmax(scores.to(sm_dtype), dim=-1).to(dtype)
    return torch.matmul(p, v)


def attention_varlen_ref(
    q: torch.Tensor,  # (T, H, D) packed
    k: torch.Tensor,  # (T, H_kv, D)
    v: torch.Tensor,  # (T, H_kv, D)
    cu_seqlens: torch.Tensor,  # (B+1,) int32
    softmax_scale: float,
    causal: bool = True,
) -> torch.Tensor:
    """Packed varlen causal attention: the algorithm flash_attn_varlen_func
    computes at the call site attention/padding_free.py:51-62 — independent
    causal attention per sequence slice [cu_seqlens[i], cu_seqlens[i+1]).
    Softmax accumulated in fp32 (flash-attn semantics)."""
    T, H, D = q.shape
    out = torch.empty_like(q)
    cu = cu_seqlens.tolist()
    for i in range(len(cu) - 1):
        s, e = cu[i], cu[i + 1]
        if e == s:
            continue
        qi = q[s:e].transpose(0, 1).unsqueeze(0)  # (1, H, si, D)
        ki = k[s:e].transpose(0, 1).unsqueeze(0)
        vi = v[s:e].transpose(0, 1).unsqueeze(0)
        oi = attention_eager_dense_ref(qi, ki, vi, softmax_scale, causal=causal, softmax_in_fp32=True)
        out[s:e] = oi.squeeze(0).transpose(0, 1)
    return out


def cross_entropy_ref(logits: torch.Tensor, labels: torch.Tensor, ignore_index: int = -100) -> torch.Tensor:
    """Mean cross entropy over non-ignored rows, fp32 log-softmax — what
    F.cross_entropy computes at model_wrapper/pretraining.py:125 and
    gpt_dolomite/main.py:200."""
    logits = logits.to(torch.float32)
    valid = labels != ignore_index
    lse = torch.logsumexp(logits, dim=-1)
    picked = logits.gather(-1, labels.clamp_min(0).unsqueeze(-1)).squeeze(-1)
    losses = (lse - picked) * valid
    n = valid.sum()
    return losses.sum() / n


def softmax_cross_entropy_fwd_bwd_ref(
    logits: torch.Tensor, labels: torch.Tensor, ignore_index: int = -100, grad_scale: float | None = None
) -> tuple[torch.Tensor, torch.Tensor]:
    """Fused CE forward+backward reference: loss (mean over valid rows) and
    dlogits = (softmax - onehot) * grad_scale/num_valid in logits dtype.
    grad_scale defaults to 1 (d loss/d logits for mean reduction)."""
    orig_dtype = logits.dtype
    l32 = logits.to(torch.float32)
    valid = labels != ignore_index
    n = valid.sum()
    lse = torch.logsumexp(l32, dim=-1)
    picked = l32.gather(-1, labels.clamp_min(0).unsqueeze(-1)).squeeze(-1)
    loss = ((lse - picked) * valid).sum() / n
    p = torch.softmax(l32, dim=-1)
    onehot = torch.zeros_like(p)
    onehot.scatter_(-1, labels.clamp_min(0).unsqueeze(-1), 1.0)
    g = (grad_scale if grad_scale is not None else 1.0) / n
    dlogits = (p - onehot) * valid.unsqueeze(-1) * g
    return loss, dlogits.to(orig_dtype)


def lm_loss_padding_free_ref(
    logits: torch.Tensor,  # (T, V)
    labels: torch.Tensor,  # (T,)
    cu_seqlens: torch.Tensor,  # (B+1,) int32
    upcast_logits_for_loss: bool = False,
) -> torch.Tensor:
    """Model-internal padding-free loss: shift by one, drop positions at
    sequence boundaries so the last token of a sequence doesn't predict the
    first of the next. Restates gpt_dolomite/main.py:185-202."""
    shift_logits = logits[:-1, :]
    shift_labels = labels[1:].clone()
    drop_loss_positions = cu_seqlens[1:-1].long() - 1
    shift_labels[drop_loss_positions] = -100
    if upcast_logits_for_loss:
        shift_logits = shift_logits.float()
    return cross_entropy_ref(shift_logits, shift_labels)


def adamw_step_ref(
    param: torch.Tensor,
    grad: torch.Tensor,
    exp_avg: torch.Tensor,
    exp_avg_sq: torch.Tensor,
    step: int,
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_decay: float,
) -> None:
    """One AdamW step, torch.optim.AdamW semantics (the reference's default
    TorchAdamW, optimization/optimizer.py:74; defaults lr 1e-5, wd 0.1,
    betas (0.9, 0.95), eps 1e-10 — arguments.py:235-249). In-place, fp32.
    Decoupled weight decay: p *= (1 - lr*wd) before the Adam update."""
    param.mul_(1 - lr * weight_decay)
    exp_avg.mul_(beta1).add_(grad, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
    bias_correction1 = 1 - beta1**step
    bias_correction2 = 1 - beta2**step
    step_size = lr / bias_correction1
    denom = (exp_avg_sq / bias_correction2).sqrt().add_(eps)
    param.addcdiv_(exp_avg, denom, value=-step_size)


# ---------------------------------------------------------------------------
# Full-model oracle
# ---------------------------------------------------------------------------


@dataclass
class OracleConfig:
    """Resolved GPTDolomite config fields (hf_models/config.py:15-110)."""

    vocab_size: int = 50257
    n_positions: int = 1024
    n_embd: int = 768
    n_layer: int = 12
    n_head: int = 12
    num_key_value_heads: int | None = None
    n_inner: int | None = None
    activation_function: str = "gelu_pytorch_tanh"
    attention_head_type: str = "mqa"
    normalization_function: str = "layernorm"
    layer_norm_epsilon: float = 1e-5
    initializer_range: float = 0.02
    scale_attn_weights: bool = True
    attention_multiplier: float | None = None
    attention_softmax_in_fp32: bool = True
    add_bias: bool = True
    position_embedding_type: str = "learned_absolute"
    rope_theta: float = 10000
    m_emb: float | None = None
    m_width: float | None = None
    m_residual: float | None = None
    upcast_logits_for_loss: bool = False
    tie_word_embeddings: bool = False

    def __post_init__(self):
        if self.n_inner is None:
            self.n_inner = 4 * self.n_embd
        if self.attention_head_type == "mha" and self.num_key_value_heads is None:
            self.num_key_value_heads = self.n_head
        elif self.attention_head_type == "mqa" and self.num_key_value_heads is None:
            self.num_key_value_heads = 1
        assert self.n_embd % self.n_head == 0

    @property
    def head_dim(self) -> int:
        return self.n_embd // self.n_head

    def softmax_scale(self) -> float:
        """attention/base.py:277-286."""
        if self.scale_attn_weights:
            if self.attention_multiplier is None:
                return 1.0 / self.head_dim**0.5
            return self.attention_multiplier
        return 1.0


def _is_glu(name: str) -> bool:
    """activations/glu.py:50."""
    return name.endswith("glu")


def _base_activation(name: str):
    """Subset of activations/base.py + glu.py:7-18 mapping used on the hot
    path (gelu_pytorch_tanh and swiglu cover every named config)."""
    glu_map = {"swiglu": "swish", "geglu": "gelu", "reglu": "relu"}
    if _is_glu(name):
        name = glu_map.get(name, name[: -len("_glu")] if name.endswith("_glu") else name)
    if name == "gelu_pytorch_tanh":
        return lambda x: F.gelu(x, approximate="tanh")
    if name == "gelu":
        return F.gelu
    if name in ("swish", "silu"):
        return F.silu
    if name == "relu":
        return F.relu
    raise ValueError(f"oracle does not restate activation {name}")


class _OracleNorm(nn.Module):
    def __init__(self, kind: str, hidden: int, eps: float):
        super().__init__()
        self.kind = kind
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(hidden))
        if kind == "layernorm":
            self.bias = nn.Parameter(torch.zeros(hidden))

    def forward(self, x):
        if self.kind == "rmsnorm":
            return rmsnorm_ref(x, self.weight, self.eps)
        return layernorm_ref(x, self.weight, self.bias, self.eps)


class _OracleAttention(nn.Module):
    """Attention block mirroring attention/base.py — fused c_attn with
    [q | k | v] layout per head type (base.py:72-81), c_proj."""

    def __init__(self, cfg: OracleConfig):
        super().__init__()
        self.cfg = cfg
        h, kv, d = cfg.n_embd, cfg.num_key_value_heads, cfg.head_dim
        self.c_attn = nn.Linear(h, h + 2 * kv * d, bias=cfg.add_bias)
        self.c_proj = nn.Linear(h, h, bias=cfg.add_bias)

    def split_qkv(self, hs: torch.Tensor):
        """QKV split on packed (T, ·) layout — attention/padding_free.py:79-116.
        Returns (T, H, D), (T, H_kv, D), (T, H_kv, D)."""
        cfg = self.cfg
        T = hs.shape[0]
        H, Hkv, D = cfg.n_head, cfg.num_key_value_heads, cfg.head_dim
        if cfg.attention_head_type == "mha":
            hs = hs.view(T, Hkv, -1)
            q, k, v = hs.chunk(3, dim=-1)
        elif cfg.attention_head_type == "gqa":
            hs = hs.view(T, Hkv, -1)
            q, k, v = hs.split(((H // Hkv) * D, D, D), dim=-1)
            q = q.reshape(T, -1, D)
        else:  # mqa
            q, k, v = hs.split((cfg.n_embd, D, D), dim=-1)
            q = q.view(T, H, D)
            k = k.unsqueeze(1)
            v = v.unsqueeze(1)
        return q, k, v

    def forward(self, hs, rope_cos_sin, cu_seqlens):
        cfg = self.cfg
        q, k, v = self.split_qkv(self.c_attn(hs))
        if cfg.position_embedding_type == "rope":
            cos, sin = rope_cos_sin  # (T, 1, D)
            q = apply_rope_ref(q, cos, sin)
            k = apply_rope_ref(k, cos, sin)
        o = attention_varlen_ref(q, k, v, cu_seqlens, cfg.softmax_scale(), causal=True)
        return self.c_proj(o.reshape(hs.shape[0], cfg.n_embd))


class _OracleMLP(nn.Module):
    """gpt_dolomite/mlp.py:12-51 — fused up+gate c_fc when GLU."""

    def __init__(self, cfg: OracleConfig):
        super().__init__()
        glu = _is_glu(cfg.activation_function)
        self.c_fc = nn.Linear(cfg.n_embd, 2 * cfg.n_inner if glu else cfg.n_inner, bias=cfg.add_bias)
        self.c_proj = nn.Linear(cfg.n_inner, cfg.n_embd, bias=cfg.add_bias)
        self.glu = glu
        self.act = _base_activation(cfg.activation_function)

    def forward(self, x):
        x = self.c_fc(x)
        if self.glu:
            a, b = x.chunk(2, dim=-1)  # activations/glu.py:26-28: x0 * act(x1)
            x = a * self.act(b)
        else:
            x = self.act(x)
        return self.c_proj(x)


class _OracleBlock(nn.Module):
    """gpt_dolomite/layer.py:49-87: ln_1 → attn → +res, ln_2 → mlp → +res,
    optional m_residual scaling."""

    def __init__(self, cfg: OracleConfig):
        super().__init__()
        self.cfg = cfg
        self.ln_1 = _OracleNorm(cfg.normalization_function, cfg.n_embd, cfg.layer_norm_epsilon)
        self.attn = _OracleAttention(cfg)
        self.ln_2 = _OracleNorm(cfg.normalization_function, cfg.n_embd, cfg.layer_norm_epsilon)
        self.mlp = _OracleMLP(cfg)

    def forward(self, hs, rope_cos_sin, cu_seqlens):
        m_res = self.cfg.m_residual
        attn_out = self.attn(self.ln_1(hs), rope_cos_sin, cu_seqlens)
        if m_res is not None:
            attn_out = attn_out * m_res
        hs = attn_out + hs
        mlp_out = self.mlp(self.ln_2(hs))
        if m_res is not None:
            mlp_out = mlp_out * m_res
        return hs + mlp_out


class _OracleTransformer(nn.Module):
    def __init__(self, cfg: OracleConfig):
        super().__init__()
        self.cfg = cfg
        self.wte = nn.Embedding(cfg.vocab_size, cfg.n_embd)
        if cfg.position_embedding_type == "learned_absolute":
            self.wpe = nn.Embedding(cfg.n_positions, cfg.n_embd)
        self.h = nn.ModuleList([_OracleBlock(cfg) for _ in range(cfg.n_layer)])
        self.ln_f = _OracleNorm(cfg.normalization_function, cfg.n_embd, cfg.layer_norm_epsilon)

    def forward(self, input_ids, position_ids, cu_seqlens, max_seqlen):
        """gpt_dolomite/base.py:170-244, packed padding-free layout (T,)."""
        cfg = self.cfg
        hs = self.wte(input_ids)
        if cfg.position_embedding_type == "learned_absolute":
            hs = hs + self.wpe(position_ids)
        if cfg.m_emb is not None:
            hs = hs * cfg.m_emb
        rope_cos_sin = None
        if cfg.position_embedding_type == "rope":
            # base.py:289-296: tables to key_length, gathered per token, unsqueezed
            cos, sin = rope_cos_sin_ref(cfg.head_dim, int(max_seqlen), cfg.rope_theta, hs.dtype)
            rope_cos_sin = (cos[position_ids].unsqueeze(1), sin[position_ids].unsqueeze(1))
        for block in self.h:
            hs = block(hs, rope_cos_sin, cu_seqlens)
        return self.ln_f(hs)


class OracleGPTDolomiteForCausalLM(nn.Module):
    """Padding-free GPTDolomiteForCausalLM restatement (gpt_dolomite/main.py).

    Parameter names match the reference state_dict exactly so that
    state dicts move 1:1 between reference, oracle, and product model.
    Packed-tensor inputs only: input_ids (T,), position_ids (T,),
    cu_seqlens (B+1,) int32, max_seqlen scalar.
    """

    def __init__(self, cfg: OracleConfig):
        super().__init__()
        self.cfg = cfg
        self.transformer = _OracleTransformer(cfg)
        if not cfg.tie_word_embeddings:
            self.lm_head = nn.Linear(cfg.n_embd, cfg.vocab_size, bias=False)

    def forward(self, input_ids, position_ids, cu_seqlens, max_seqlen, labels=None):
        hs = self.transformer(input_ids, position_ids, cu_seqlens, max_seqlen)
        if self.cfg.tie_word_embeddings:
            logits = F.linear(hs, self.transformer.wte.weight)  # main.py:172-177
        else:
            logits = self.lm_head(hs)
        if self.cfg.m_width is not None:
            logits = logits / self.cfg.m_width  # main.py:160-161
        loss = None
        if labels is not None:
            loss = lm_loss_padding_free_ref(logits, labels, cu_seqlens, self.cfg.upcast_logits_for_loss)
        return logits, loss
