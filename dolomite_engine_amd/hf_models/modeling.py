"""GPTDolomite model — MI355X-native implementation of the reference's
drop-in API surface (gpt_dolomite/main.py, base.py, layer.py, mlp.py;
modeling_utils/attention/*). Three attention implementations:

  - "eager"  : dense math attention, fp32 softmax (attention/base.py:171-269)
  - "sdpa"   : torch scaled_dot_product_attention (attention/sdpa.py)
  - "flash_attention_2" + use_padding_free_transformer: THE hot path —
    packed (T,) layout through the hand-written gfx950 kernels
    (fused RMSNorm+residual, packed-QKV RoPE, varlen flash attention,
    fused cross-entropy) behind the C-ABI in include/dolomite_hip.h.

Forward signature, list-input handling, loss semantics (boundary drops) and
state-dict names match the reference exactly (drop-in boundary).
"""

import math

import torch
import torch.nn as nn
import torch.nn.functional as F
from transformers import PreTrainedModel
from transformers.modeling_outputs import BaseModelOutputWithPast, CausalLMOutputWithPast

from ..ops import (
    QKVLayout,
    fused_cross_entropy,
    fused_layernorm,
    fused_rmsnorm,
    rope_packed_qkv,
    varlen_attention,
)
from .config import GPTDolomiteConfig


# ---------------------------------------------------------------------------
# Parameterized modules (modeling_utils/linear.py, embedding.py)
# ---------------------------------------------------------------------------


class ParameterizedLinear(nn.Linear):
    def __init__(self, in_features, out_features, bias=True, device=None, dtype=None, std=None):
        self.std = std
        super().__init__(in_features, out_features, bias, device, dtype)

    def forward(self, input):
        # Training on GPU with a live ZeRO engine: route through the
        # deferred-wgrad linear so dW/db run on the engine's side stream,
        # off the backward critical path (zero.py DeferredWgradLinear).
        if self.training and input.is_cuda and torch.is_grad_enabled():
            from ..zero import DeferredWgradLinear, _WgradSink

            if _WgradSink.current is not None:
                return DeferredWgradLinear.apply(input, self.weight, self.bias)
        return super().forward(input)

    @torch.no_grad()
    def reset_parameters(self) -> None:
        if self.std is None:
            super().reset_parameters()
        else:
            nn.init.normal_(self.weight, mean=0, std=self.std)
            if getattr(self, "bias", None) is not None:
                self.bias.zero_()


class ParameterizedEmbedding(nn.Embedding):
    def __init__(self, num_embeddings, embedding_dim, std=None, **kwargs):
        self.std = std
        super().__init__(num_embeddings, embedding_dim, **kwargs)

    @torch.no_grad()
    def reset_parameters(self) -> None:
        if self.std is None:
            super().reset_parameters()
        else:
            self.weight.data.normal_(mean=0, std=self.std)
            if self.padding_idx is not None:
                self.weight.data[self.padding_idx].zero_()


# ---------------------------------------------------------------------------
# Normalization (fused HIP on GPU; same math on CPU)
# ---------------------------------------------------------------------------


class DolomiteNorm(nn.Module):
    """RMSNorm (rmsnorm/base.py:18-25) or LayerNorm ('torch' impl), with
    optional fused residual-add. forward(x, residual) -> (y, pre-norm sum)."""

    def __init__(self, kind: str, hidden_size: int, eps: float):
        super().__init__()
        if kind not in ("rmsnorm", "layernorm"):
            raise NotImplementedError(f"normalization_function {kind}")
        self.kind = kind
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(hidden_size))
        if kind == "layernorm":
            self.bias = nn.Parameter(torch.zeros(hidden_size))

    def reset_parameters(self) -> None:
        with torch.no_grad():
            self.weight.fill_(1.0)
            if self.kind == "layernorm":
                self.bias.zero_()

    def forward(self, x, residual=None):
        if self.kind == "rmsnorm":
            return fused_rmsnorm(x, self.weight, self.eps, residual)
        return fused_layernorm(x, self.weight, self.bias, self.eps, residual)


# ---------------------------------------------------------------------------
# RoPE tables (position_embedding/rope.py:7-56)
# ---------------------------------------------------------------------------


class RoPE(nn.Module):
    def __init__(self, head_dim: int, max_position_embeddings: int, base: float):
        super().__init__()
        self.head_dim = head_dim
        self.base = float(base)
        self.mscale = 1.0
        self.max_seq_len_cached = 0
        self.reset_parameters(max_position_embeddings)

    def _get_inv_freq(self, device) -> torch.Tensor:
        return 1.0 / (
            self.base ** (torch.arange(0, self.head_dim, 2, dtype=torch.float32, device=device) / self.head_dim)
        )

    def reset_parameters(self, seq_len: int | None = None) -> None:
        self._set_cache(seq_len or self.max_seq_len_cached, torch.device("cpu"))

    @torch.no_grad()
    def _set_cache(self, seq_len: int, device) -> None:
        # plain attributes, NOT registered buffers: transformers-5
        # from_pretrained materializes non-persistent buffers from the meta
        # device UNINITIALIZED (to_empty), which silently poisons the tables.
        # Plain attributes are untouched by the module system; device moves
        # and meta-context construction are handled by the recompute check.
        self.max_seq_len_cached = seq_len
        if device is not None and torch.device(device).type == "meta":
            device = "cpu"
        inv_freq = self._get_inv_freq(device)
        t = torch.arange(seq_len, dtype=torch.float32, device=device)
        freqs = torch.outer(t, inv_freq)
        emb = torch.cat((freqs, freqs), dim=-1)
        self.cos_cached = emb.cos() * self.mscale
        self.sin_cached = emb.sin() * self.mscale

    def forward(self, seq_len: int, device) -> tuple[torch.Tensor, torch.Tensor]:
        if seq_len > self.max_seq_len_cached or self.cos_cached.device != torch.device(device):
            self._set_cache(max(seq_len, self.max_seq_len_cached), device)
        return self.cos_cached[:seq_len], self.sin_cached[:seq_len]


class YaRNScaledRoPE(RoPE):
    """NTK-by-parts YaRN scaling (reference position_embedding/rope.py:59-142):
    per-dim blend of interpolated and extrapolated inverse frequencies with
    a linear ramp between the beta_fast/beta_slow correction dims, and the
    attention-magnitude mscale folded into the tables."""

    def __init__(self, head_dim: int, max_position_embeddings: int, base: float,
                 scale: float = 1.0, original_max_position_embeddings: int = 2048,
                 extrapolation_factor: float = 1.0, attn_factor: float = 1.0,
                 beta_fast: int = 32, beta_slow: int = 1):
        self.scale = float(scale)
        self.original_max_position_embeddings = original_max_position_embeddings
        self.extrapolation_factor = extrapolation_factor
        self.attn_factor = attn_factor
        self.beta_fast = beta_fast
        self.beta_slow = beta_slow
        super().__init__(head_dim, max_position_embeddings, base)
        self.mscale = (0.1 * math.log(self.scale) + 1.0 if self.scale > 1 else 1.0) * attn_factor
        # rebuild the cached tables with the final mscale (the base __init__
        # built them before mscale was known)
        self.reset_parameters(max_position_embeddings)

    def _get_inv_freq(self, device) -> torch.Tensor:
        dim = self.head_dim
        pos_freqs = self.base ** (torch.arange(0, dim, 2, dtype=torch.float32, device=device) / dim)
        inv_extrapolation = 1.0 / pos_freqs
        inv_interpolation = 1.0 / (self.scale * pos_freqs)

        def correction_dim(num_rotations):
            return (dim * math.log(self.original_max_position_embeddings / (num_rotations * 2 * math.pi))) / (
                2 * math.log(self.base)
            )

        low = max(math.floor(correction_dim(self.beta_fast)), 0)
        high = min(math.ceil(correction_dim(self.beta_slow)), dim - 1)
        if low == high:
            high += 0.001  # prevent singularity
        ramp = ((torch.arange(dim // 2, dtype=torch.float32, device=device) - low) / (high - low)).clamp(0, 1)
        inv_freq_mask = (1 - ramp) * self.extrapolation_factor
        return inv_interpolation * (1 - inv_freq_mask) + inv_extrapolation * inv_freq_mask


class Alibi(nn.Module):
    """Bloom-style alibi slopes and bias (reference
    position_embedding/alibi.py:8-45): bias[b, h, k] = slope_h * position_k
    (row-constant-shifted equivalent of slope * (k - q) under softmax)."""

    def __init__(self, num_heads: int):
        super().__init__()
        self.num_heads = num_heads
        self.reset_parameters()

    def reset_parameters(self) -> None:
        n = self.num_heads
        closest = 2 ** math.floor(math.log2(n))
        base = torch.tensor(2 ** (-(2 ** -(math.log2(closest) - 3))), dtype=torch.float32)
        slopes = torch.pow(base, torch.arange(1, 1 + closest, dtype=torch.int32))
        if closest != n:
            extra_base = torch.tensor(2 ** (-(2 ** -(math.log2(2 * closest) - 3))), dtype=torch.float32)
            extra = torch.pow(extra_base, torch.arange(1, 1 + 2 * min(closest, n - closest), 2, dtype=torch.int32))
            slopes = torch.cat([slopes, extra], dim=0)
        self.slopes = slopes  # plain attribute (see RoPE cache note)

    def forward(self, attention_mask, batch_size: int, key_length: int, device, dtype) -> torch.Tensor:
        slopes = self.slopes.to(device)
        if attention_mask is None:
            pos = torch.arange(key_length, device=device, dtype=torch.float32).unsqueeze(0).unsqueeze(0)
            pos = pos.expand(batch_size, -1, -1)
        else:
            pos = (attention_mask.cumsum(dim=-1) - 1).masked_fill_(attention_mask == 0, 0).unsqueeze(1)
        return (slopes.unsqueeze(0).unsqueeze(-1) * pos).to(dtype)  # (B, H, K)


def apply_rotary_dense(x, cos, sin):
    """x: (B, H, S, D); cos/sin: (B, 1, S, D) in x dtype (rope.py:104-121)."""
    x1, x2 = torch.chunk(x, 2, dim=-1)
    rotated = torch.cat((-x2, x1), dim=-1)
    return (x * cos) + (rotated * sin)


# ---------------------------------------------------------------------------
# Attention
# ---------------------------------------------------------------------------


class Attention(nn.Module):
    """Fused-c_attn attention (attention/base.py:16-112). The padding-free
    forward keeps QKV packed end-to-end (no unpack copies)."""

    def __init__(self, config: GPTDolomiteConfig, causal: bool, layer_idx: int | None = None):
        super().__init__()
        self.causal = causal
        self.hidden_size = config.n_embd
        self.num_heads = config.n_head
        self.num_key_value_heads = config.num_key_value_heads
        self.head_dim = config.head_dim
        self.attention_head_type = config.attention_head_type
        self.position_embedding_type = config.position_embedding_type
        self.layer_idx = layer_idx
        self.attention_softmax_in_fp32 = config.attention_softmax_in_fp32
        self.softmax_scale = config.softmax_scale()
        self.layout = QKVLayout.make(self.num_heads, self.num_key_value_heads, self.head_dim, self.attention_head_type)

        init_method = config.init_method
        initializer_range = config.initializer_range
        m_width = config.m_width
        std = initializer_range
        if init_method == "mup":
            std /= math.sqrt(m_width)
        self.c_attn = ParameterizedLinear(
            self.hidden_size,
            self.hidden_size + 2 * self.num_key_value_heads * self.head_dim,
            bias=config.add_bias,
            std=std,
        )
        std = initializer_range / math.sqrt(2 * config.n_layer)
        if init_method == "mup":
            std /= math.sqrt(m_width)
        self.c_proj = ParameterizedLinear(self.hidden_size, self.hidden_size, bias=config.add_bias, std=std)

        self.attn_pdrop = config.attn_pdrop
        self.attn_dropout = nn.Identity() if config.attn_pdrop == 0 else nn.Dropout(config.attn_pdrop)
        self.resid_dropout = nn.Identity() if config.resid_pdrop == 0 else nn.Dropout(config.resid_pdrop)

    # ---- padding-free packed path (the hot path) ----
    def forward_padding_free(self, hidden_states, rope_cos_sin, cu_seqlens, max_seqlen):
        if self.training and self.attn_pdrop > 0:
            raise NotImplementedError("attention dropout > 0 is not supported on the padding-free path")
        qkv = self.c_attn(hidden_states)
        if self.position_embedding_type == "rope":
            cos, sin = rope_cos_sin  # (T, D) fp32
            qkv = rope_packed_qkv(qkv, cos, sin, self.layout)
        attn_output = varlen_attention(qkv, cu_seqlens, max_seqlen, self.layout, self.softmax_scale)
        attn_output = self.c_proj(attn_output)
        return self.resid_dropout(attn_output)

    # ---- dense eager / sdpa paths ----
    def _split_dense(self, qkv):
        B, S = qkv.shape[:2]
        H, Hkv, D = self.num_heads, self.num_key_value_heads, self.head_dim
        if self.attention_head_type == "mha":
            h = qkv.view(B, S, H, 3 * D).transpose(1, 2)
            q, k, v = h.chunk(3, dim=-1)
        elif self.attention_head_type == "gqa":
            h = qkv.view(B, S, Hkv, -1)
            G = H // Hkv
            q, k, v = h.split((G * D, D, D), dim=-1)
            q = q.reshape(B, S, H, D).transpose(1, 2)
            k = k.transpose(1, 2)
            v = v.transpose(1, 2)
        else:
            q, k, v = qkv.split((H * D, D, D), dim=-1)
            q = q.view(B, S, H, D).transpose(1, 2)
            k = k.unsqueeze(1)
            v = v.unsqueeze(1)
        return q, k, v

    def forward_dense(self, hidden_states, attention_bias, rope_cos_sin, implementation):
        B, S, _ = hidden_states.shape
        H, Hkv, D = self.num_heads, self.num_key_value_heads, self.head_dim
        q, k, v = self._split_dense(self.c_attn(hidden_states))
        if self.position_embedding_type == "rope":
            cos, sin = rope_cos_sin  # (B, 1, S, D) model dtype
            q = apply_rotary_dense(q, cos, sin)
            k = apply_rotary_dense(k, cos, sin)

        if Hkv != H:
            if Hkv == 1:
                k = k.expand(-1, H, -1, -1)
                v = v.expand(-1, H, -1, -1)
            else:
                k = k.repeat_interleave(H // Hkv, dim=1)
                v = v.repeat_interleave(H // Hkv, dim=1)

        if implementation == "sdpa":
            out = F.scaled_dot_product_attention(
                q,
                k,
                v,
                attn_mask=attention_bias,
                dropout_p=self.attn_pdrop if self.training else 0,
                is_causal=self.causal and attention_bias is None and S > 1,
                scale=self.softmax_scale,
            )
        else:  # eager (attention/base.py:205-269)
            dtype = q.dtype
            sm_dtype = torch.float32 if self.attention_softmax_in_fp32 else dtype
            scores = torch.matmul(q, k.transpose(-1, -2)) * self.softmax_scale
            if attention_bias is not None:
                scores = scores + attention_bias
            elif self.causal:
                mask = torch.ones(S, S, dtype=torch.bool, device=q.device).tril()
                scores = scores.masked_fill(~mask, torch.finfo(scores.dtype).min)
            p = F.softmax(scores.to(sm_dtype), dim=-1).to(dtype)
            p = self.attn_dropout(p)
            out = torch.matmul(p, v)

        out = out.transpose(1, 2).reshape(B, S, H * D)
        out = self.c_proj(out)
        return self.resid_dropout(out)


# ---------------------------------------------------------------------------
# MLP (gpt_dolomite/mlp.py)
# ---------------------------------------------------------------------------


_GLU_BASE = {"swiglu": "silu", "geglu": "gelu_tanh", "reglu": "relu"}


def _activation(name: str):
    base = name
    glu = name.endswith("glu")
    if glu:
        base = _GLU_BASE.get(name, name[: -len("_glu")] if name.endswith("_glu") else None)
        if base is None:
            raise NotImplementedError(f"activation {name}")
    if base in ("gelu_pytorch_tanh", "gelu_tanh"):
        fn = lambda x: F.gelu(x, approximate="tanh")
    elif base == "gelu":
        fn = F.gelu
    elif base in ("silu", "swish"):
        fn = F.silu
    elif base == "relu":
        fn = F.relu
    else:
        raise NotImplementedError(f"activation {name}")
    return fn, glu


class MLP(nn.Module):
    def __init__(self, config: GPTDolomiteConfig):
        super().__init__()
        hidden_size = config.n_embd
        intermediate = config.n_inner
        self.act, self.is_glu = _activation(config.activation_function)

        init_method = config.init_method
        std = config.initializer_range
        if init_method == "mup":
            std /= math.sqrt(config.m_width)
        self.c_fc = ParameterizedLinear(
            hidden_size, 2 * intermediate if self.is_glu else intermediate, bias=config.add_bias, std=std
        )
        std = config.initializer_range / math.sqrt(2 * config.n_layer)
        if init_method == "mup":
            std /= math.sqrt(config.m_width)
        self.c_proj = ParameterizedLinear(intermediate, hidden_size, bias=config.add_bias, std=std)
        self.dropout = nn.Identity() if config.resid_pdrop == 0 else nn.Dropout(config.resid_pdrop)

    def forward(self, x):
        x = self.c_fc(x)
        if self.is_glu:
            a, b = x.chunk(2, dim=-1)  # glu.py:26-28: x0 * act(x1)
            x = a * self.act(b)
        else:
            x = self.act(x)
        return self.dropout(self.c_proj(x))


# ---------------------------------------------------------------------------
# Block (gpt_dolomite/layer.py) — padding-free variant fuses the residual
# adds into the norms: forward(delta, residual) -> (delta', residual').
# ---------------------------------------------------------------------------


class GPTDolomiteBlock(nn.Module):
    def __init__(self, config: GPTDolomiteConfig, use_padding_free_transformer: bool, layer_idx: int | None = None):
        super().__init__()
        self.m_residual = config.m_residual
        self.ln_1 = DolomiteNorm(config.normalization_function, config.n_embd, config.layer_norm_epsilon)
        self.attn = Attention(config, causal=True, layer_idx=layer_idx)
        self.ln_2 = DolomiteNorm(config.normalization_function, config.n_embd, config.layer_norm_epsilon)
        self.mlp = MLP(config)

    def forward_padding_free(self, delta, residual, rope_cos_sin, cu_seqlens, max_seqlen):
        y, s = self.ln_1(delta, residual)
        attn_out = self.attn.forward_padding_free(y, rope_cos_sin, cu_seqlens, max_seqlen)
        if self.m_residual is not None:
            attn_out = attn_out * self.m_residual
        y, s = self.ln_2(attn_out, s)
        mlp_out = self.mlp(y)
        if self.m_residual is not None:
            mlp_out = mlp_out * self.m_residual
        return mlp_out, s

    def forward_dense(self, hidden_states, attention_bias, rope_cos_sin, implementation):
        y, _ = self.ln_1(hidden_states)
        attn_out = self.attn.forward_dense(y, attention_bias, rope_cos_sin, implementation)
        if self.m_residual is not None:
            attn_out = attn_out * self.m_residual
        hidden_states = attn_out + hidden_states
        y, _ = self.ln_2(hidden_states)
        mlp_out = self.mlp(y)
        if self.m_residual is not None:
            mlp_out = mlp_out * self.m_residual
        return hidden_states + mlp_out


# ---------------------------------------------------------------------------
# Model
# ---------------------------------------------------------------------------


class GPTDolomitePreTrainedModel(PreTrainedModel):
    config_class = GPTDolomiteConfig
    base_model_prefix = "transformer"
    causal = True
    _no_split_modules = ["GPTDolomiteBlock"]
    _supports_sdpa = True
    _supports_flash_attn_2 = True

    def __init__(self, config: GPTDolomiteConfig, *args, **kwargs):
        # mirror reference GPTDolomitePreTrainedModel.__init__ (base.py:30-63)
        self._use_padding_free_transformer = kwargs.pop("use_padding_free_transformer", False)
        kwargs.pop("normalization_implementation", None)  # single native impl

        impl = getattr(config, "_attn_implementation", None) or "sdpa"
        if impl == "flash_attention_2":
            # our own HIP varlen attention, not the flash-attn package:
            # keep transformers-5's availability validator out of the way
            # during super().__init__, then restore the requested value
            config._attn_implementation = "sdpa"
        super().__init__(config, *args, **kwargs)
        if impl == "flash_attention_2":
            config._attn_implementation = impl

        self.attention_implementation = impl
        self._tied_word_embeddings = config.tie_word_embeddings
        if self._use_padding_free_transformer:
            assert impl == "flash_attention_2", "padding free transformer only works with flash attention"
        assert impl in ("eager", "sdpa", "flash_attention_2"), impl

    def _init_weights(self, module):
        if isinstance(module, (nn.Embedding, nn.Linear, DolomiteNorm)):
            module.reset_parameters()

    def prepare_inputs_for_model(
        self, input_ids, inputs_embeds, position_ids, token_type_ids, labels, cu_seqlens, max_seqlen,
        past_key_values, attention_mask, use_cache, output_attentions,
    ):
        """Reference base.py:68-115 + hf_models/utils.py:20-58."""
        if self._use_padding_free_transformer:
            if isinstance(input_ids, list) or isinstance(inputs_embeds, list):
                msg = "{v} should not be passed for list inputs"
                assert cu_seqlens is None, msg.format(v="cu_seqlens")
                assert max_seqlen is None, msg.format(v="max_seqlen")
                assert attention_mask is None, msg.format(v="attention_mask")
                device = next(self.parameters()).device
                seqlens = torch.tensor([0] + [len(x) for x in input_ids])
                cu_seqlens = seqlens.cumsum(dim=-1).to(device, torch.int32)
                max_seqlen = int(seqlens.max())
                if position_ids is None:
                    position_ids = [list(range(len(x))) for x in input_ids]
                position_ids = torch.tensor([p for seq in position_ids for p in seq], device=device)
                input_ids = torch.tensor([t for seq in input_ids for t in seq], device=device)
                if token_type_ids is not None:
                    token_type_ids = torch.tensor([t for seq in token_type_ids for t in seq], device=device)
                if labels is not None:
                    labels = torch.tensor([t for seq in labels for t in seq], device=device)
            else:
                assert cu_seqlens is not None, "cu_seqlens needs to be specified for tensor inputs"
                assert position_ids is not None, "position_ids needs to be specified when specifying cu_seqlens"
                assert max_seqlen is not None, "max_seqlen needs to be specified when specifying cu_seqlens"
                assert attention_mask is None, "attention_mask should not be passed when specifying cu_seqlens"
            if use_cache or past_key_values is not None:
                raise NotImplementedError("KV caching is not supported with padding_free transformer")
        assert not output_attentions
        return input_ids, position_ids, token_type_ids, labels, cu_seqlens, max_seqlen


class GPTDolomiteModel(GPTDolomitePreTrainedModel):
    block_class = GPTDolomiteBlock  # swapped by model families (MoE)

    def __init__(self, config: GPTDolomiteConfig, **kwargs):
        super().__init__(config, **kwargs)
        self.embed_dim = config.n_embd
        self.m_emb = config.m_emb
        self.head_dim = config.head_dim
        self.wte = ParameterizedEmbedding(config.vocab_size, self.embed_dim, std=config.initializer_range)
        self.drop = nn.Identity() if config.embd_pdrop == 0 else nn.Dropout(config.embd_pdrop)
        self.h = nn.ModuleList(
            [
                self.block_class(config, self._use_padding_free_transformer, layer_idx=i)
                for i in range(config.n_layer)
            ]
        )
        self.ln_f = DolomiteNorm(config.normalization_function, self.embed_dim, config.layer_norm_epsilon)

        self.position_embedding_type = config.position_embedding_type
        if self.position_embedding_type == "learned_absolute":
            self.wpe = ParameterizedEmbedding(config.n_positions, self.embed_dim, std=config.initializer_range)
        elif self.position_embedding_type == "rope":
            # transformers-5 normalizes rope_scaling=None into a default dict
            rs = config.rope_scaling
            if rs is None or (isinstance(rs, dict) and rs.get("rope_type", rs.get("type", "default")) == "default"):
                self.rope = RoPE(self.head_dim, config.n_positions, config.rope_theta)
            else:
                # reference base.py:534-547: YaRN-scaled tables
                self.rope = YaRNScaledRoPE(
                    self.head_dim, config.n_positions, config.rope_theta,
                    scale=rs["factor"],
                    original_max_position_embeddings=rs["original_max_position_embeddings"],
                )
        elif self.position_embedding_type == "alibi":
            # reference base.py:261-287 + alibi.py; supported on the dense
            # eager/sdpa paths (the reference's flash path drops it too)
            assert not self._use_padding_free_transformer, (
                "alibi is not supported on the padding-free/flash path (matches the reference)"
            )
            self.alibi = Alibi(config.n_head)
        else:
            raise NotImplementedError(f"position_embedding_type {self.position_embedding_type}")

        self.post_init()

    def get_input_embeddings(self):
        return self.wte

    def set_input_embeddings(self, value):
        self.wte = value

    def forward(
        self,
        input_ids=None,
        past_key_values=None,
        attention_mask=None,
        token_type_ids=None,
        position_ids=None,
        inputs_embeds=None,
        use_cache=None,
        output_hidden_states=None,
        return_dict=None,
        cu_seqlens=None,
        max_seqlen=None,
    ):
        assert past_key_values is None and not use_cache, "KV cache is not implemented (training engine)"
        assert inputs_embeds is None, "inputs_embeds input is not implemented"
        assert not output_hidden_states, "output_hidden_states is not implemented"

        if self._use_padding_free_transformer:
            hidden_states = self._forward_padding_free(input_ids, position_ids, token_type_ids, cu_seqlens, max_seqlen)
        else:
            hidden_states = self._forward_dense(input_ids, attention_mask, position_ids, token_type_ids)

        return BaseModelOutputWithPast(last_hidden_state=hidden_states)

    # ---- the hot path ----
    def _forward_padding_free(self, input_ids, position_ids, token_type_ids, cu_seqlens, max_seqlen):
        hs = self.wte(input_ids)
        if token_type_ids is not None:
            hs = hs + self.wte(token_type_ids)
        if self.position_embedding_type == "learned_absolute":
            hs = hs + self.wpe(position_ids)
        hs = self.drop(hs)
        if self.m_emb is not None:
            hs = hs * self.m_emb

        rope_cos_sin = None
        if self.position_embedding_type == "rope":
            cos, sin = self.rope(int(max_seqlen), hs.device)  # (S, D) fp32
            # match the reference's table rounding (cast to model dtype,
            # base.py:289-296) while the kernel multiplies in fp32
            cos = cos.to(hs.dtype).float()[position_ids].contiguous()
            sin = sin.to(hs.dtype).float()[position_ids].contiguous()
            rope_cos_sin = (cos, sin)

        delta, residual = hs, None
        for block in self.h:
            if getattr(block, "_gradient_checkpointing", False) and self.training:
                delta, residual = torch.utils.checkpoint.checkpoint(
                    block.forward_padding_free, delta, residual, rope_cos_sin, cu_seqlens, max_seqlen,
                    use_reentrant=False,
                )
            else:
                delta, residual = block.forward_padding_free(delta, residual, rope_cos_sin, cu_seqlens, max_seqlen)
        hidden_states, _ = self.ln_f(delta, residual)
        return hidden_states

    # ---- dense eager/sdpa path ----
    def _forward_dense(self, input_ids, attention_mask, position_ids, token_type_ids):
        B, S = input_ids.shape
        device = input_ids.device
        if position_ids is None:
            position_ids = torch.arange(S, dtype=torch.long, device=device).unsqueeze(0).expand(B, -1)

        hs = self.wte(input_ids)
        if token_type_ids is not None:
            hs = hs + self.wte(token_type_ids)
        if self.position_embedding_type == "learned_absolute":
            hs = hs + self.wpe(position_ids)
        hs = self.drop(hs)
        if self.m_emb is not None:
            hs = hs * self.m_emb

        rope_cos_sin = None
        if self.position_embedding_type == "rope":
            cos, sin = self.rope(S, device)
            cos = cos.to(hs.dtype)[position_ids].unsqueeze(1)  # (B, 1, S, D)
            sin = sin.to(hs.dtype)[position_ids].unsqueeze(1)
            rope_cos_sin = (cos, sin)

        attention_bias = None
        if attention_mask is not None:
            # (B, S) padding mask & causal -> additive float bias
            # (reference base.py:299-350, 553-580)
            causal = torch.ones(S, S, dtype=torch.bool, device=device).tril().unsqueeze(0)
            allowed = causal & attention_mask.unsqueeze(1).to(torch.bool)
            bias = torch.zeros(B, S, S, dtype=hs.dtype, device=device)
            bias = bias.masked_fill(~allowed, torch.finfo(hs.dtype).min)
            attention_bias = bias.unsqueeze(1)
        if self.position_embedding_type == "alibi":
            # (B, H, 1, K) additive slopes*position bias on top of the mask
            # (reference base.py:484-503, _get_maybe_causal_mask:569-598).
            # Faithful quirk: the reference's SDPA branch folds alibi into
            # the mask ONLY when a padding mask exists — sdpa + alibi with
            # attention_mask=None silently runs without alibi. We mirror
            # that for drop-in parity; the eager branch always applies it.
            if self.attention_implementation == "sdpa" and attention_mask is None:
                pass
            else:
                ab = self.alibi(attention_mask, B, S, device, hs.dtype).unsqueeze(2)
                if attention_bias is None:
                    causal = torch.ones(S, S, dtype=torch.bool, device=device).tril()
                    cb = torch.zeros(S, S, dtype=hs.dtype, device=device)
                    attention_bias = cb.masked_fill(~causal, torch.finfo(hs.dtype).min).unsqueeze(0).unsqueeze(0)
                attention_bias = attention_bias + ab

        for block in self.h:
            if getattr(block, "_gradient_checkpointing", False) and self.training:
                hs = torch.utils.checkpoint.checkpoint(
                    block.forward_dense, hs, attention_bias, rope_cos_sin, self.attention_implementation,
                    use_reentrant=False,
                )
            else:
                hs = block.forward_dense(hs, attention_bias, rope_cos_sin, self.attention_implementation)
        hs, _ = self.ln_f(hs)
        return hs


class GPTDolomiteForCausalLM(GPTDolomitePreTrainedModel):
    _tied_weights_keys = {"lm_head.weight": "transformer.wte.weight"}
    model_class = GPTDolomiteModel  # swapped by model families (MoE)

    def __init__(self, config: GPTDolomiteConfig, **kwargs):
        super().__init__(config, **kwargs)
        self.transformer = self.model_class(config, use_padding_free_transformer=self._use_padding_free_transformer)
        if not self._tied_word_embeddings:
            self.lm_head = ParameterizedLinear(
                config.n_embd, config.vocab_size, bias=False, std=config.initializer_range
            )
        self.m_width = config.m_width
        self.upcast_logits_for_loss = config.upcast_logits_for_loss
        self.post_init()

    def get_input_embeddings(self):
        return self.transformer.wte

    def set_input_embeddings(self, value):
        self.transformer.wte = value

    def tie_weights(self, *args, **kwargs):
        # when tied, logits are computed as F.linear(hs, wte.weight) directly
        # (reference main.py:172-177) — there is no lm_head module to tie
        if not self._tied_word_embeddings:
            super().tie_weights(*args, **kwargs)

    def get_output_embeddings(self):
        if not self._tied_word_embeddings:
            return self.lm_head

    def set_output_embeddings(self, new_embeddings):
        if not self._tied_word_embeddings:
            self.lm_head = new_embeddings

    def forward(
        self,
        input_ids=None,
        past_key_values=None,
        attention_mask=None,
        token_type_ids=None,
        position_ids=None,
        inputs_embeds=None,
        labels=None,
        use_cache=None,
        output_attentions=None,
        output_hidden_states=None,
        return_dict=None,
        cu_seqlens=None,
        max_seqlen=None,
    ):
        """Drop-in signature of the reference (gpt_dolomite/main.py:95-170)."""
        input_ids, position_ids, token_type_ids, labels, cu_seqlens, max_seqlen = self.prepare_inputs_for_model(
            input_ids=input_ids,
            inputs_embeds=inputs_embeds,
            position_ids=position_ids,
            token_type_ids=token_type_ids,
            labels=labels,
            cu_seqlens=cu_seqlens,
            max_seqlen=max_seqlen,
            past_key_values=past_key_values,
            attention_mask=attention_mask,
            use_cache=use_cache,
            output_attentions=output_attentions,
        )

        transformer_outputs = self.transformer(
            input_ids,
            past_key_values=past_key_values,
            attention_mask=attention_mask,
            token_type_ids=token_type_ids,
            position_ids=position_ids,
            use_cache=use_cache,
            output_hidden_states=output_hidden_states,
            return_dict=return_dict,
            cu_seqlens=cu_seqlens,
            max_seqlen=max_seqlen,
        )
        hidden_states = transformer_outputs.last_hidden_state

        lm_logits = self.get_lm_logits(hidden_states)
        if self.m_width is not None:
            lm_logits = lm_logits / self.m_width

        loss = self.get_autoregressive_language_modeling_loss(lm_logits, labels, cu_seqlens)

        return CausalLMOutputWithPast(loss=loss, logits=lm_logits)

    def get_lm_logits(self, hidden_states):
        return (
            F.linear(hidden_states, self.transformer.wte.weight)
            if self._tied_word_embeddings
            else self.lm_head(hidden_states)
        )

    def get_autoregressive_language_modeling_loss(self, lm_logits, labels, cu_seqlens):
        """Reference gpt_dolomite/main.py:179-202 (shift + boundary drops)."""
        if labels is None:
            return None
        if self._use_padding_free_transformer:
            shift_logits = lm_logits[:-1, :]
            shift_labels = labels[1:].clone()
            drop_loss_positions = cu_seqlens[1:-1].long() - 1
            shift_labels[drop_loss_positions] = -100
        else:
            shift_logits = lm_logits[..., :-1, :].contiguous()
            shift_labels = labels[..., 1:].contiguous()
        if self.upcast_logits_for_loss:
            shift_logits = shift_logits.float()
        return fused_cross_entropy(shift_logits.reshape(-1, shift_logits.size(-1)), shift_labels.reshape(-1))


def apply_gradient_checkpointing(model, gradient_checkpointing_method: str = "block", checkpoint_every: int = 1, **_):
    """Per-block activation checkpointing (reference
    gradient_checkpointing/block.py:13-34: every `checkpoint_every`-th
    GPTDolomiteBlock recomputes its forward in backward)."""
    assert gradient_checkpointing_method == "block", gradient_checkpointing_method
    # Resolve the block class from the model itself (reference block.py
    # looks the block class up by name) so MoEDolomiteBlock and any future
    # family is wrapped too, not just GPTDolomiteBlock.
    block_class = None
    for module in model.modules():
        if getattr(module, "block_class", None) is not None:
            block_class = module.block_class
            break
    if block_class is None:
        block_class = GPTDolomiteBlock
    idx = 0
    marked = 0
    for module in model.modules():
        if isinstance(module, block_class):
            if idx % checkpoint_every == 0:
                module._gradient_checkpointing = True
                marked += 1
            idx += 1
    assert marked > 0, f"no {block_class.__name__} blocks found to checkpoint"
