"""MoEDolomite — sparse-MoE model family (reference
hf_models/models/moe_dolomite/). Round-1 ships the reference's own eager
SparseMoE semantics (moe/base.py:53-181: top-k routing, fp32 softmax over the
selected experts' logits, expert-sorted grouped linears, gated index_add
combine); the grouped-GEMM HIP kernel replacing the scattermoe path
(moe/scatter.py) is the committed next row.

Attention/norm/embedding/loss are the same MI355X modules as GPTDolomite —
the padding-free varlen HIP attention path works for MoE too."""

import math

import torch
import torch.nn as nn
import torch.nn.functional as F
from transformers.modeling_outputs import MoeCausalLMOutputWithPast, MoeModelOutputWithPast

from ..ops.functional import MoERowsCombine, MoERowsGather
from .config import GPTDolomiteConfig
from .modeling import (
    Attention,
    DolomiteNorm,
    GPTDolomiteForCausalLM,
    GPTDolomiteModel,
    ParameterizedLinear,
    _activation,
)


class MoEDolomiteConfig(GPTDolomiteConfig):
    """Reference moe_dolomite/config.py:6-110 — CommonConfig + MoE fields."""

    model_type = "moe_dolomite"

    def __init__(
        self,
        num_experts: int = 8,
        num_experts_per_tok: int = 2,
        output_router_logits: bool = False,
        router_aux_loss_coef: float = 0.001,
        **kwargs,
    ) -> None:
        self.num_experts = num_experts
        self.num_experts_per_tok = num_experts_per_tok
        self.output_router_logits = output_router_logits
        self.router_aux_loss_coef = router_aux_loss_coef
        super().__init__(**kwargs)


class ParameterizedExperts(nn.Module):
    """Per-expert weights (E, out, in); grouped forward over expert-sorted
    rows (reference moe/base.py:12-50)."""

    def __init__(self, num_experts, in_features, out_features, add_bias=True, std=None):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(num_experts, out_features, in_features))
        self.bias = nn.Parameter(torch.empty(num_experts, out_features)) if add_bias else None
        self.std = std
        self.num_experts = num_experts
        self.in_features = in_features
        self.out_features = out_features
        self.reset_parameters()

    @torch.no_grad()
    def reset_parameters(self) -> None:
        nn.init.normal_(self.weight, mean=0, std=self.std if self.std is not None else 0.02)
        if self.bias is not None:
            self.bias.zero_()

    def forward(self, input: torch.Tensor, num_tokens_per_expert: torch.Tensor) -> torch.Tensor:
        if input.is_cuda:
            # one grouped-GEMM HIP launch over all experts (csrc/moe_gemm.hip,
            # the scattermoe-path replacement); None only for unsupported
            # dtype/dims, where the eager per-expert loop below still applies
            from ..ops import grouped_expert_gemm

            y = grouped_expert_gemm(input, self.weight, self.bias, num_tokens_per_expert)
            if y is not None:
                return y
        pieces = input.split(num_tokens_per_expert.tolist(), dim=0)
        outs = [
            F.linear(pieces[i], self.weight[i], None if self.bias is None else self.bias[i])
            for i in range(self.num_experts)
        ]
        return torch.cat(outs, dim=0)


class SparseMoE(nn.Module):
    """Eager top-k MoE MLP (reference moe/base.py:53-181)."""

    def __init__(self, config: MoEDolomiteConfig):
        super().__init__()
        self.num_experts = config.num_experts
        self.top_k = config.num_experts_per_tok
        self.hidden_size = config.n_embd
        self.intermediate_size = config.n_inner
        self.act, self.is_glu = _activation(config.activation_function)

        self.gate = ParameterizedLinear(self.hidden_size, config.num_experts, bias=False, std=config.initializer_range)
        std = config.initializer_range
        if config.init_method == "mup":
            std /= math.sqrt(config.m_width)
        self.c_fc = ParameterizedExperts(
            config.num_experts,
            self.hidden_size,
            2 * self.intermediate_size if self.is_glu else self.intermediate_size,
            add_bias=config.add_bias,
            std=std,
        )
        std = config.initializer_range / math.sqrt(2 * config.n_layer)
        if config.init_method == "mup":
            std /= math.sqrt(config.m_width)
        self.c_proj = ParameterizedExperts(
            config.num_experts, self.intermediate_size, self.hidden_size, add_bias=config.add_bias, std=std
        )
        self.dropout = nn.Identity() if config.resid_pdrop == 0 else nn.Dropout(config.resid_pdrop)

    def forward(self, hidden_states: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
        orig_shape = hidden_states.shape
        hidden_states = hidden_states.view(-1, self.hidden_size)
        total_q = hidden_states.shape[0]

        # routing (base.py:96-106): top-k logits -> fp32 softmax over the k
        router_logits = self.gate(hidden_states)
        router_weights, selected_experts = router_logits.topk(self.top_k, dim=-1)
        router_weights = F.softmax(router_weights.float(), dim=-1).type_as(hidden_states)

        # expert assignment (base.py:108-127): sort token-expert pairs by expert
        selected_flat = selected_experts.flatten()
        num_tokens_per_expert = selected_flat.bincount(minlength=self.num_experts)
        _, index_sorted_experts = selected_flat.sort(0)
        batch_index = index_sorted_experts // self.top_k
        batch_gates = router_weights.flatten()[index_sorted_experts]
        # inverse of the expert sort (inv[pair] = slot): lets the gather
        # backward and the scatter-back run through the deterministic
        # combine kernel instead of torch's atomic index_add
        n_pairs = selected_flat.numel()
        inv = torch.empty(n_pairs, dtype=torch.int32, device=hidden_states.device)
        inv[index_sorted_experts] = torch.arange(n_pairs, dtype=torch.int32, device=hidden_states.device)

        expert_inputs = MoERowsGather.apply(hidden_states, batch_index, inv, self.top_k)
        h = self.c_fc(expert_inputs, num_tokens_per_expert)
        if self.is_glu:
            a, b = h.chunk(2, dim=-1)
            h = a * self.act(b)
        else:
            h = self.act(h)
        h = self.c_proj(h, num_tokens_per_expert)
        h = h * batch_gates.unsqueeze(-1)
        out = MoERowsCombine.apply(h, inv, batch_index, total_q, self.top_k)
        out = self.dropout(out.view(orig_shape))
        return out, router_logits


class MoEDolomiteBlock(nn.Module):
    """moe_dolomite/layer.py: ln_1 -> attn -> ln_2 -> SparseMoE; router
    logits returned alongside. Padding-free variant keeps the fused
    residual-in-norm flow of GPTDolomiteBlock."""

    def __init__(self, config: MoEDolomiteConfig, use_padding_free_transformer: bool, layer_idx=None):
        super().__init__()
        self.m_residual = config.m_residual
        self.ln_1 = DolomiteNorm(config.normalization_function, config.n_embd, config.layer_norm_epsilon)
        self.attn = Attention(config, causal=True, layer_idx=layer_idx)
        self.ln_2 = DolomiteNorm(config.normalization_function, config.n_embd, config.layer_norm_epsilon)
        self.mlp = SparseMoE(config)  # named `mlp` in the reference state dict (layer.py:44-49)

    def forward_padding_free(self, delta, residual, rope_cos_sin, cu_seqlens, max_seqlen):
        y, s = self.ln_1(delta, residual)
        attn_out = self.attn.forward_padding_free(y, rope_cos_sin, cu_seqlens, max_seqlen)
        if self.m_residual is not None:
            attn_out = attn_out * self.m_residual
        y, s = self.ln_2(attn_out, s)
        moe_out, router_logits = self.mlp(y)
        if self.m_residual is not None:
            moe_out = moe_out * self.m_residual
        return moe_out, s, router_logits

    def forward_dense(self, hidden_states, attention_bias, rope_cos_sin, implementation):
        y, _ = self.ln_1(hidden_states)
        attn_out = self.attn.forward_dense(y, attention_bias, rope_cos_sin, implementation)
        if self.m_residual is not None:
            attn_out = attn_out * self.m_residual
        hidden_states = attn_out + hidden_states
        y, _ = self.ln_2(hidden_states)
        moe_out, router_logits = self.mlp(y)
        if self.m_residual is not None:
            moe_out = moe_out * self.m_residual
        return hidden_states + moe_out, router_logits


class MoEDolomiteModel(GPTDolomiteModel):
    config_class = MoEDolomiteConfig
    block_class = MoEDolomiteBlock

    def _init_weights(self, module):
        if isinstance(module, ParameterizedExperts):
            module.reset_parameters()
        else:
            super()._init_weights(module)

    def forward(self, input_ids=None, past_key_values=None, attention_mask=None, token_type_ids=None,
                position_ids=None, inputs_embeds=None, use_cache=None, output_hidden_states=None,
                return_dict=None, cu_seqlens=None, max_seqlen=None, output_router_logits=False):
        assert past_key_values is None and not use_cache, "KV cache is not implemented (training engine)"
        assert inputs_embeds is None and not output_hidden_states

        all_router_logits = () if output_router_logits else None

        if self._use_padding_free_transformer:
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
                cos, sin = self.rope(int(max_seqlen), hs.device)
                rope_cos_sin = (
                    cos.to(hs.dtype).float()[position_ids].contiguous(),
                    sin.to(hs.dtype).float()[position_ids].contiguous(),
                )
            delta, residual = hs, None
            for block in self.h:
                if getattr(block, "_gradient_checkpointing", False) and self.training:
                    delta, residual, rl = torch.utils.checkpoint.checkpoint(
                        block.forward_padding_free, delta, residual, rope_cos_sin, cu_seqlens, max_seqlen,
                        use_reentrant=False,
                    )
                else:
                    delta, residual, rl = block.forward_padding_free(
                        delta, residual, rope_cos_sin, cu_seqlens, max_seqlen
                    )
                if output_router_logits:
                    all_router_logits += (rl,)
            hidden_states, _ = self.ln_f(delta, residual)
        else:
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
                rope_cos_sin = (
                    cos.to(hs.dtype)[position_ids].unsqueeze(1),
                    sin.to(hs.dtype)[position_ids].unsqueeze(1),
                )
            attention_bias = None
            if attention_mask is not None:
                causal = torch.ones(S, S, dtype=torch.bool, device=device).tril().unsqueeze(0)
                allowed = causal & attention_mask.unsqueeze(1).to(torch.bool)
                bias = torch.zeros(B, S, S, dtype=hs.dtype, device=device)
                bias = bias.masked_fill(~allowed, torch.finfo(hs.dtype).min)
                attention_bias = bias.unsqueeze(1)
            for block in self.h:
                hs, rl = block.forward_dense(hs, attention_bias, rope_cos_sin, self.attention_implementation)
                if output_router_logits:
                    all_router_logits += (rl,)
            hidden_states, _ = self.ln_f(hs)

        return MoeModelOutputWithPast(last_hidden_state=hidden_states, router_logits=all_router_logits)


class MoEDolomiteForCausalLM(GPTDolomiteForCausalLM):
    config_class = MoEDolomiteConfig
    model_class = MoEDolomiteModel

    def __init__(self, config: MoEDolomiteConfig, **kwargs):
        super().__init__(config, **kwargs)
        self.router_aux_loss_coef = config.router_aux_loss_coef
        self.num_experts = config.num_experts
        self.num_experts_per_tok = config.num_experts_per_tok
        # Deliberate divergence from the reference: moe_dolomite/main.py:47-48
        # raises for padding_free + output_router_logits because the mixtral
        # helper there wants (B, S, E) + attention_mask. The packed layout
        # has NO pad tokens, so the load-balancing loss is computed directly
        # on the concatenated (T, E) router logits — the packed-path MoE
        # bench trains the reference's actual objective.

    def _init_weights(self, module):
        if isinstance(module, ParameterizedExperts):
            module.reset_parameters()
        else:
            super()._init_weights(module)

    def forward(self, input_ids=None, past_key_values=None, attention_mask=None, token_type_ids=None,
                position_ids=None, inputs_embeds=None, labels=None, use_cache=None, output_attentions=None,
                output_hidden_states=None, return_dict=None, cu_seqlens=None, max_seqlen=None,
                output_router_logits=None):
        """Reference moe_dolomite/main.py:50-130: CLM loss + mixtral-style
        load-balancing aux loss scaled by router_aux_loss_coef."""
        if output_router_logits is None:
            output_router_logits = self.config.output_router_logits

        input_ids, position_ids, token_type_ids, labels, cu_seqlens, max_seqlen = self.prepare_inputs_for_model(
            input_ids=input_ids, inputs_embeds=inputs_embeds, position_ids=position_ids,
            token_type_ids=token_type_ids, labels=labels, cu_seqlens=cu_seqlens, max_seqlen=max_seqlen,
            past_key_values=past_key_values, attention_mask=attention_mask, use_cache=use_cache,
            output_attentions=output_attentions,
        )

        transformer_outputs = self.transformer(
            input_ids, attention_mask=attention_mask, token_type_ids=token_type_ids,
            position_ids=position_ids, cu_seqlens=cu_seqlens, max_seqlen=max_seqlen,
            output_router_logits=output_router_logits,
        )
        hidden_states = transformer_outputs.last_hidden_state

        lm_logits = self.get_lm_logits(hidden_states)
        if self.m_width is not None:
            lm_logits = lm_logits / self.m_width

        loss = self.get_autoregressive_language_modeling_loss(lm_logits, labels, cu_seqlens)
        aux_loss = None
        if output_router_logits:
            from transformers.models.mixtral.modeling_mixtral import load_balancing_loss_func

            aux_loss = load_balancing_loss_func(
                transformer_outputs.router_logits, self.num_experts, self.num_experts_per_tok
            )
            if loss is not None:
                loss = loss + self.router_aux_loss_coef * aux_loss

        return MoeCausalLMOutputWithPast(
            loss=loss, aux_loss=aux_loss, logits=lm_logits,
            router_logits=transformer_outputs.router_logits,
        )
