"""Shim that makes the upstream reference (ibm-granite/dolomite-engine, mounted
read-only at /root/reference) importable in THIS container only, so that golden
vectors can be generated from the reference's own CPU eager path.

TEST INFRASTRUCTURE ONLY. The product package (dolomite_engine_amd) must never
import this module; /root/reference does not exist on the GPU box. Golden
vectors generated through this shim are committed under tests/golden/ together
with the generating script (oracle/gen_golden.py).

Why a shim is needed:
  1. torch 2.10 renamed torch.distributed._tensor.placement_types._Partial ->
     Partial; the reference imports _Partial (modeling_utils_TP/linear.py:8).
  2. transformers 5.x PretrainedConfig drops custom attributes set in
     __init__ of subclasses; re-inject them after construction
     (reference fields: hf_models/config.py:50-78).
  3. tie_word_embeddings must be False (transformers-5 expects a dict
     _tied_weights_keys; the reference has a list, main.py:12).
"""

import sys

REFERENCE_PATH = "/root/reference"


def import_reference_hf_models():
    if REFERENCE_PATH not in sys.path:
        sys.path.insert(0, REFERENCE_PATH)

    import torch.distributed._tensor.placement_types as ptypes

    if not hasattr(ptypes, "_Partial"):
        ptypes._Partial = ptypes.Partial

    import dolomite_engine.hf_models as hf_models

    return hf_models


# Field resolution mirroring the reference CommonConfig.__init__
# (/root/reference/dolomite_engine/hf_models/config.py:15-110).
_CONFIG_DEFAULTS = dict(
    vocab_size=50257,
    n_positions=1024,
    n_embd=768,
    n_layer=12,
    n_head=12,
    num_key_value_heads=None,
    n_inner=None,
    activation_function="gelu_pytorch_tanh",
    attention_head_type="mqa",
    resid_pdrop=0.1,
    embd_pdrop=0.1,
    attn_pdrop=0.1,
    normalization_function="layernorm",
    layer_norm_epsilon=1e-5,
    initializer_range=0.02,
    scale_attn_weights=True,
    attention_multiplier=None,
    use_cache=True,
    attention_softmax_in_fp32=True,
    add_bias=True,
    position_embedding_type="learned_absolute",
    rope_theta=10000,
    rope_scaling=None,
    m_emb=None,
    m_width=None,
    m_residual=None,
    init_method="normal",
    upcast_logits_for_loss=False,
)


def make_reference_config(**kwargs):
    """Build a reference GPTDolomiteConfig under transformers 5.x.

    Re-injects every custom field with the same resolution logic the
    reference constructor applies (config.py:50-110): n_inner defaulting to
    4*n_embd, num_key_value_heads resolution per attention_head_type, and
    the multi_query flag.
    """
    hf_models = import_reference_hf_models()

    kwargs.setdefault("tie_word_embeddings", False)
    config = hf_models.GPTDolomiteConfig(**kwargs)

    fields = dict(_CONFIG_DEFAULTS)
    for k in fields:
        if k in kwargs:
            fields[k] = kwargs[k]

    if fields["n_inner"] is None:
        fields["n_inner"] = 4 * fields["n_embd"]
    aht = fields["attention_head_type"]
    if aht == "mha" and fields["num_key_value_heads"] is None:
        fields["num_key_value_heads"] = fields["n_head"]
    elif aht == "mqa" and fields["num_key_value_heads"] is None:
        fields["num_key_value_heads"] = 1
    fields["multi_query"] = aht == "mqa"

    for k, v in fields.items():
        object.__setattr__(config, k, v)

    return config


def make_reference_model(config, attn_implementation="eager", **kwargs):
    hf_models = import_reference_hf_models()
    config._attn_implementation = attn_implementation
    model = hf_models.GPTDolomiteForCausalLM(config, **kwargs)
    return model


# MoE family (reference moe_dolomite/config.py adds four fields on top of
# CommonConfig; same transformers-5 attribute re-injection applies).
_MOE_EXTRA_DEFAULTS = dict(
    num_experts=8,
    num_experts_per_tok=2,
    output_router_logits=False,
    router_aux_loss_coef=0.001,
)


def make_reference_moe_config(**kwargs):
    hf_models = import_reference_hf_models()

    kwargs.setdefault("tie_word_embeddings", False)
    config = hf_models.MoEDolomiteConfig(**kwargs)

    fields = dict(_CONFIG_DEFAULTS)
    fields.update(_MOE_EXTRA_DEFAULTS)
    for k in fields:
        if k in kwargs:
            fields[k] = kwargs[k]

    if fields["n_inner"] is None:
        fields["n_inner"] = 4 * fields["n_embd"]
    aht = fields["attention_head_type"]
    if aht == "mha" and fields["num_key_value_heads"] is None:
        fields["num_key_value_heads"] = fields["n_head"]
    elif aht == "mqa" and fields["num_key_value_heads"] is None:
        fields["num_key_value_heads"] = 1
    fields["multi_query"] = aht == "mqa"

    for k, v in fields.items():
        object.__setattr__(config, k, v)

    return config


def make_reference_moe_model(config, attn_implementation="eager", **kwargs):
    hf_models = import_reference_hf_models()
    config._attn_implementation = attn_implementation
    return hf_models.MoEDolomiteForCausalLM(config, moe_implementation="eager", **kwargs)
