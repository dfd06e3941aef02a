"""Model conversion — llama <-> gpt_dolomite (reference
hf_models/model_conversion/llama.py + modeling_utils/attention/utils.py:18-107
and gpt_dolomite/mlp.py:53-58). Local paths only (no hub downloads).

The fused c_attn layout per head type:
  mha: per head   [q_i | k_i | v_i]
  gqa: per group  [q_{iG..iG+G-1} | k_i | v_i]
  mqa:            [q_all | k | v]
and fused c_fc for GLU: [up ; gate].
"""

import json
from pathlib import Path

import torch

from .config import GPTDolomiteConfig


def interleave_up_gate_tensor_for_mlp(up_weight, gate_weight):
    return torch.cat([up_weight, gate_weight])


def split_up_gate_tensor_for_mlp(c_fc_weight):
    return c_fc_weight.chunk(2)


def interleave_query_key_value_tensor_for_attention(q, k, v, num_heads, num_key_value_heads, head_dim, head_type):
    if head_type == "mha":
        parts = []
        for i in range(num_heads):
            s = i * head_dim
            parts += [q[s : s + head_dim], k[s : s + head_dim], v[s : s + head_dim]]
        return torch.cat(parts)
    if head_type == "gqa":
        G = num_heads // num_key_value_heads
        parts = []
        for i in range(num_key_value_heads):
            parts.append(q[i * G * head_dim : (i + 1) * G * head_dim])
            parts.append(k[i * head_dim : (i + 1) * head_dim])
            parts.append(v[i * head_dim : (i + 1) * head_dim])
        return torch.cat(parts)
    if head_type == "mqa":
        return torch.cat([q, k, v])
    raise ValueError(head_type)


def split_query_key_value_tensor_for_attention(w, num_heads, num_key_value_heads, head_dim, head_type):
    shape_rest = w.shape[1:]
    if head_type == "mha":
        q, k, v = w.view(num_heads, -1).chunk(3, -1)
        return (q.reshape(-1, *shape_rest), k.reshape(-1, *shape_rest), v.reshape(-1, *shape_rest))
    if head_type == "gqa":
        G = num_heads // num_key_value_heads
        w2 = w.view(num_key_value_heads, G + 2, -1)
        q, k, v = w2.split((G, 1, 1), 1)
        return (q.reshape(-1, *shape_rest), k.reshape(-1, *shape_rest), v.reshape(-1, *shape_rest))
    if head_type == "mqa":
        return w.split((num_heads * head_dim, head_dim, head_dim))
    raise ValueError(head_type)


def _load_hf_state_dict(path: Path) -> dict:
    from safetensors.torch import load_file

    sd = {}
    files = sorted(path.glob("*.safetensors"))
    if files:
        for f in files:
            sd.update(load_file(str(f)))
    else:
        sd = torch.load(path / "pytorch_model.bin", weights_only=True)
    return sd


def import_from_huggingface_llama(pretrained_model_name_or_path: str, save_path: str) -> None:
    """Llama checkpoint dir -> gpt_dolomite checkpoint dir
    (reference model_conversion/llama.py:13-35)."""
    from safetensors.torch import save_file

    src = Path(pretrained_model_name_or_path)
    with open(src / "config.json") as f:
        oc = json.load(f)
    assert oc.get("hidden_act", "silu") == "silu"

    n_head = oc["num_attention_heads"]
    n_kv = oc.get("num_key_value_heads", n_head)
    if n_head == n_kv:
        head_type = "mha"
    elif n_kv == 1:
        head_type = "mqa"
    else:
        head_type = "gqa"

    config = GPTDolomiteConfig(
        vocab_size=oc["vocab_size"],
        n_positions=oc.get("max_position_embeddings", 2048),
        n_embd=oc["hidden_size"],
        n_layer=oc["num_hidden_layers"],
        n_head=n_head,
        num_key_value_heads=n_kv,
        attention_head_type=head_type,
        position_embedding_type="rope",
        n_inner=oc["intermediate_size"],
        activation_function="swiglu",
        normalization_function="rmsnorm",
        layer_norm_epsilon=oc.get("rms_norm_eps", 1e-6),
        add_bias=oc.get("attention_bias", False),
        tie_word_embeddings=oc.get("tie_word_embeddings", False),
        initializer_range=oc.get("initializer_range", 0.02),
        rope_theta=oc.get("rope_theta", 10000),
        attn_pdrop=oc.get("attention_dropout", 0.0),
        resid_pdrop=0.0,
        embd_pdrop=0.0,
        bos_token_id=oc.get("bos_token_id"),
        eos_token_id=oc.get("eos_token_id"),
        pad_token_id=oc.get("pad_token_id"),
    )

    hf = _load_hf_state_dict(src)
    head_dim = config.head_dim
    sd = {
        "transformer.wte.weight": hf["model.embed_tokens.weight"],
        "transformer.ln_f.weight": hf["model.norm.weight"],
    }
    if "lm_head.weight" in hf:
        sd["lm_head.weight"] = hf["lm_head.weight"]

    for i in range(config.n_layer):
        p = f"model.layers.{i}"
        o = f"transformer.h.{i}"
        sd[f"{o}.ln_1.weight"] = hf[f"{p}.input_layernorm.weight"]
        sd[f"{o}.ln_2.weight"] = hf[f"{p}.post_attention_layernorm.weight"]
        sd[f"{o}.mlp.c_fc.weight"] = interleave_up_gate_tensor_for_mlp(
            hf[f"{p}.mlp.up_proj.weight"], hf[f"{p}.mlp.gate_proj.weight"]
        )
        sd[f"{o}.mlp.c_proj.weight"] = hf[f"{p}.mlp.down_proj.weight"]
        sd[f"{o}.attn.c_attn.weight"] = interleave_query_key_value_tensor_for_attention(
            hf[f"{p}.self_attn.q_proj.weight"],
            hf[f"{p}.self_attn.k_proj.weight"],
            hf[f"{p}.self_attn.v_proj.weight"],
            n_head,
            n_kv,
            head_dim,
            head_type,
        )
        sd[f"{o}.attn.c_proj.weight"] = hf[f"{p}.self_attn.o_proj.weight"]
        if f"{p}.self_attn.q_proj.bias" in hf:
            sd[f"{o}.attn.c_attn.bias"] = interleave_query_key_value_tensor_for_attention(
                hf[f"{p}.self_attn.q_proj.bias"],
                hf[f"{p}.self_attn.k_proj.bias"],
                hf[f"{p}.self_attn.v_proj.bias"],
                n_head,
                n_kv,
                head_dim,
                head_type,
            )
            sd[f"{o}.attn.c_proj.bias"] = hf[f"{p}.self_attn.o_proj.bias"]

    out = Path(save_path)
    out.mkdir(parents=True, exist_ok=True)
    save_file({k: v.contiguous() for k, v in sd.items()}, str(out / "model.safetensors"), metadata={"format": "pt"})
    config.save_pretrained(save_path)


def export_to_huggingface_llama(pretrained_model_name_or_path: str, save_path: str) -> None:
    """gpt_dolomite checkpoint dir -> llama checkpoint dir
    (reference model_conversion/llama.py:38-205 inverse mapping)."""
    from safetensors.torch import save_file

    src = Path(pretrained_model_name_or_path)
    config = GPTDolomiteConfig.from_pretrained(str(src))
    assert config.normalization_function == "rmsnorm"
    assert config.activation_function == "swiglu"
    assert config.position_embedding_type == "rope"

    sd = _load_hf_state_dict(src)
    n_head, n_kv, head_dim = config.n_head, config.num_key_value_heads, config.head_dim
    head_type = config.attention_head_type

    out_sd = {
        "model.embed_tokens.weight": sd["transformer.wte.weight"],
        "model.norm.weight": sd["transformer.ln_f.weight"],
    }
    if "lm_head.weight" in sd:
        out_sd["lm_head.weight"] = sd["lm_head.weight"]
    for i in range(config.n_layer):
        p = f"model.layers.{i}"
        o = f"transformer.h.{i}"
        out_sd[f"{p}.input_layernorm.weight"] = sd[f"{o}.ln_1.weight"]
        out_sd[f"{p}.post_attention_layernorm.weight"] = sd[f"{o}.ln_2.weight"]
        up, gate = split_up_gate_tensor_for_mlp(sd[f"{o}.mlp.c_fc.weight"])
        out_sd[f"{p}.mlp.up_proj.weight"] = up
        out_sd[f"{p}.mlp.gate_proj.weight"] = gate
        out_sd[f"{p}.mlp.down_proj.weight"] = sd[f"{o}.mlp.c_proj.weight"]
        q, k, v = split_query_key_value_tensor_for_attention(
            sd[f"{o}.attn.c_attn.weight"], n_head, n_kv, head_dim, head_type
        )
        out_sd[f"{p}.self_attn.q_proj.weight"] = q
        out_sd[f"{p}.self_attn.k_proj.weight"] = k
        out_sd[f"{p}.self_attn.v_proj.weight"] = v
        out_sd[f"{p}.self_attn.o_proj.weight"] = sd[f"{o}.attn.c_proj.weight"]

    hf_config = {
        "architectures": ["LlamaForCausalLM"],
        "model_type": "llama",
        "vocab_size": config.vocab_size,
        "hidden_size": config.n_embd,
        "intermediate_size": config.n_inner,
        "num_hidden_layers": config.n_layer,
        "num_attention_heads": config.n_head,
        "num_key_value_heads": config.num_key_value_heads,
        "hidden_act": "silu",
        "max_position_embeddings": config.n_positions,
        "rms_norm_eps": config.layer_norm_epsilon,
        "rope_theta": config.rope_theta,
        "attention_bias": config.add_bias,
        "tie_word_embeddings": config.tie_word_embeddings,
        "bos_token_id": config.bos_token_id,
        "eos_token_id": config.eos_token_id,
    }
    out = Path(save_path)
    out.mkdir(parents=True, exist_ok=True)
    save_file({k: v.contiguous() for k, v in out_sd.items()}, str(out / "model.safetensors"), metadata={"format": "pt"})
    with open(out / "config.json", "w") as f:
        json.dump(hf_config, f, indent=2)


_IMPORTERS = {"llama": import_from_huggingface_llama}
_EXPORTERS = {"llama": export_to_huggingface_llama}


def import_from_huggingface(pretrained_model_name_or_path: str, save_path: str) -> None:
    """Reference model_conversion/__init__.py:19-27 (llama only; other
    families are out of scope this round)."""
    src = Path(pretrained_model_name_or_path)
    with open(src / "config.json") as f:
        model_type = json.load(f)["model_type"]
    if model_type not in _IMPORTERS:
        raise NotImplementedError(f"import for model_type {model_type} (only {list(_IMPORTERS)} this round)")
    _IMPORTERS[model_type](pretrained_model_name_or_path, save_path)


def export_to_huggingface(pretrained_model_name_or_path: str, save_path: str, model_type: str) -> None:
    if model_type not in _EXPORTERS:
        raise NotImplementedError(f"export for model_type {model_type}")
    _EXPORTERS[model_type](pretrained_model_name_or_path, save_path)
