"""Llama <-> gpt_dolomite conversion round trip (reference
model_conversion/llama.py; parity harness mirrors the reference's
single_gpu/model_conversion_test.py: exact weight mapping + logits match)."""

import json

import pytest
import torch

from dolomite_engine_amd.hf_models import GPTDolomiteConfig, GPTDolomiteForCausalLM
from dolomite_engine_amd.hf_models.conversion import (
    export_to_huggingface_llama,
    import_from_huggingface,
)


def _make_tiny_llama(tmp_path, kv_heads):
    from transformers import LlamaConfig, LlamaForCausalLM

    torch.manual_seed(11)
    cfg = LlamaConfig(
        vocab_size=256,
        hidden_size=64,
        intermediate_size=128,
        num_hidden_layers=2,
        num_attention_heads=4,
        num_key_value_heads=kv_heads,
        max_position_embeddings=128,
        rms_norm_eps=1e-6,
        attention_bias=False,
        tie_word_embeddings=False,
        bos_token_id=0,
        eos_token_id=1,
    )
    model = LlamaForCausalLM(cfg)
    model.eval()
    src = tmp_path / "llama"
    model.save_pretrained(str(src))
    return model, src


@pytest.mark.parametrize("kv_heads", [4, 2, 1])  # mha, gqa, mqa
def test_llama_import_logits_match(tmp_path, kv_heads):
    llama, src = _make_tiny_llama(tmp_path, kv_heads)
    dst = tmp_path / "dolomite"
    import_from_huggingface(str(src), str(dst))

    cfg = GPTDolomiteConfig.from_pretrained(str(dst))
    cfg._attn_implementation = "eager"
    model = GPTDolomiteForCausalLM.from_pretrained(str(dst), config=cfg)
    model.eval()

    ids = torch.randint(0, 256, (2, 10), generator=torch.Generator().manual_seed(3))
    with torch.no_grad():
        ours = model(input_ids=ids).logits
        theirs = llama(input_ids=ids).logits
    torch.testing.assert_close(ours, theirs, rtol=1e-4, atol=2e-5)


def test_llama_round_trip_weights_exact(tmp_path):
    from safetensors.torch import load_file

    _, src = _make_tiny_llama(tmp_path, 2)
    mid = tmp_path / "dolomite"
    back = tmp_path / "llama2"
    import_from_huggingface(str(src), str(mid))
    export_to_huggingface_llama(str(mid), str(back))

    orig = {}
    for f in src.glob("*.safetensors"):
        orig.update(load_file(str(f)))
    rt = load_file(str(back / "model.safetensors"))
    for k, v in orig.items():
        if k.endswith("rotary_emb.inv_freq"):
            continue
        assert k in rt, k
        torch.testing.assert_close(rt[k], v, rtol=0, atol=0, msg=lambda m: f"{k}: {m}")
