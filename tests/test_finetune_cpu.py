"""Finetuning path end-to-end on CPU: config #1-style run (tiny model, eager,
debug dataset) for both dense and padding-free collates, plus unshard."""

import json

import pytest
import torch
import yaml

from dolomite_engine_amd import finetune

TINY_FT_CONFIG = {
    "model_args": {
        "model_class": "AutoModelForCausalLM",
        "pretrained_config": {
            "model_type": "gpt_dolomite",
            "vocab_size": 512,
            "n_positions": 128,
            "n_embd": 64,
            "n_layer": 2,
            "n_head": 4,
            "n_inner": 128,
            "attention_head_type": "mqa",
            "position_embedding_type": "rope",
            "normalization_function": "rmsnorm",
            "activation_function": "gelu_pytorch_tanh",
            "resid_pdrop": 0.0,
            "embd_pdrop": 0.0,
            "attn_pdrop": 0.0,
            "tie_word_embeddings": False,
            "bos_token_id": 0,
            "eos_token_id": 1,
            "pad_token_id": 2,
        },
        "attention_implementation": "eager",
    },
    "tuning_args": {"tuning_method": "full_finetuning"},
    "training_parameters": {
        "num_training_steps": 3,
        "micro_batch_size": 4,
        "gradient_accumulation_steps": 1,
        "gradient_clipping": 1.0,
        "loss_mask": "output_only",
    },
    "optimizer_args": {"class_args": {"lr": 1e-3, "weight_decay": 0.1, "betas": [0.9, 0.95], "eps": 1e-10}},
    "lr_scheduler_args": {"num_warmup_steps": 1, "lr_decay_style": "cosine"},
    "mixed_precision_args": {"dtype": "fp32"},
    "distributed_args": {"stage": 2, "overlap_comm": False},
    "datasets": [
        {
            "class_name": "DebugDataset",
            "class_args": {"num_examples": 64, "vocab_size": 512},
            "data_name": "debug",
            "data_sampling_ratio": 1,
        }
    ],
    "random_args": {"seed": 5},
}


def _run(tmp_path, overrides):
    cfg = json.loads(json.dumps(TINY_FT_CONFIG))
    for path, v in overrides.items():
        d = cfg
        keys = path.split(".")
        for k in keys[:-1]:
            d = d[k]
        d[keys[-1]] = v
    cfg["save_args"] = {"save_path": str(tmp_path / "ckpt")}
    p = tmp_path / "cfg.yml"
    with open(p, "w") as f:
        yaml.safe_dump(cfg, f)
    finetune.main(["--config", str(p)])
    return tmp_path / "ckpt"


def test_finetune_dense_eager(tmp_path):
    ckpt = _run(tmp_path, {})
    assert (ckpt / "global_step3" / "model").exists()


def test_finetune_padding_free_cpu(tmp_path):
    ckpt = _run(
        tmp_path,
        {
            "model_args.attention_implementation": "flash_attention_2",
            "model_args.use_padding_free_transformer": True,
        },
    )
    assert (ckpt / "global_step3" / "model").exists()


def test_unshard(tmp_path):
    ckpt = _run(tmp_path, {})
    from dolomite_engine_amd.unshard import unshard_checkpoint

    out = tmp_path / "consolidated"
    unshard_checkpoint(str(ckpt), None, str(out))
    assert list(out.glob("*.safetensors"))
    from dolomite_engine_amd.hf_models import GPTDolomiteConfig, GPTDolomiteForCausalLM

    cfg = GPTDolomiteConfig.from_pretrained(str(out))
    cfg._attn_implementation = "eager"
    model = GPTDolomiteForCausalLM.from_pretrained(str(out), config=cfg)
    ids = torch.randint(0, 512, (1, 8))
    out_m = model(input_ids=ids, labels=ids)
    assert torch.isfinite(out_m.loss)


def test_jsonl_tokenized_dataset(tmp_path):
    rows = [{"input_ids": [3, 4, 5], "output_ids": [6, 7]}, {"input_ids": [8], "output_ids": [9, 10, 11]}]
    p = tmp_path / "data.jsonl"
    with open(p, "w") as f:
        for r in rows:
            f.write(json.dumps(r) + "\n")
    ckpt = _run(
        tmp_path,
        {"datasets": [{"class_name": "JSONLinesTokenizedDataset", "class_args": {"data_path": str(p)}, "data_name": "j"}]},
    )
    assert (ckpt / "global_step3" / "model").exists()
