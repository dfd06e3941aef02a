"""Drop-in model surface (reference: dolomite_engine/hf_models)."""

from .config import GPTDolomiteConfig
from .modeling import (
    apply_gradient_checkpointing,
    GPTDolomiteBlock,
    GPTDolomiteForCausalLM,
    GPTDolomiteModel,
    GPTDolomitePreTrainedModel,
    ParameterizedEmbedding,
    ParameterizedLinear,
)


def register_model_classes() -> None:
    """Register into HF Auto classes (reference register_hf.py:35-45)."""
    from transformers import AutoConfig, AutoModel, AutoModelForCausalLM

    AutoConfig.register("gpt_dolomite", GPTDolomiteConfig, exist_ok=True)
    AutoModel.register(GPTDolomiteConfig, GPTDolomiteModel, exist_ok=True)
    AutoModelForCausalLM.register(GPTDolomiteConfig, GPTDolomiteForCausalLM, exist_ok=True)
