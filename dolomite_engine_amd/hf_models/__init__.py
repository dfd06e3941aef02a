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
from .moe import (
    MoEDolomiteBlock,
    MoEDolomiteConfig,
    MoEDolomiteForCausalLM,
    MoEDolomiteModel,
    ParameterizedExperts,
    SparseMoE,
)


def register_model_classes() -> None:
    """Register into HF Auto classes (reference register_hf.py:35-45)."""
    from transformers import AutoConfig, AutoModel, AutoModelForCausalLM

    AutoConfig.register("gpt_dolomite", GPTDolomiteConfig, exist_ok=True)
    AutoModel.register(GPTDolomiteConfig, GPTDolomiteModel, exist_ok=True)
    AutoModelForCausalLM.register(GPTDolomiteConfig, GPTDolomiteForCausalLM, exist_ok=True)
    AutoConfig.register("moe_dolomite", MoEDolomiteConfig, exist_ok=True)
    AutoModel.register(MoEDolomiteConfig, MoEDolomiteModel, exist_ok=True)
    AutoModelForCausalLM.register(MoEDolomiteConfig, MoEDolomiteForCausalLM, exist_ok=True)
