"""Model wrappers — reference model_wrapper/{base,pretraining,finetuning}.py
reimplemented for the MI355X engine (single-node data parallel, no TP)."""

import torch
import torch.nn as nn

from .hf_models import (
    GPTDolomiteConfig,
    GPTDolomiteForCausalLM,
    MoEDolomiteConfig,
    MoEDolomiteForCausalLM,
)
from .ops import fused_cross_entropy
from .utils import string_to_torch_dtype

# model family registry (reference resolves via AutoConfig/model_type)
_MODEL_CLASSES = {
    "gpt_dolomite": (GPTDolomiteConfig, GPTDolomiteForCausalLM),
    "moe_dolomite": (MoEDolomiteConfig, MoEDolomiteForCausalLM),
}


class ModelWrapper(nn.Module):
    """Builds the model from model_name (a saved directory) or
    pretrained_config (reference model_wrapper/base.py:151-244)."""

    def __init__(
        self,
        model_name: str | None,
        pretrained_config: dict | None,
        dtype: torch.dtype | str,
        attention_implementation: str,
        use_padding_free_transformer: bool,
        efficient_initialization: bool = False,
        device: torch.device | str | None = None,
    ):
        super().__init__()
        if isinstance(dtype, str):
            dtype = string_to_torch_dtype(dtype)
        self.dtype = dtype
        self._init_device = torch.device(device) if device is not None else None
        self.attention_implementation = attention_implementation
        self.use_padding_free_transformer = use_padding_free_transformer

        if use_padding_free_transformer:
            assert attention_implementation == "flash_attention_2", (
                "padding free transformer only works with flash attention"
            )

        if model_name is None:
            cfg_kwargs = dict(pretrained_config)
            model_type = cfg_kwargs.pop("model_type", "gpt_dolomite")
            if model_type not in _MODEL_CLASSES:
                raise NotImplementedError(
                    f"model family '{model_type}' is out of scope (DESIGN.md §6: "
                    f"rnn/crosslayer/dense-moe families); built: {sorted(_MODEL_CLASSES)}"
                )
            config_class, model_class = _MODEL_CLASSES[model_type]
            config = config_class(**cfg_kwargs)
            config._attn_implementation = attention_implementation
            self.config = config
            import contextlib

            dev_ctx = torch.device(self._init_device) if self._init_device is not None else contextlib.nullcontext()
            with dev_ctx:
                self.model = model_class(config, use_padding_free_transformer=use_padding_free_transformer)
        else:
            from transformers import AutoConfig

            model_type = AutoConfig.from_pretrained(model_name).model_type
            if model_type not in _MODEL_CLASSES:
                raise NotImplementedError(
                    f"model family '{model_type}' is out of scope (DESIGN.md §6); "
                    f"built: {sorted(_MODEL_CLASSES)}"
                )
            config_class, model_class = _MODEL_CLASSES[model_type]
            config = config_class.from_pretrained(model_name)
            config._attn_implementation = attention_implementation
            self.config = config
            self.model = model_class.from_pretrained(
                model_name, config=config, use_padding_free_transformer=use_padding_free_transformer
            )
        self.model = self.model.to(dtype)

        self.upcast_logits_for_loss = getattr(config, "upcast_logits_for_loss", False)

    def save_pretrained(self, path: str) -> None:
        self.model.save_pretrained(path, safe_serialization=True)


class ModelWrapperForPretraining(ModelWrapper):
    """Loss-external pretraining forward (model_wrapper/pretraining.py:89-229):
    trims (B, S+1) -> inputs/labels, static cu_seqlens/position_ids buffers,
    optional document-boundary reset via eos scan."""

    def __init__(
        self,
        micro_batch_size: int,
        sequence_length: int,
        reset_attention_mask: bool = False,
        reset_position_ids: bool = False,
        eos_token_id: int | None = None,
        **kwargs,
    ):
        super().__init__(**kwargs)
        self.micro_batch_size = micro_batch_size
        self.sequence_length = sequence_length
        self.reset_attention_mask = reset_attention_mask
        self.reset_position_ids = reset_position_ids
        self.eos_token_id = eos_token_id if eos_token_id is not None else self.config.eos_token_id
        if reset_position_ids:
            assert reset_attention_mask, "reset_attention_mask should be specified with reset_position_ids"
        self._buffers_device = None

    def _setup_static_buffers(self, device) -> None:
        B, S = self.micro_batch_size, self.sequence_length
        self._cu_seqlens = torch.arange(0, B * S + 1, S, dtype=torch.int32, device=device)
        self._max_seqlen = S
        self._position_ids = torch.arange(0, S, device=device).repeat(B)
        self._buffers_device = device

    def forward(self, batch: dict) -> torch.Tensor:
        tokens: torch.Tensor = batch["text"]
        device = next(self.model.parameters()).device
        tokens = tokens.to(device)
        input_ids = tokens[:, :-1]
        labels = tokens[:, 1:]

        if self.use_padding_free_transformer:
            B, S = input_ids.shape
            flat = input_ids.reshape(-1)
            if self.reset_attention_mask:
                cu_seqlens, max_seqlen, position_ids = self._document_boundaries(flat, B, S)
            else:
                if self._buffers_device != device:
                    self._setup_static_buffers(device)
                cu_seqlens, max_seqlen, position_ids = self._cu_seqlens, self._max_seqlen, self._position_ids
            out = self.model(
                input_ids=flat,
                position_ids=position_ids,
                cu_seqlens=cu_seqlens,
                max_seqlen=max_seqlen,
            )
            logits = out.logits
            labels = labels.reshape(-1)
        else:
            out = self.model(input_ids=input_ids)
            logits = out.logits.reshape(-1, out.logits.shape[-1])
            labels = labels.reshape(-1)

        if self.upcast_logits_for_loss:
            logits = logits.float()
        return fused_cross_entropy(logits.view(-1, logits.size(-1)), labels)

    def _document_boundaries(self, flat_input_ids, B, S):
        """reset_attention_mask eos-scan (pretraining.py:136-158)."""
        device = flat_input_ids.device
        ends = flat_input_ids == self.eos_token_id
        ends = ends.clone()
        ends[S - 1 :: S] = True  # row boundaries always end a document window
        cu = ends.nonzero(as_tuple=True)[0] + 1
        cu_seqlens = torch.cat([torch.zeros(1, dtype=torch.long, device=device), cu]).to(torch.int32)
        seqlen = cu_seqlens[1:] - cu_seqlens[:-1]
        max_seqlen = int(seqlen.max())
        if self.reset_position_ids:
            position_ids = torch.cat(
                [torch.arange(0, int(i), 1, dtype=torch.long, device=device) for i in seqlen]
            )
        else:
            if self._buffers_device != device:
                self._setup_static_buffers(device)
            position_ids = self._position_ids
        return cu_seqlens, max_seqlen, position_ids


class ModelWrapperForFinetuning(ModelWrapper):
    """Loss-internal finetuning forward (model_wrapper/finetuning.py:11-103):
    padding-free batches arrive as lists of unpadded token ids."""

    def forward(self, batch: dict) -> torch.Tensor:
        if self.use_padding_free_transformer:
            out = self.model(input_ids=batch["input_ids"], labels=batch["labels"])
        else:
            out = self.model(
                input_ids=batch["input_ids"],
                attention_mask=batch.get("attention_mask"),
                labels=batch["labels"],
            )
        return out.loss
