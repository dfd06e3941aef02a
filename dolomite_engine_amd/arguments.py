"""YAML / CLI argument surface — accepts the reference's TrainingArgs schema
verbatim for the hot path (reference arguments.py:30-447). Out-of-scope
options (DeepSpeed backend, tensor parallel, PEFT tuning, fp8) raise with a
clear message instead of being silently ignored.

Entry: get_args(path) or parse_args(argv) for `--config x.yml`.
"""

import argparse
from enum import Enum
from typing import Any

import yaml
from pydantic import BaseModel, ConfigDict, Field


class BaseArgs(BaseModel):
    model_config = ConfigDict(extra="forbid", protected_namespaces=(), arbitrary_types_allowed=True)

    def to_dict(self) -> dict:
        return self.model_dump(mode="json")


class Mode(Enum):
    training = "training"
    inference = "inference"
    unsharding = "unsharding"


class TuningMethod(Enum):
    pretraining = "pretraining"
    full_finetuning = "full_finetuning"
    # out of scope: prompt_tuning, lora


class LossMask(Enum):
    output_only = "output_only"
    no_mask = "no_mask"


class AttentionImplementation(Enum):
    eager = "eager"
    sdpa = "sdpa"
    flash_attention_2 = "flash_attention_2"


class DistributedBackend(Enum):
    torch = "torch"
    # deepspeed is out of scope (north_star: "no DeepSpeed")


class LRDecaySchedule(Enum):
    constant = "constant"
    cosine = "cosine"
    exponential = "exponential"
    linear = "linear"
    power = "power"


class RandomArgs(BaseArgs):
    seed: int = 42


class TokenizerArgs(BaseArgs):
    tokenizer_name: str | None = None
    additional_special_tokens: list[str] | None = None


class ModelArgs(BaseArgs):
    model_name: str | None = None
    pretrained_config: dict | None = None
    model_class: str = "AutoModelForCausalLM"
    trust_remote_code: bool = False
    attention_implementation: AttentionImplementation | None = None
    use_padding_free_transformer: bool = False
    efficient_initialization: bool = False
    reset_attention_mask: bool = False
    reset_position_ids: bool = False

    def model_post_init(self, __context: Any) -> None:
        assert self.model_class in ("AutoModelForCausalLM",), f"unexpected model_class ({self.model_class})"
        assert self.model_name is not None or self.pretrained_config is not None


class TuningArgs(BaseArgs):
    tuning_method: TuningMethod
    prompt_tuning_args: dict | None = None
    lora_args: dict | None = None

    def model_post_init(self, __context: Any) -> None:
        assert self.prompt_tuning_args is None and self.lora_args is None, "PEFT is out of scope for this engine"


class TrainingParameters(BaseArgs):
    ignore_sampling_proportion_for_validation: bool = False
    num_training_steps: int | None = None
    gradient_accumulation_steps: int = 1
    eval_interval: int | None = None
    micro_batch_size: int = Field(default=None)
    sequence_length: int | None = None  # pretraining
    eval_during_training: bool = True
    loss_mask: LossMask = LossMask.output_only
    gradient_clipping: float | None = 1


class SaveArgs(BaseArgs):
    save_path: str
    save_interval: int | None = None
    save_optimizer: bool = True


class LoadArgs(BaseArgs):
    load_path: str
    iteration: int | None = None
    load_optimizer: bool = True
    load_lr_scheduler: bool = True
    load_rng_state: bool = True
    load_dataloader_state: bool = True
    load_experiments_tracker_state: bool = True
    load_starting_iteration: bool = True
    resume_learning_rate: bool = True


class DatasetArgs(BaseArgs):
    class_name: str
    class_args: dict = {}
    data_name: str | None = None
    input_format: str = "__input__"
    output_format: str = "__output__"
    data_sampling_ratio: int | None = None
    max_input_tokens: int | None = None
    max_output_tokens: int | None = None


class OptimizerArgs(BaseArgs):
    class_name: str = "TorchAdamW"
    params_group_method: str | None = None
    class_args: dict = {
        "lr": 1e-5,
        "weight_decay": 0.1,
        "betas": [0.9, 0.95],
        "eps": 1e-10,
    }

    def model_post_init(self, __context: Any) -> None:
        assert self.class_name in ("TorchAdamW", "TorchAdam"), (
            f"optimizer {self.class_name} is not implemented in the MI355X engine "
            "(fused HIP AdamW backs TorchAdamW/TorchAdam)"
        )
        assert self.params_group_method is None, "mup params groups not implemented yet"


class LRSchedulerArgs(BaseArgs):
    num_warmup_steps: int = 200
    num_constant_steps: int = 0
    num_decay_steps: int | None = None
    lr_decay_style: LRDecaySchedule = LRDecaySchedule.cosine
    lr_decay_factor: float = 0.1
    extra_lr_scheduler_args: dict = {}


class MixedPrecisionArgs(BaseArgs):
    dtype: str = "fp32"
    fp8_backend: str | None = None

    def model_post_init(self, __context: Any) -> None:
        self.dtype = {"fp32": "fp32", "float32": "fp32", "bf16": "bf16", "bfloat16": "bf16", "fp16": "fp16", "float16": "fp16"}[
            self.dtype
        ]
        assert self.fp8_backend is None, "fp8 is out of scope"


class ZeroTopologyArgs(BaseArgs):
    data_parallel_replication_world_size: int | None = None
    data_parallel_sharding_world_size: int | None = None


class DistributedArgs(BaseArgs):
    stage: int = 3
    distributed_backend: DistributedBackend = DistributedBackend.torch
    overlap_comm: bool = True
    contiguous_gradients: bool = False
    cpu_offload: bool = False
    gradient_checkpointing_method: str | None = None
    gradient_checkpointing_args: dict = {}
    zero_topology: ZeroTopologyArgs = ZeroTopologyArgs()
    zero_quantized_weights: bool = False
    zero_quantized_gradients: bool = False
    communication_dtype: str | None = None
    torch_compile: bool = False
    dispatching_dataloader: bool = False
    tensor_parallel_size: int = 1
    tensor_parallel_word_embeddings: bool = False
    sequence_parallel: bool = False
    data_parallel_size: int | None = None
    timeout_minutes: int | None = None
    fsdp_algorithm: int = 2

    def model_post_init(self, __context: Any) -> None:
        assert self.tensor_parallel_size == 1, "tensor parallel is out of scope (north_star: data-parallel only)"
        assert not self.sequence_parallel, "sequence parallel is a TP sub-mode — out of scope"
        assert not self.cpu_offload, "cpu_offload not implemented"
        assert not self.zero_quantized_weights and not self.zero_quantized_gradients
        assert not self.torch_compile, "the engine uses hand-written HIP kernels, not torch.compile"
        assert self.stage in (0, 1, 2, 3), self.stage
        if self.gradient_checkpointing_method is not None:
            assert self.gradient_checkpointing_method == "block", self.gradient_checkpointing_method


class LoggingArgs(BaseArgs):
    logging_level: str = "INFO"
    log_interval: int = 1
    aim_args: dict | None = None
    wandb_args: dict | None = None
    experiments_tracker_name: str | None = None
    use_colored_logs: bool = False
    torch_profiler_trace_path: str | None = None


class ResearchArgs(BaseArgs):
    scalar_attention: bool = False
    neft_alpha: float | None = None


class TrainingArgs(BaseArgs):
    random_args: RandomArgs = RandomArgs()
    tokenizer_args: TokenizerArgs = TokenizerArgs()
    model_args: ModelArgs
    tuning_args: TuningArgs
    optimizer_args: OptimizerArgs = OptimizerArgs()
    lr_scheduler_args: LRSchedulerArgs = LRSchedulerArgs()
    datasets: list[DatasetArgs] = []
    save_args: SaveArgs | None = None
    load_args: LoadArgs | None = None
    training_parameters: TrainingParameters | None = None
    logging_args: LoggingArgs = LoggingArgs()
    mixed_precision_args: MixedPrecisionArgs = MixedPrecisionArgs()
    distributed_args: DistributedArgs = DistributedArgs()
    research_args: ResearchArgs = ResearchArgs()

    def model_post_init(self, __context: Any) -> None:
        if self.model_args is not None and self.model_args.use_padding_free_transformer:
            assert (
                self.model_args.attention_implementation == AttentionImplementation.flash_attention_2
            ), "padding free transformer only works with flash attention"


def load_yaml(path: str) -> dict:
    with open(path) as f:
        return yaml.safe_load(f)


def get_args(config: str | dict, mode: Mode = Mode.training) -> TrainingArgs:
    if isinstance(config, str):
        config = load_yaml(config)
    assert mode == Mode.training
    return TrainingArgs(**config)


def parse_args(argv: list[str] | None = None) -> TrainingArgs:
    parser = argparse.ArgumentParser()
    parser.add_argument("--config", type=str, required=True)
    ns = parser.parse_args(argv)
    return get_args(ns.config)
