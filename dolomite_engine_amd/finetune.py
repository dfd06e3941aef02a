"""Finetuning entry point — drop-in for `python -m dolomite_engine.finetune
--config x.yml` (reference finetune.py:214-311). Loss computed inside the
model (shift + document-boundary drops) on padding-free list batches, or on
padded dense batches for eager/sdpa."""

import time

import torch

from .arguments import TrainingArgs, parse_args
from .checkpointing import save_checkpoint
from .data import get_finetuning_dataloader
from .model_wrapper import ModelWrapperForFinetuning
from .optimization import get_scheduler
from .train_utils import train_step
from .utils import get_rank, init_distributed, log_rank_0
from .zero import ZeRO2Engine


def build_engine(args: TrainingArgs):
    model_wrapper = ModelWrapperForFinetuning(
        model_name=args.model_args.model_name,
        pretrained_config=args.model_args.pretrained_config,
        dtype=args.mixed_precision_args.dtype,
        attention_implementation=(
            args.model_args.attention_implementation.value
            if args.model_args.attention_implementation is not None
            else "sdpa"
        ),
        use_padding_free_transformer=args.model_args.use_padding_free_transformer,
    )
    if args.distributed_args.gradient_checkpointing_method is not None:
        from .hf_models import apply_gradient_checkpointing

        apply_gradient_checkpointing(
            model_wrapper.model,
            args.distributed_args.gradient_checkpointing_method,
            **args.distributed_args.gradient_checkpointing_args,
        )
    if torch.cuda.is_available():
        model_wrapper.model.cuda()

    oa = args.optimizer_args.class_args
    engine = ZeRO2Engine(
        model_wrapper.model,
        lr=oa.get("lr", 1e-5),
        betas=tuple(oa.get("betas", (0.9, 0.95))),
        eps=oa.get("eps", 1e-10),
        weight_decay=oa.get("weight_decay", 0.1),
        overlap_comm=args.distributed_args.overlap_comm,
    )
    lr_scheduler = get_scheduler(
        oa.get("lr", 1e-5), args.lr_scheduler_args, args.training_parameters.num_training_steps
    )
    return model_wrapper, engine, lr_scheduler


def main(argv=None):
    args = parse_args(argv)
    init_distributed(args.distributed_args.timeout_minutes)
    torch.manual_seed(args.random_args.seed + get_rank())

    model_wrapper, engine, lr_scheduler = build_engine(args)
    tp = args.training_parameters

    train_iter = get_finetuning_dataloader(
        args.datasets,
        micro_batch_size=tp.micro_batch_size,
        use_padding_free_transformer=args.model_args.use_padding_free_transformer,
        loss_mask_output_only=tp.loss_mask.value == "output_only",
        seed=args.random_args.seed,
        pad_token_id=model_wrapper.config.pad_token_id,
    )

    model_wrapper.train()
    for global_step in range(1, tp.num_training_steps + 1):
        t0 = time.perf_counter()
        loss, grad_norm = train_step(
            model_wrapper, engine, lr_scheduler, train_iter, tp.gradient_accumulation_steps, tp.gradient_clipping
        )
        dt = time.perf_counter() - t0
        if global_step % args.logging_args.log_interval == 0:
            log_rank_0(
                f"step = {global_step}, loss_step = {loss:.5f}, grad_norm = {grad_norm:.3f}, "
                f"learning_rate = {lr_scheduler.get_lr():.3e}, step time (sec) = {dt:.3f}"
            )
        if args.save_args is not None and args.save_args.save_interval and global_step % args.save_args.save_interval == 0:
            save_checkpoint(
                args.save_args.save_path, global_step, model_wrapper,
                engine if args.save_args.save_optimizer else None, lr_scheduler,
                metadata={}, args_dict=args.to_dict(), save_optimizer=args.save_args.save_optimizer,
            )

    if args.save_args is not None:
        save_checkpoint(
            args.save_args.save_path, tp.num_training_steps, model_wrapper,
            engine if args.save_args.save_optimizer else None, lr_scheduler,
            metadata={}, args_dict=args.to_dict(), save_optimizer=args.save_args.save_optimizer,
        )


if __name__ == "__main__":
    main()
