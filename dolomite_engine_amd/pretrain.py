"""Pretraining entry point — drop-in for `python -m dolomite_engine.pretrain
--config x.yml` (reference pretrain.py:283-375), launched under torchrun with
one rank per GPU over RCCL."""

import time

import torch

from .arguments import TrainingArgs, parse_args
from .checkpointing import load_checkpoint_for_training, save_checkpoint
from .data import SyntheticPretrainingDataLoader
from .model_wrapper import ModelWrapperForPretraining
from .optimization import get_scheduler
from .train_utils import get_model_tflops, train_step
from .utils import get_rank, get_world_size, init_distributed
from .zero import ZeRO2Engine


def build_engine(args: TrainingArgs):
    """model wrapper + ZeRO-2 engine + scheduler from TrainingArgs."""
    tp = args.training_parameters
    model_wrapper = ModelWrapperForPretraining(
        micro_batch_size=tp.micro_batch_size,
        sequence_length=tp.sequence_length,
        reset_attention_mask=args.model_args.reset_attention_mask,
        reset_position_ids=args.model_args.reset_position_ids,
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
    lr_scheduler = get_scheduler(oa.get("lr", 1e-5), args.lr_scheduler_args, tp.num_training_steps)
    return model_wrapper, engine, lr_scheduler


@torch.no_grad()
def evaluate(val_iter, model_wrapper, global_step: int, eval_steps: int) -> float:
    """Validation loop (reference pretrain.py:221-279): mean loss over
    eval_steps batches, AVG-reduced over the data-parallel group."""
    import torch.distributed as dist

    from .utils import is_initialized

    model_wrapper.eval()
    loss_sum = 0
    for _ in range(eval_steps):
        batch = next(val_iter)
        loss_sum = loss_sum + model_wrapper(batch)
    loss_mean = loss_sum / eval_steps
    if is_initialized():
        dist.all_reduce(loss_mean, op=dist.ReduceOp.AVG if dist.get_backend() == "nccl" else dist.ReduceOp.SUM)
        if dist.get_backend() != "nccl":
            loss_mean = loss_mean / dist.get_world_size()
    loss_mean = loss_mean.item()
    from .tracking import ExperimentsTracker, track_val_metrics

    track_val_metrics(global_step, loss_mean, ExperimentsTracker())
    model_wrapper.train()
    return loss_mean


def train(args: TrainingArgs, model_wrapper, engine, lr_scheduler, train_loader, starting_step=0, metadata=None,
          val_loader=None, eval_steps: int = 8):
    tp = args.training_parameters
    ga = tp.gradient_accumulation_steps
    world = get_world_size()
    tokens_per_step = tp.micro_batch_size * tp.sequence_length * ga * world
    tflops_per_step = (
        get_model_tflops(
            model_wrapper.config,
            tp.micro_batch_size,
            tp.sequence_length,
            args.distributed_args.gradient_checkpointing_method is not None,
        )
        * ga
        * world
    )

    from .tracking import ExperimentsTracker, RunningMean, track_train_metrics

    tracker = ExperimentsTracker(
        args.logging_args.experiments_tracker_name,
        args.save_args.save_path if args.save_args is not None else None,
    )
    running = RunningMean()

    train_iter = iter(train_loader)
    model_wrapper.train()

    # torch profiler hook (reference train_utils.py:182-194: wait 5 / warmup
    # 5 / active 1, rank 0, tensorboard trace)
    profiler = None
    if args.logging_args.torch_profiler_trace_path is not None and get_rank() == 0:
        profiler = torch.profiler.profile(
            schedule=torch.profiler.schedule(wait=5, warmup=5, active=1, repeat=1),
            on_trace_ready=torch.profiler.tensorboard_trace_handler(args.logging_args.torch_profiler_trace_path),
            record_shapes=True,
            profile_memory=True,
        )
        profiler.__enter__()

    for global_step in range(starting_step + 1, tp.num_training_steps + 1):
        t0 = time.perf_counter()
        loss, grad_norm = train_step(model_wrapper, engine, lr_scheduler, train_iter, ga, tp.gradient_clipping)
        if profiler is not None:
            profiler.step()
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0

        loss_mean = running.add_loss(loss)
        if global_step % args.logging_args.log_interval == 0:
            track_train_metrics(
                global_step,
                loss,
                grad_norm,
                lr_scheduler.get_lr(),
                tracker,
                loss_mean,
                flops=tflops_per_step * 1e12 / dt,
                billion_tokens_per_day=tokens_per_step / dt * 86400 / 1e9,
                step_time=dt,
            )

        if (
            tp.eval_during_training
            and tp.eval_interval is not None
            and val_loader is not None
            and global_step % tp.eval_interval == 0
        ):
            evaluate(iter(val_loader), model_wrapper, global_step, eval_steps=eval_steps)

        if args.save_args is not None and args.save_args.save_interval and global_step % args.save_args.save_interval == 0:
            save_checkpoint(
                args.save_args.save_path,
                global_step,
                model_wrapper,
                engine if args.save_args.save_optimizer else None,
                lr_scheduler,
                train_loader,
                metadata={"consumed_samples": global_step * tp.micro_batch_size * ga * world},
                args_dict=args.to_dict(),
                save_optimizer=args.save_args.save_optimizer,
            )

    if profiler is not None:
        profiler.__exit__(None, None, None)

    if args.save_args is not None:
        save_checkpoint(
            args.save_args.save_path,
            tp.num_training_steps,
            model_wrapper,
            engine if args.save_args.save_optimizer else None,
            lr_scheduler,
            train_loader,
            metadata={"consumed_samples": tp.num_training_steps * tp.micro_batch_size * ga * world},
            args_dict=args.to_dict(),
            save_optimizer=args.save_args.save_optimizer,
        )


def main(argv=None):
    args = parse_args(argv)
    init_distributed(args.distributed_args.timeout_minutes)
    torch.manual_seed(args.random_args.seed + get_rank())

    model_wrapper, engine, lr_scheduler = build_engine(args)

    tp = args.training_parameters
    val_loader = None
    eval_steps = 8
    if args.datasets and args.datasets[0].class_name == "MegatronDataset":
        # Full reference wiring (data/megatron/__init__.py:18-110): weighted
        # blend across data_path entries, document-level train/val/test
        # split, eval_steps from class_args.
        from .megatron import MegatronDataLoader, build_train_val_test_datasets, train_val_test_samples

        ca = args.datasets[0].class_args
        eval_steps = int(ca.get("eval_steps", 8))
        world = get_world_size()
        sizes = train_val_test_samples(
            tp.num_training_steps, tp.micro_batch_size, tp.gradient_accumulation_steps,
            tp.eval_interval if tp.eval_during_training else None, eval_steps, world,
        )
        fim_rate = float(ca.get("fim_rate", 0))
        tok = None
        if fim_rate:
            from transformers import AutoTokenizer

            tok = AutoTokenizer.from_pretrained(args.tokenizer_args.tokenizer_name)
            # megatron-tokenizer façade the FIM transform expects
            tok.eod = tok.eos_token_id
            tok.detokenize = tok.decode
            tok.tokenize = lambda s_, _t=tok: _t.encode(s_, add_special_tokens=False)
        train_ds, val_ds, _test_ds = build_train_val_test_datasets(
            ca["data_path"], ca.get("split", "969,30,1"), sizes,
            seq_length=tp.sequence_length, seed=int(ca.get("seed", args.random_args.seed)),
            fim_rate=fim_rate, fim_spm_rate=float(ca.get("fim_spm_rate", 0.5)), tokenizer=tok,
        )
        train_loader = MegatronDataLoader(train_ds, tp.micro_batch_size)
        if val_ds is not None and tp.eval_during_training and tp.eval_interval is not None:
            val_loader = MegatronDataLoader(val_ds, tp.micro_batch_size)
    else:
        train_loader = SyntheticPretrainingDataLoader(
            tp.micro_batch_size, tp.sequence_length, model_wrapper.config.vocab_size, seed=args.random_args.seed
        )

    starting_step, metadata = 0, None
    if args.load_args is not None:
        starting_step, metadata = load_checkpoint_for_training(
            args.load_args.load_path,
            model_wrapper,
            engine,
            lr_scheduler,
            train_loader,
            iteration=args.load_args.iteration,
            load_optimizer=args.load_args.load_optimizer,
            load_lr_scheduler=args.load_args.load_lr_scheduler,
            load_rng_state=args.load_args.load_rng_state,
            load_dataloader_state=args.load_args.load_dataloader_state,
        )
        if not args.load_args.load_starting_iteration:
            starting_step = 0

    if val_loader is None and tp.eval_during_training and tp.eval_interval is not None:
        # held-out synthetic stream (different seed space than training)
        val_loader = SyntheticPretrainingDataLoader(
            tp.micro_batch_size, tp.sequence_length, model_wrapper.config.vocab_size,
            seed=args.random_args.seed + 777777,
        )

    train(args, model_wrapper, engine, lr_scheduler, train_loader, starting_step, metadata,
          val_loader=val_loader, eval_steps=eval_steps)


if __name__ == "__main__":
    main()
