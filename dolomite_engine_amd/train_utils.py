"""Train-step and metrics (reference train_utils.py:18-236)."""

import torch
import torch.distributed as dist

from .utils import is_initialized
from .zero import ZeRO2Engine


def get_next_batch(it):
    return next(it)


def train_step(
    model_wrapper,
    engine: ZeRO2Engine,
    lr_scheduler,
    train_iter,
    gradient_accumulation_steps: int,
    gradient_clipping: float | None,
) -> tuple[float, float]:
    """Grad-accum microsteps without sync, boundary microstep with overlapped
    reduce-scatter, clip, fused AdamW step, param all-gather
    (reference train_utils.py:18-116; loss semantics identical: full loss
    backwarded each microstep, summed loss / ga reported, AVG over ranks)."""
    engine.zero_grad()
    engine.set_sync(False)

    loss = torch.zeros((), device=engine.device)
    for _ in range(gradient_accumulation_steps - 1):
        batch = get_next_batch(train_iter)
        loss_micro = model_wrapper(batch)
        loss = loss + loss_micro.detach()
        loss_micro.backward()

    engine.set_sync(True)
    batch = get_next_batch(train_iter)
    loss_micro = model_wrapper(batch)
    loss = loss + loss_micro.detach()
    loss_micro.backward()

    grad_norm = engine.step(lr=lr_scheduler.get_lr(), grad_clip=gradient_clipping)
    lr_scheduler.step()

    loss = loss / gradient_accumulation_steps
    if is_initialized():
        dist.all_reduce(loss, op=dist.ReduceOp.AVG if dist.get_backend() == "nccl" else dist.ReduceOp.SUM)
        if dist.get_backend() != "nccl":
            loss = loss / dist.get_world_size()

    return loss.item(), 0.0 if grad_norm is None else float(grad_norm)


def get_model_tflops(config, batch_size: int, sequence_length: int, gradient_checkpointing: bool = False) -> float:
    """Analytic flops per step in TFLOP (reference train_utils.py:197-236)."""
    b, s = batch_size, sequence_length
    h, f, n, l, v = config.n_embd, config.n_inner, config.n_head, config.n_layer, config.vocab_size
    k = config.num_key_value_heads
    glu = config.activation_function.endswith("glu")

    mlp_flops = 4 * b * s * h * f
    if glu:
        mlp_flops += 2 * b * s * h * f
    attention_flops = 4 * b * s * h * (h * (1 + k / n) + s)
    forward_flops = attention_flops + mlp_flops
    backward_flops = (3 if gradient_checkpointing else 2) * forward_flops
    model_flops = l * (forward_flops + backward_flops)
    model_flops += 6 * b * s * h * v
    return model_flops / 1e12
