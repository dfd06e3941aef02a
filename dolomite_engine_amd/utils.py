"""Process-group + logging utilities (reference: utils/parallel.py,
utils/__init__.py:28-59, utils/logger.py). Data-parallel only: the engine's
process group is the world group over RCCL (backend "nccl" on ROCm) or gloo
on CPU test hosts."""

import logging
import os
from datetime import timedelta

import torch
import torch.distributed as dist

logger = logging.getLogger("dolomite_engine_amd")


def log_rank_0(msg: str, level: int = logging.INFO) -> None:
    if get_rank() == 0:
        logger.log(level, msg)
        if not logger.handlers:
            print(msg, flush=True)


def is_initialized() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if is_initialized() else 0


def get_world_size() -> int:
    return dist.get_world_size() if is_initialized() else 1


def init_distributed(timeout_minutes: int | None = None, backend: str | None = None) -> None:
    """torchrun-launched init (reference utils/parallel.py:46-76). One process
    per GPU over RCCL; falls back to gloo when no GPU (CPU tests)."""
    if is_initialized():
        return
    if "RANK" not in os.environ:
        # single-process run without torchrun
        return
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    kwargs = {}
    if timeout_minutes is not None:
        kwargs["timeout"] = timedelta(minutes=timeout_minutes)
    dist.init_process_group(backend=backend, **kwargs)
    if torch.cuda.is_available():
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))


def string_to_torch_dtype(s: str) -> torch.dtype:
    return {"fp32": torch.float32, "bf16": torch.bfloat16, "fp16": torch.float16}[s]
