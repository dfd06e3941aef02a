"""Checkpoint save/load — reference checkpointing.py:50-263 layout:

  save_path/global_step{N}/
    model/                      (rank-0 full model save_pretrained)
    optimizer/optimizer-{rank}.pt   (ZeRO-2 per-rank shard states)
    lr_scheduler.pt
    rng_state/rng_state-{rank}.pt
    dataloader/dataloader-{rank}.pt
    metadata.json
    training_config.yml
  save_path/latest_checkpointed_iteration.json
"""

import json
import random
from pathlib import Path

import numpy as np
import torch
import torch.distributed as dist
import yaml

from .utils import get_rank, is_initialized


def _step_dir(save_path: str, global_step: int) -> Path:
    return Path(save_path) / f"global_step{global_step}"


def save_checkpoint(
    save_path: str,
    global_step: int,
    model_wrapper,
    engine,
    lr_scheduler,
    train_loader=None,
    metadata: dict | None = None,
    args_dict: dict | None = None,
    save_optimizer: bool = True,
) -> None:
    rank = get_rank()
    d = _step_dir(save_path, global_step)
    (d / "rng_state").mkdir(parents=True, exist_ok=True)
    (d / "optimizer").mkdir(exist_ok=True)
    (d / "dataloader").mkdir(exist_ok=True)

    if rank == 0:
        model_wrapper.save_pretrained(str(d / "model"))
        torch.save(lr_scheduler.state_dict(), d / "lr_scheduler.pt")
        with open(d / "metadata.json", "w") as f:
            json.dump(metadata or {}, f)
        if args_dict is not None:
            with open(d / "training_config.yml", "w") as f:
                yaml.safe_dump(args_dict, f)

    if save_optimizer and engine is not None:
        torch.save(engine.state_dict(), d / "optimizer" / f"optimizer-{rank}.pt")

    rng = {
        "random_rng_state": random.getstate(),
        "np_rng_state": np.random.get_state(),
        "torch_rng_state": torch.get_rng_state(),
        "cuda_rng_state": torch.cuda.get_rng_state() if torch.cuda.is_available() else None,
    }
    torch.save(rng, d / "rng_state" / f"rng_state-{rank}.pt")

    if train_loader is not None and hasattr(train_loader, "state_dict"):
        torch.save(train_loader.state_dict(), d / "dataloader" / f"dataloader-{rank}.pt")

    if is_initialized():
        dist.barrier()
    if rank == 0:
        with open(Path(save_path) / "latest_checkpointed_iteration.json", "w") as f:
            json.dump({"latest_checkpointed_iteration": global_step}, f)


def latest_iteration(load_path: str) -> int:
    with open(Path(load_path) / "latest_checkpointed_iteration.json") as f:
        return json.load(f)["latest_checkpointed_iteration"]


def load_checkpoint_for_training(
    load_path: str,
    model_wrapper,
    engine,
    lr_scheduler,
    train_loader=None,
    iteration: int | None = None,
    load_optimizer: bool = True,
    load_lr_scheduler: bool = True,
    load_rng_state: bool = True,
    load_dataloader_state: bool = True,
) -> tuple[int, dict]:
    rank = get_rank()
    it = iteration if iteration is not None else latest_iteration(load_path)
    d = _step_dir(load_path, it)

    sd = GPTDolomite_load_state_dict(d / "model")
    # Strict load except for tied weights the save path deduplicates
    # (lm_head.weight aliases the embedding when tie_word_embeddings):
    # resuming from a checkpoint with missing/renamed keys must fail loudly,
    # not silently train from partially-initialized weights.
    missing, unexpected = model_wrapper.model.load_state_dict(sd, strict=False)
    tied_ok = {"lm_head.weight"} if getattr(model_wrapper.model.config, "tie_word_embeddings", True) else set()
    bad_missing = [k for k in missing if k not in tied_ok]
    if bad_missing or unexpected:
        raise RuntimeError(
            f"checkpoint/model mismatch loading {d}: missing={bad_missing} unexpected={list(unexpected)}"
        )
    # repointed flat buffers: copy loaded values into the engine's storage
    if engine is not None:
        for b in engine.buckets:
            b.master.copy_(b.flat_param[b.shard_slice].float())

    if load_optimizer and engine is not None:
        opt_dir = d / "optimizer"
        saved = sorted(opt_dir.glob("optimizer-*.pt"), key=lambda p: int(p.stem.split("-")[1]))
        sd0 = torch.load(saved[min(rank, len(saved) - 1)], weights_only=False)
        if sd0["world_size"] != engine.world:
            # world size changed: reshard from every saved rank's shard
            all_shards = [torch.load(p, weights_only=False) for p in saved]
            engine.load_state_dict(all_shards[0], all_shards=all_shards)
        else:
            engine.load_state_dict(torch.load(opt_dir / f"optimizer-{rank}.pt", weights_only=False))
    if load_lr_scheduler:
        lr_scheduler.load_state_dict(torch.load(d / "lr_scheduler.pt", weights_only=False))
    if load_rng_state:
        rng = torch.load(d / "rng_state" / f"rng_state-{rank}.pt", weights_only=False)
        random.setstate(rng["random_rng_state"])
        np.random.set_state(rng["np_rng_state"])
        torch.set_rng_state(rng["torch_rng_state"])
        if rng["cuda_rng_state"] is not None and torch.cuda.is_available():
            torch.cuda.set_rng_state(rng["cuda_rng_state"])
    if load_dataloader_state and train_loader is not None:
        p = d / "dataloader" / f"dataloader-{rank}.pt"
        if p.exists():
            train_loader.load_state_dict(torch.load(p, weights_only=False))

    with open(d / "metadata.json") as f:
        metadata = json.load(f)
    return it, metadata


def GPTDolomite_load_state_dict(model_dir: Path) -> dict:
    from safetensors.torch import load_file

    files = sorted(Path(model_dir).glob("*.safetensors"))
    sd = {}
    for f in files:
        sd.update(load_file(str(f)))
    if not sd:  # pytorch_model.bin fallback
        sd = torch.load(Path(model_dir) / "pytorch_model.bin", weights_only=False)
    return sd
