"""Checkpoint -> consolidated safetensors model dir (reference unshard.py +
checkpointing.py:266-402, DP variant: ZeRO-2 keeps full bf16 params on every
rank and rank 0 saves a consolidated model/, so unsharding re-exports it)."""

import argparse
import shutil
from pathlib import Path

from .checkpointing import latest_iteration


def unshard_checkpoint(load_path: str, iteration: int | None, save_path: str) -> None:
    it = iteration if iteration is not None else latest_iteration(load_path)
    model_dir = Path(load_path) / f"global_step{it}" / "model"
    assert model_dir.exists(), f"{model_dir} not found"
    out = Path(save_path)
    out.mkdir(parents=True, exist_ok=True)
    for f in model_dir.iterdir():
        shutil.copy2(f, out / f.name)


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--load-path", required=True)
    p.add_argument("--iteration", type=int, default=None)
    p.add_argument("--save-path", required=True)
    ns = p.parse_args(argv)
    unshard_checkpoint(ns.load_path, ns.iteration, ns.save_path)


if __name__ == "__main__":
    main()
