"""Experiments tracking — reference utils/tracking.py:42-148 surface with an
offline JSONL backend (aim/wandb need network; the tracker name selects the
sink). Metric names match the reference exactly
(train_utils.py:119-179: loss_step, loss_running_mean, learning_rate,
grad_norm, FLOPS, "throughput (B tokens/day)", "step time (sec)")."""

import json
import logging
from pathlib import Path

from .utils import get_rank, log_rank_0


class RunningMean:
    """reference utils/miscellaneous RunningMean (window 100)."""

    def __init__(self, window: int = 100):
        self.window = window
        self.values = []

    def add_loss(self, v: float) -> float:
        self.values.append(v)
        if len(self.values) > self.window:
            self.values.pop(0)
        return sum(self.values) / len(self.values)


class ExperimentsTracker:
    """reference utils/experiments_tracker.py surface: tracker_name selects
    aim / wandb (used when the library is importable, as the reference
    does) with the offline 'jsonl' sink always available."""

    def __init__(self, tracker_name: str | None = None, run_dir: str | None = None, **init_kwargs):
        self.enabled = get_rank() == 0
        self._fh = None
        self._aim_run = None
        self._wandb = None
        if not self.enabled:
            return
        if tracker_name == "jsonl" and run_dir is not None:
            Path(run_dir).mkdir(parents=True, exist_ok=True)
            self._fh = open(Path(run_dir) / "metrics.jsonl", "a")
        elif tracker_name == "aim":
            try:
                from aim import Run  # noqa: PLC0415

                self._aim_run = Run(repo=init_kwargs.get("repo", run_dir), experiment=init_kwargs.get("experiment"))
            except ImportError:
                log_rank_0("tracker 'aim' requested but aim is not installed; falling back to logs", logging.WARNING)
        elif tracker_name == "wandb":
            try:
                import wandb  # noqa: PLC0415

                self._wandb = wandb
                wandb.init(**init_kwargs)
            except ImportError:
                log_rank_0("tracker 'wandb' requested but wandb is not installed; falling back to logs", logging.WARNING)
        elif tracker_name is not None:
            log_rank_0(f"unknown tracker '{tracker_name}'; falling back to logs", logging.WARNING)

    def track(self, values: dict, step: int | None = None, context: str | None = None) -> None:
        if self._fh is not None:
            self._fh.write(json.dumps({"step": step, "context": context, **values}) + "\n")
            self._fh.flush()
        if self._aim_run is not None:
            for k, v in values.items():
                self._aim_run.track(v, name=k, step=step, context={"subset": context} if context else None)
        if self._wandb is not None:
            prefix = f"{context}/" if context else ""
            self._wandb.log({f"{prefix}{k}": v for k, v in values.items()}, step=step)

    def finish(self) -> None:
        if self._fh is not None:
            self._fh.close()
        if self._aim_run is not None:
            self._aim_run.close()
        if self._wandb is not None:
            self._wandb.finish()


def track_train_metrics(
    global_step: int,
    train_loss_step: float,
    grad_norm_step: float,
    current_lr: float,
    experiments_tracker: ExperimentsTracker,
    loss_running_mean: float,
    flops: float | None = None,
    billion_tokens_per_day: float | None = None,
    step_time: float | None = None,
) -> None:
    """Reference train_utils.py:119-179 metric names and terminal format."""
    message = {"loss_step": train_loss_step, "loss_running_mean": loss_running_mean, "learning_rate": current_lr}
    if grad_norm_step is not None:
        message["grad_norm"] = grad_norm_step
    if flops is not None:
        message["FLOPS"] = flops
    if billion_tokens_per_day is not None:
        message["throughput (B tokens/day)"] = billion_tokens_per_day
    if step_time is not None:
        message["step time (sec)"] = step_time
    experiments_tracker.track(message, step=global_step, context="train")

    text = (
        f"step = {global_step}, train_loss (batch) = {train_loss_step:.4f}, "
        f"train_loss (running_mean) = {loss_running_mean:.4f}, "
        f"learning_rate = {current_lr:.3E}"
    )
    if grad_norm_step is not None:
        text += f", grad_norm = {grad_norm_step:.2f}"
    if flops is not None:
        text += f", FLOPS = {flops:.2f}"
    if billion_tokens_per_day is not None:
        text += f", throughput = {billion_tokens_per_day:.2f} B tokens/day"
    if step_time is not None:
        text += f", step_time = {step_time:.3f} sec"
    log_rank_0(text)


def track_val_metrics(global_step: int, val_loss: float, experiments_tracker: ExperimentsTracker, group_name=None):
    """Reference train_utils track_val_metrics."""
    experiments_tracker.track({"loss": val_loss}, step=global_step, context="val")
    log_rank_0(f"step = {global_step}, val_loss = {val_loss:.4f}" + (f" ({group_name})" if group_name else ""))
