"""LR schedules — restatement of the reference's closures
(optimization/scheduler.py:9-182), pinned by tests/golden/scheduler.pt.
The optimizer itself is the fused HIP AdamW inside ZeRO2Engine (zero.py)."""

import math


def _linear(m, c, x):
    return m * x + c


def _cosine(a, b, t, x):
    return a * (1 + math.cos(math.pi * x / t)) / 2 + b


def _exponential(a, b, t, x):
    return a * math.exp(-x / t) + b


def _power(a, b, x):
    return a * (x**b)


class LRScheduler:
    """lr(step) = base_lr * factor(step); counter semantics match the
    reference's LambdaLR wiring (factor(0) applies to the first optimizer
    step, .step() advances after each optimizer step)."""

    def __init__(
        self,
        base_lr: float,
        num_warmup_steps: int,
        num_constant_steps: int,
        num_decay_steps: int | None,
        num_training_steps: int | None,
        lr_decay_style: str,
        lr_decay_factor: float,
        extra_lr_scheduler_args: dict | None = None,
    ):
        self.base_lr = base_lr
        self.style = lr_decay_style
        self.lr_warmup_boundary = num_warmup_steps
        self.lr_constant_boundary = self.lr_warmup_boundary + num_constant_steps
        self.lr_decay_boundary = num_training_steps
        if num_decay_steps is not None:
            self.lr_decay_boundary = self.lr_constant_boundary + num_decay_steps
        self.lr_decay_factor = lr_decay_factor
        self.extra = extra_lr_scheduler_args or {}
        self._step = 0

        if self.style == "constant":
            assert num_decay_steps in (0, None), "num_decay_steps should be 0 for constant schedule"
        if self.style == "power":
            assert num_constant_steps == 0, "num_constant_steps should be 0 for power law scheduler"
            a, b, c = self.extra["a"], self.extra["b"], self.extra["c"]
            self._max_lr_during_warmup = min(1, _power(a=a / base_lr, b=b, x=num_warmup_steps * c))

    def _factor(self, n: int) -> float:
        wb, cb, db, f = self.lr_warmup_boundary, self.lr_constant_boundary, self.lr_decay_boundary, self.lr_decay_factor
        if self.style == "constant":
            return _linear(1 / wb, 0, n) if (wb > 0 and n <= wb) else 1
        if self.style == "cosine":
            if wb > 0 and n <= wb:
                return _linear(1 / wb, 0, n)
            if n <= cb:
                return 1
            if n <= db:
                return _cosine(a=1 - f, b=f, t=db - cb, x=n - cb)
            return f
        if self.style == "linear":
            if wb > 0 and n <= wb:
                return _linear(1 / wb, 0, n)
            if n <= cb:
                return 1
            if n <= db:
                return _linear(m=(f - 1) / (db - cb), c=1, x=n - cb)
            return f
        if self.style == "exponential":
            if wb > 0 and n <= wb:
                return _linear(1 / wb, 0, n)
            if n <= cb:
                return 1
            return _exponential(
                a=(1 - f) * math.e / (math.e - 1), b=(f * math.e - 1) / (math.e - 1), t=db - cb, x=n - cb
            )
        if self.style == "power":
            a, b, c = self.extra["a"], self.extra["b"], self.extra["c"]
            if wb > 0 and n <= wb:
                return _linear(m=self._max_lr_during_warmup / wb, c=0, x=n)
            return min(1, _power(a=a / self.base_lr, b=b, x=n * c))
        raise ValueError(f"invalid lr_decay_style ({self.style})")

    def get_lr(self) -> float:
        return self.base_lr * self._factor(self._step)

    def step(self) -> None:
        self._step += 1

    def state_dict(self) -> dict:
        return {"step": self._step}

    def load_state_dict(self, sd: dict) -> None:
        self._step = sd["step"]


def get_scheduler(base_lr: float, args, num_training_steps: int | None = None) -> LRScheduler:
    return LRScheduler(
        base_lr=base_lr,
        num_warmup_steps=args.num_warmup_steps,
        num_constant_steps=args.num_constant_steps,
        num_decay_steps=args.num_decay_steps,
        num_training_steps=num_training_steps,
        lr_decay_style=args.lr_decay_style.value if hasattr(args.lr_decay_style, "value") else args.lr_decay_style,
        lr_decay_factor=args.lr_decay_factor,
        extra_lr_scheduler_args=args.extra_lr_scheduler_args,
    )
