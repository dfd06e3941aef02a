"""Data pipeline (thin, round 1): synthetic Megatron-shaped pretraining
batches ({"text": (micro_batch_size, sequence_length+1) int64}, reference
data/megatron returns sequence_length+1 windows), and a list-packing
finetune collator (data/utils.py:8-93, padding-free branch).

The on-disk Megatron mmap .bin/.idx pipeline is a "next" row.
"""

import torch

from .utils import get_rank


class SyntheticPretrainingDataLoader:
    """Deterministic synthetic token stream: seed varies per (rank, step) so
    ranks see different data (bench contract: data='synthetic')."""

    def __init__(self, micro_batch_size: int, sequence_length: int, vocab_size: int, seed: int = 1234):
        self.micro_batch_size = micro_batch_size
        self.sequence_length = sequence_length
        self.vocab_size = vocab_size
        self.seed = seed
        self.step = 0

    def state_dict(self):
        return {"step": self.step}

    def load_state_dict(self, sd):
        self.step = sd["step"]

    def __iter__(self):
        return self

    def __next__(self):
        g = torch.Generator().manual_seed(self.seed + 100003 * get_rank() + self.step)
        self.step += 1
        text = torch.randint(
            0, self.vocab_size, (self.micro_batch_size, self.sequence_length + 1), generator=g, dtype=torch.int64
        )
        return {"text": text}


def collate_padding_free(examples: list[dict], loss_mask_output_only: bool = True) -> dict:
    """Padding-free finetune collate (data/utils.py:50-61): lists of unpadded
    ids; labels -100 on the prompt when loss_mask == output_only."""
    input_ids = []
    labels = []
    for ex in examples:
        inp = list(ex["input"]) + list(ex["output"])
        input_ids.append(inp)
        if loss_mask_output_only:
            labels.append([-100] * len(ex["input"]) + list(ex["output"]))
        else:
            labels.append(list(inp))
    return {"input_ids": input_ids, "labels": labels}
