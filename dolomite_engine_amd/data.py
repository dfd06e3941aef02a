"""Data pipeline: synthetic Megatron-shaped pretraining batches
({"text": (micro_batch_size, sequence_length+1) int64}, reference
data/megatron returns sequence_length+1 windows), and a list-packing
finetune collator (data/utils.py:8-93, padding-free branch).

The on-disk Megatron mmap .bin/.idx pipeline (blending, splits, FIM,
native index builders) lives in megatron.py.
"""

import torch

from .utils import get_rank, get_world_size


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


class DebugDataset:
    """Synthetic finetune examples (reference data/debugging.py analogue):
    deterministic token-id lists; no tokenizer/network needed."""

    def __init__(self, num_examples=256, max_input_tokens=32, max_output_tokens=16, vocab_size=512, seed=42):
        g = torch.Generator().manual_seed(seed)
        self.examples = []
        for _ in range(num_examples):
            li = int(torch.randint(2, max_input_tokens, (1,), generator=g))
            lo = int(torch.randint(1, max_output_tokens, (1,), generator=g))
            self.examples.append(
                {
                    "input": torch.randint(0, vocab_size, (li,), generator=g).tolist(),
                    "output": torch.randint(0, vocab_size, (lo,), generator=g).tolist(),
                }
            )

    def __len__(self):
        return len(self.examples)

    def __getitem__(self, i):
        return self.examples[i]


class JSONLinesTokenizedDataset:
    """Pre-tokenized jsonl: rows {"input_ids": [...], "output_ids": [...]}.
    (The reference tokenizes raw jsonl with an AutoTokenizer at load time,
    data/huggingface.py — there is no network here, so this engine consumes
    pre-tokenized rows; the collate/loss-mask semantics are identical.)"""

    def __init__(self, data_path: str):
        import json as _json

        self.examples = []
        with open(data_path) as f:
            for line in f:
                if line.strip():
                    row = _json.loads(line)
                    self.examples.append({"input": row["input_ids"], "output": row["output_ids"]})

    def __len__(self):
        return len(self.examples)

    def __getitem__(self, i):
        return self.examples[i]


_DATASET_CLASSES = {"DebugDataset": DebugDataset, "JSONLinesTokenizedDataset": JSONLinesTokenizedDataset}


def collate_padded(examples, pad_token_id, loss_mask_output_only=True):
    """Dense-path collate (data/utils.py:63-93): right-pad input_ids,
    attention_mask, labels (-100 on pad and, with output_only, the prompt)."""
    seqs, labels = [], []
    for ex in examples:
        inp = list(ex["input"]) + list(ex["output"])
        seqs.append(inp)
        lab = ([-100] * len(ex["input"]) if loss_mask_output_only else list(ex["input"])) + list(ex["output"])
        labels.append(lab)
    maxlen = max(len(s) for s in seqs)
    B = len(seqs)
    input_ids = torch.full((B, maxlen), pad_token_id, dtype=torch.long)
    attention_mask = torch.zeros(B, maxlen, dtype=torch.long)
    lab_t = torch.full((B, maxlen), -100, dtype=torch.long)
    for i, (s, l) in enumerate(zip(seqs, labels)):
        input_ids[i, : len(s)] = torch.tensor(s)
        attention_mask[i, : len(s)] = 1
        lab_t[i, : len(l)] = torch.tensor(l)
    return {"input_ids": input_ids, "attention_mask": attention_mask, "labels": lab_t}


class FinetuningIterator:
    """Infinite shuffled iterator over blended datasets with the
    padding-free or padded collate (BlendedDistributedSampler-lite:
    rank-sharded by example index)."""

    def __init__(self, datasets, micro_batch_size, use_padding_free_transformer, loss_mask_output_only, seed, pad_token_id):
        self.datasets = datasets
        self.mbs = micro_batch_size
        self.padding_free = use_padding_free_transformer
        self.loss_mask_output_only = loss_mask_output_only
        self.pad_token_id = pad_token_id
        self.seed = seed
        self.epoch = 0
        self._indices = []
        self._pos = 0

    def _reshuffle(self):
        total = [(d, i) for d in range(len(self.datasets)) for i in range(len(self.datasets[d]))]
        g = torch.Generator().manual_seed(self.seed + self.epoch)
        perm = torch.randperm(len(total), generator=g).tolist()
        world, rank = get_world_size(), get_rank()
        self._indices = [total[p] for p in perm][rank::max(world, 1)]
        self._pos = 0
        self.epoch += 1

    def state_dict(self):
        return {"epoch": self.epoch, "pos": self._pos}

    def load_state_dict(self, sd):
        self.epoch = sd["epoch"] - 1
        self._reshuffle()
        self._pos = sd["pos"]

    def __iter__(self):
        return self

    def __next__(self):
        batch = []
        while len(batch) < self.mbs:
            if self._pos >= len(self._indices):
                self._reshuffle()
            d, i = self._indices[self._pos]
            self._pos += 1
            batch.append(self.datasets[d][i])
        if self.padding_free:
            return collate_padding_free(batch, self.loss_mask_output_only)
        return collate_padded(batch, self.pad_token_id, self.loss_mask_output_only)


def get_finetuning_dataloader(dataset_args, micro_batch_size, use_padding_free_transformer, loss_mask_output_only, seed, pad_token_id):
    datasets = []
    for da in dataset_args:
        cls = _DATASET_CLASSES.get(da.class_name)
        if cls is None:
            raise NotImplementedError(
                f"dataset class {da.class_name} (available offline: {sorted(_DATASET_CLASSES)})"
            )
        datasets.append(cls(**da.class_args))
    assert datasets, "finetuning needs at least one dataset"
    return FinetuningIterator(
        datasets, micro_batch_size, use_padding_free_transformer, loss_mask_output_only, seed, pad_token_id
    )
