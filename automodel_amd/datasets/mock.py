"""Synthetic datasets for benchmarking and tests.

Reference behavior: nemo_automodel/components/datasets/llm/mock*.py
(MockIterableDataset powers the published benchmarks — BASELINE.md notes the
reference's numbers are measured on mock data).
"""

from __future__ import annotations

import torch
from torch.utils.data import Dataset, IterableDataset


class MockIterableDataset(IterableDataset):
    """Infinite stream of random token batches with labels = shifted inputs."""

    def __init__(self, seq_len: int = 4096, vocab_size: int = 128256, seed: int = 1234,
                 num_samples: int | None = None):
        self.seq_len = seq_len
        self.vocab_size = vocab_size
        self.seed = seed
        self.num_samples = num_samples

    def __iter__(self):
        g = torch.Generator().manual_seed(self.seed)
        n = 0
        while self.num_samples is None or n < self.num_samples:
            ids = torch.randint(0, self.vocab_size, (self.seq_len + 1,), generator=g)
            yield {"input_ids": ids[:-1], "labels": ids[1:].clone()}
            n += 1


class MockClassificationDataset(Dataset):
    """Sequence-classification samples: tokens + one class id."""

    def __init__(self, num_samples: int = 64, seq_len: int = 32, vocab_size: int = 1000,
                 num_labels: int = 2, seed: int = 0):
        self.num_samples = num_samples
        self.seq_len = seq_len
        self.vocab_size = vocab_size
        self.num_labels = num_labels
        self.seed = seed

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx: int):
        g = torch.Generator().manual_seed(self.seed + idx)
        return {
            "input_ids": torch.randint(0, self.vocab_size, (self.seq_len,), generator=g),
            "labels": torch.randint(0, self.num_labels, (1,), generator=g),
        }


class MockDataset(Dataset):
    """Finite map-style mock dataset (deterministic per index)."""

    def __init__(self, num_samples: int = 128, seq_len: int = 512, vocab_size: int = 32000,
                 seed: int = 1234):
        self.num_samples = num_samples
        self.seq_len = seq_len
        self.vocab_size = vocab_size
        self.seed = seed

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx: int):
        g = torch.Generator().manual_seed(self.seed + idx)
        ids = torch.randint(0, self.vocab_size, (self.seq_len + 1,), generator=g)
        return {"input_ids": ids[:-1], "labels": ids[1:].clone()}
