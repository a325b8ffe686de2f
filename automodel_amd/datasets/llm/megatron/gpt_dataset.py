"""GPT pretraining dataset over indexed token streams + blended datasets.

Reference behavior: nemo_automodel/components/datasets/llm/megatron/
gpt_dataset.py + builder.py (doc/sample/shuffle indices via the C++ helpers,
weighted blending across datasets).
"""

from __future__ import annotations

import numpy as np
import torch
from torch.utils.data import Dataset

from automodel_amd.datasets.llm.megatron.build import load_helpers
from automodel_amd.datasets.llm.megatron.indexed_dataset import IndexedDataset


class GPTDataset(Dataset):
    """Tiles the (epoch-replicated, shuffled) document stream into fixed
    seq_length+1 token windows."""

    def __init__(self, indexed: IndexedDataset, seq_length: int,
                 num_samples: int | None = None, seed: int = 1234):
        self.ds = indexed
        self.seq_length = seq_length
        helpers = load_helpers()

        tokens_per_epoch = self.ds.total_tokens
        samples_per_epoch = max(1, (tokens_per_epoch - 1) // seq_length)
        num_samples = num_samples or samples_per_epoch
        num_epochs = max(1, -(-num_samples // samples_per_epoch))

        rng = np.random.RandomState(seed)
        doc_idx = np.tile(np.arange(len(self.ds), dtype=np.int32), num_epochs)
        rng.shuffle(doc_idx)
        self.doc_idx = doc_idx
        self.sample_idx = helpers.build_sample_idx(
            np.asarray(self.ds.sizes, dtype=np.int32), doc_idx,
            seq_length, num_epochs, tokens_per_epoch,
        )
        n_avail = self.sample_idx.shape[0] - 1
        self.shuffle_idx = helpers.build_shuffle_idx(n_avail, seed + 1)
        self.num_samples = min(num_samples, n_avail)

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx: int) -> dict:
        i = int(self.shuffle_idx[idx % len(self.shuffle_idx)])
        doc_a, off_a = self.sample_idx[i]
        doc_b, off_b = self.sample_idx[i + 1]
        want = self.seq_length + 1
        parts = []
        if doc_a == doc_b:
            parts.append(self.ds.get(self.doc_idx[doc_a], off_a, want))
        else:
            parts.append(self.ds.get(self.doc_idx[doc_a], off_a))
            for d in range(doc_a + 1, doc_b):
                parts.append(self.ds.get(self.doc_idx[d]))
            parts.append(self.ds.get(self.doc_idx[doc_b], 0, off_b + 1))
        tokens = np.concatenate(parts)[:want]
        if len(tokens) < want:  # stream end: pad by wrapping
            tokens = np.concatenate([tokens, np.zeros(want - len(tokens), dtype=tokens.dtype)])
        t = torch.from_numpy(np.ascontiguousarray(tokens)).long()
        return {"input_ids": t[:-1], "labels": t[1:].clone()}


class BlendedDataset(Dataset):
    """Weighted mixture of datasets via the C++ blending-index builder
    (reference builder.py + helpers.cpp build_blending_indices)."""

    def __init__(self, datasets: list[Dataset], weights: list[float], size: int):
        assert len(datasets) == len(weights) and datasets
        self.datasets = datasets
        w = np.asarray(weights, dtype=np.float64)
        w = w / w.sum()
        helpers = load_helpers()
        self.dataset_index = np.zeros(size, dtype=np.int16)
        self.dataset_sample_index = np.zeros(size, dtype=np.int64)
        helpers.build_blending_indices(self.dataset_index, self.dataset_sample_index,
                                       w, len(datasets), size)
        self.size = size

    def __len__(self):
        return self.size

    def __getitem__(self, idx: int):
        d = int(self.dataset_index[idx])
        s = int(self.dataset_sample_index[idx]) % len(self.datasets[d])
        return self.datasets[d][s]
