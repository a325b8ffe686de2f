"""Sequence packing: greedy-knapsack offline packing -> THD cu_seqlens batches.

Reference behavior: nemo_automodel/components/datasets/llm/packed_sequence.py
:268 (pack_dataset: greedy knapsack into bins <= packed_sequence_size,
cu_seqlens metadata for varlen attention) and datasets/loader.py:193
(ThdPackingConfig).
"""

from __future__ import annotations

import torch
from torch.utils.data import Dataset

IGNORE_INDEX = -100


def greedy_knapsack(lengths: list[int], max_len: int) -> list[list[int]]:
    """Pack sample indices into bins of total length <= max_len.

    Sorted-descending first-fit (the reference's strategy): near-optimal fill
    with deterministic output.
    """
    order = sorted(range(len(lengths)), key=lambda i: -lengths[i])
    bins: list[list[int]] = []
    bin_space: list[int] = []
    for i in order:
        L = lengths[i]
        if L > max_len:
            continue  # drop over-long samples (reference behavior: truncate/drop)
        placed = False
        for b in range(len(bins)):
            if bin_space[b] >= L:
                bins[b].append(i)
                bin_space[b] -= L
                placed = True
                break
        if not placed:
            bins.append([i])
            bin_space.append(max_len - L)
    return bins


class PackedDataset(Dataset):
    """Offline-packs a map-style dataset of {input_ids, labels} samples into
    fixed bins with cu_seqlens (THD layout)."""

    def __init__(self, dataset: Dataset, packed_sequence_size: int,
                 pad_to_size: bool = True, pad_token_id: int = 0):
        self.dataset = dataset
        self.size = packed_sequence_size
        self.pad_to_size = pad_to_size
        self.pad_token_id = pad_token_id
        lengths = [len(dataset[i]["input_ids"]) for i in range(len(dataset))]
        self.bins = greedy_knapsack(lengths, packed_sequence_size)

    def __len__(self):
        return len(self.bins)

    def __getitem__(self, idx: int) -> dict:
        ids_parts, label_parts, cu = [], [], [0]
        for i in self.bins[idx]:
            s = self.dataset[i]
            ids_parts.append(torch.as_tensor(s["input_ids"]))
            label_parts.append(torch.as_tensor(s["labels"]))
            cu.append(cu[-1] + len(s["input_ids"]))
        input_ids = torch.cat(ids_parts)
        labels = torch.cat(label_parts)
        if self.pad_to_size and len(input_ids) < self.size:
            pad = self.size - len(input_ids)
            input_ids = torch.nn.functional.pad(input_ids, (0, pad), value=self.pad_token_id)
            labels = torch.nn.functional.pad(labels, (0, pad), value=IGNORE_INDEX)
            cu.append(self.size)
        return {
            "input_ids": input_ids,
            "labels": labels,
            "cu_seqlens": torch.tensor(cu, dtype=torch.int32),
        }


def thd_collate(batch: list[dict]) -> dict:
    """Collate packed samples into one THD batch: concatenate along tokens,
    merge cu_seqlens (reference distributed/thd_utils.py:85)."""
    ids = torch.cat([b["input_ids"] for b in batch]).unsqueeze(0)
    labels = torch.cat([b["labels"] for b in batch]).unsqueeze(0)
    cu, offset = [torch.tensor([0], dtype=torch.int32)], 0
    for b in batch:
        cu.append(b["cu_seqlens"][1:] + offset)
        offset += len(b["input_ids"])
    return {"input_ids": ids, "labels": labels, "cu_seqlens": torch.cat(cu)}


def block_causal_mask(cu_seqlens: torch.Tensor, total_len: int | None = None) -> torch.Tensor:
    """Dense [T, T] block-causal mask for packed docs (reference
    packed_sequence.py:396-455) — used by the sdpa/eager paths; the HIP
    varlen kernel consumes cu_seqlens directly."""
    T = total_len or int(cu_seqlens[-1])
    mask = torch.zeros(T, T, dtype=torch.bool)
    for a, b in zip(cu_seqlens[:-1].tolist(), cu_seqlens[1:].tolist()):
        mask[a:b, a:b] = torch.ones(b - a, b - a, dtype=torch.bool).tril()
    return mask
