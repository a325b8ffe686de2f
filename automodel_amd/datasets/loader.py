"""Dataloader construction with DP sharding and padding collation.

Reference behavior: nemo_automodel/components/datasets/loader.py:572
(DataloaderConfig builds a StatefulDataLoader with DP-rank sharding and
collators). This implementation keeps a light Stateful wrapper so dataloader
position participates in checkpoint resume.
"""

from __future__ import annotations

from typing import Any, Iterator

import torch
from torch.utils.data import DataLoader, Dataset, DistributedSampler, IterableDataset

IGNORE_INDEX = -100


def padded_collate(batch: list[dict], pad_token_id: int = 0) -> dict:
    """Pad variable-length samples to the batch max; labels padded with -100.
    Packed samples (with cu_seqlens) collate into one THD batch instead."""
    if "cu_seqlens" in batch[0]:
        from automodel_amd.datasets.llm.packed_sequence import thd_collate

        return thd_collate(batch)
    keys = batch[0].keys()
    out = {}
    max_len = max(len(b["input_ids"]) for b in batch)
    for key in keys:
        if key in ("input_ids", "labels", "attention_mask", "position_ids"):
            pad_val = IGNORE_INDEX if key == "labels" else pad_token_id
            rows = []
            for b in batch:
                t = torch.as_tensor(b[key])
                if len(t) < max_len:
                    t = torch.nn.functional.pad(t, (0, max_len - len(t)), value=pad_val)
                rows.append(t)
            out[key] = torch.stack(rows)
        else:
            out[key] = [b[key] for b in batch]
    return out


class StatefulLoader:
    """Wraps a DataLoader; tracks batches yielded for checkpoint resume."""

    def __init__(self, loader: DataLoader, sampler: DistributedSampler | None = None):
        self.loader = loader
        self.sampler = sampler
        self.batches_yielded = 0
        self.epoch = 0

    def set_epoch(self, epoch: int) -> None:
        self.epoch = epoch
        if self.sampler is not None:
            self.sampler.set_epoch(epoch)

    def __iter__(self) -> Iterator:
        skip = self.batches_yielded
        for i, batch in enumerate(self.loader):
            if i < skip:
                continue
            self.batches_yielded = i + 1
            yield batch
        self.batches_yielded = 0

    def __len__(self):
        return len(self.loader)

    def state_dict(self) -> dict:
        return {"batches_yielded": self.batches_yielded, "epoch": self.epoch}

    def load_state_dict(self, state: dict) -> None:
        self.batches_yielded = state["batches_yielded"]
        self.epoch = state["epoch"]


def build_dataloader(
    dataset: Dataset | IterableDataset,
    batch_size: int = 1,
    shuffle: bool = True,
    num_workers: int = 0,
    pad_token_id: int = 0,
    dp_rank: int = 0,
    dp_world: int = 1,
    drop_last: bool = True,
    seed: int = 42,
) -> StatefulLoader:
    collate = lambda b: padded_collate(b, pad_token_id)  # noqa: E731
    if isinstance(dataset, IterableDataset):
        # iterable datasets shard by skipping (rank r takes every dp_world-th)
        class _Sharded(IterableDataset):
            def __init__(self, ds, rank, world):
                self.ds, self.rank, self.world = ds, rank, world

            def __iter__(self):
                for i, x in enumerate(self.ds):
                    if i % self.world == self.rank:
                        yield x

        ds = _Sharded(dataset, dp_rank, dp_world) if dp_world > 1 else dataset
        loader = DataLoader(ds, batch_size=batch_size, num_workers=num_workers,
                            collate_fn=collate, drop_last=drop_last)
        return StatefulLoader(loader)
    sampler = None
    if dp_world > 1:
        sampler = DistributedSampler(dataset, num_replicas=dp_world, rank=dp_rank,
                                     shuffle=shuffle, seed=seed, drop_last=drop_last)
    loader = DataLoader(
        dataset, batch_size=batch_size, sampler=sampler,
        shuffle=(shuffle and sampler is None), num_workers=num_workers,
        collate_fn=collate, drop_last=drop_last,
    )
    return StatefulLoader(loader, sampler)
