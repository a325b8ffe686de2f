"""Mock VLM dataset + collator (reference datasets/vlm/mock.py, collate_fns.py)."""

from __future__ import annotations

import torch
from torch.utils.data import Dataset

IGNORE_INDEX = -100


class MockVLMDataset(Dataset):
    """Samples with one image each: [text..., <image tokens>, text...]."""

    def __init__(self, num_samples: int = 16, seq_len: int = 128, vocab_size: int = 1024,
                 image_size: int = 224, patch_size: int = 14, image_token_id: int = 151655,
                 num_patches: int | None = None, seed: int = 0):
        self.num_samples = num_samples
        self.seq_len = seq_len
        self.vocab_size = vocab_size
        self.image_size = image_size
        self.image_token_id = image_token_id
        self.num_patches = num_patches or (image_size // patch_size) ** 2
        assert self.num_patches < seq_len, "seq too short for image patches"
        self.seed = seed

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx: int) -> dict:
        g = torch.Generator().manual_seed(self.seed + idx)
        n_text = self.seq_len - self.num_patches
        pre = torch.randint(0, min(self.vocab_size, 10000), (n_text // 2,), generator=g)
        post = torch.randint(0, min(self.vocab_size, 10000), (n_text - n_text // 2,), generator=g)
        img_tokens = torch.full((self.num_patches,), self.image_token_id, dtype=torch.long)
        ids = torch.cat([pre, img_tokens, post])
        labels = ids.roll(-1)
        labels[ids == self.image_token_id] = IGNORE_INDEX
        labels[-1] = IGNORE_INDEX
        pixels = torch.randn(3, self.image_size, self.image_size, generator=g)
        return {"input_ids": ids, "labels": labels, "pixel_values": pixels}


def vlm_collate(batch: list[dict]) -> dict:
    return {
        "input_ids": torch.stack([b["input_ids"] for b in batch]),
        "labels": torch.stack([b["labels"] for b in batch]),
        "pixel_values": torch.stack([b["pixel_values"] for b in batch]),
    }
