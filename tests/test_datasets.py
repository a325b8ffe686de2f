"""Data subsystem tests: packing, instruction datasets, megatron pipeline
(C++ index builders), blending."""

import numpy as np
import pytest
import torch

from automodel_amd.datasets.llm.packed_sequence import (
    PackedDataset, block_causal_mask, greedy_knapsack, thd_collate,
)
from automodel_amd.datasets.llm.instruction import (
    ChatDataset, ColumnMappedTextInstructionDataset,
)
from automodel_amd.datasets.mock import MockDataset


class FakeTokenizer:
    eos_token_id = 0

    def encode(self, text):
        return [ord(c) % 200 + 1 for c in text]


def test_greedy_knapsack_fills_bins():
    bins = greedy_knapsack([10, 20, 30, 40, 5], max_len=50)
    for b in bins:
        assert sum([10, 20, 30, 40, 5][i] for i in b) <= 50
    assert sorted(i for b in bins for i in b) == [0, 1, 2, 3, 4]


def test_packed_dataset_cu_seqlens():
    class VarLen(torch.utils.data.Dataset):
        lens = [10, 14, 6, 30, 3]

        def __len__(self):
            return len(self.lens)

        def __getitem__(self, i):
            L = self.lens[i]
            return {"input_ids": torch.arange(L), "labels": torch.arange(L)}

    ds = PackedDataset(VarLen(), packed_sequence_size=32)
    total = 0
    for i in range(len(ds)):
        s = ds[i]
        assert len(s["input_ids"]) == 32
        cu = s["cu_seqlens"]
        assert cu[0] == 0 and cu[-1] == 32
        assert (cu[1:] >= cu[:-1]).all()
        total += int((s["labels"] != -100).sum())
    assert total == sum(VarLen.lens)

    batch = thd_collate([ds[0], ds[1]] if len(ds) > 1 else [ds[0]])
    assert batch["input_ids"].shape[0] == 1
    assert batch["cu_seqlens"][-1] == batch["input_ids"].shape[1]


def test_block_causal_mask():
    cu = torch.tensor([0, 3, 5], dtype=torch.int32)
    m = block_causal_mask(cu)
    assert m[0, 0] and m[2, 0] and not m[0, 2]
    assert not m[3, 0] and m[4, 3] and not m[3, 4]


def test_instruction_dataset_answer_only_loss():
    tok = FakeTokenizer()
    rows = [{"context": "ctx.", "question": "q?", "answer": "ans"}]
    ds = ColumnMappedTextInstructionDataset(rows, tok)
    s = ds[0]
    n_ignore = int((s["labels"] == -100).sum())
    assert n_ignore > 0 and n_ignore < len(s["labels"])
    assert len(s["input_ids"]) == len(s["labels"])


def test_chat_dataset_assistant_only():
    tok = FakeTokenizer()
    rows = [{"messages": [
        {"role": "user", "content": "hello"},
        {"role": "assistant", "content": "world"},
    ]}]
    ds = ChatDataset(rows, tok)
    s = ds[0]
    assert (s["labels"] == -100).any() and (s["labels"] != -100).any()


def test_megatron_indexed_dataset_roundtrip(tmp_path):
    from automodel_amd.datasets.llm.megatron.indexed_dataset import (
        IndexedDataset, IndexedDatasetWriter,
    )

    prefix = str(tmp_path / "toks")
    w = IndexedDatasetWriter(prefix, dtype=np.int32)
    docs = [np.arange(10), np.arange(100, 105), np.arange(50, 80)]
    for d in docs:
        w.add_document(d)
    w.finalize()
    ds = IndexedDataset(prefix)
    assert len(ds) == 3 and ds.total_tokens == 45
    assert (ds[1] == docs[1]).all()
    assert (ds.get(2, offset=5, length=3) == np.array([55, 56, 57])).all()


def test_megatron_gpt_dataset_windows(tmp_path):
    from automodel_amd.datasets.llm.megatron.gpt_dataset import GPTDataset
    from automodel_amd.datasets.llm.megatron.indexed_dataset import (
        IndexedDataset, IndexedDatasetWriter,
    )

    prefix = str(tmp_path / "toks")
    w = IndexedDatasetWriter(prefix)
    rng = np.random.RandomState(0)
    for _ in range(20):
        w.add_document(rng.randint(0, 1000, size=rng.randint(5, 50)))
    w.finalize()
    ds = GPTDataset(IndexedDataset(prefix), seq_length=16, num_samples=8, seed=7)
    assert len(ds) == 8
    for i in range(len(ds)):
        s = ds[i]
        assert s["input_ids"].shape == (16,) and s["labels"].shape == (16,)
    # deterministic
    a, b = GPTDataset(IndexedDataset(prefix), 16, 8, seed=7)[0], ds[0]
    assert torch.equal(a["input_ids"], b["input_ids"])


def test_megatron_blended_dataset_weights():
    from automodel_amd.datasets.llm.megatron.gpt_dataset import BlendedDataset

    d1 = MockDataset(num_samples=50, seq_len=4, seed=1)
    d2 = MockDataset(num_samples=50, seq_len=4, seed=2)
    bd = BlendedDataset([d1, d2], weights=[0.75, 0.25], size=200)
    counts = np.bincount(bd.dataset_index, minlength=2)
    assert abs(counts[0] / 200 - 0.75) < 0.02
    _ = bd[0], bd[199]


def test_varlen_attention_matches_per_doc():
    from automodel_amd.ops.attention import attention_ref, flash_attention_varlen

    torch.manual_seed(0)
    H, Hk, D = 4, 2, 16
    lens = [10, 25, 7]
    T = sum(lens)
    cu = torch.tensor([0, 10, 35, 42], dtype=torch.int32)
    q = torch.randn(1, T, H, D)
    k = torch.randn(1, T, Hk, D)
    v = torch.randn(1, T, Hk, D)
    out = flash_attention_varlen(q, k, v, cu, backend="sdpa")
    for a, b in zip(cu[:-1].tolist(), cu[1:].tolist()):
        ref = attention_ref(q[:, a:b], k[:, a:b], v[:, a:b], causal=True)
        assert torch.allclose(out[:, a:b], ref, atol=1e-4), (a, b)


def test_packed_recipe_end_to_end(tmp_path):
    """Packed THD batches through the recipe (block-diagonal attention)."""
    from automodel_amd.config.loader import ConfigNode
    from automodel_amd.recipes.llm.train_ft import (
        TrainFinetuneRecipeForNextTokenPrediction,
    )

    cfg = ConfigNode({
        "seed": 0,
        "model": {"config": {
            "vocab_size": 128, "hidden_size": 32, "intermediate_size": 64,
            "num_hidden_layers": 2, "num_attention_heads": 2,
            "num_key_value_heads": 1, "max_position_embeddings": 128,
        }, "dtype": "float32"},
        "loss_fn": {"backend": "chunked", "chunk_size": 64},
        "optimizer": {"lr": 1e-3},
        "step_scheduler": {"grad_acc_steps": 1, "max_steps": 2},
        "dataloader": {
            "dataset": {
                "_target_": "automodel_amd.datasets.llm.packed_sequence.PackedDataset",
                "dataset": {
                    "_target_": "automodel_amd.datasets.mock.MockDataset",
                    "num_samples": 12, "seq_len": 17, "vocab_size": 128,
                },
                "packed_sequence_size": 64,
            },
            "batch_size": 1,
        },
        "output_dir": str(tmp_path / "packed"),
    })
    r = TrainFinetuneRecipeForNextTokenPrediction(cfg)
    r.setup()
    r.run_train_validation_loop()
    assert r.step_scheduler.step == 2
