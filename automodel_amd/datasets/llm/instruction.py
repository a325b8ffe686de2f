"""Instruction / chat SFT datasets from local files (no hub access needed).

Reference behavior: nemo_automodel/components/datasets/llm/ — SQuAD,
column-mapped instruction datasets, chat datasets with template application
and answer-only loss masking.
"""

from __future__ import annotations

import json
import os
from typing import Any, Callable

import torch
from torch.utils.data import Dataset

IGNORE_INDEX = -100


class ColumnMappedTextInstructionDataset(Dataset):
    """Rows from json/jsonl mapped by column names to (context, question,
    answer); loss computed on the answer tokens only (answer_only_loss)."""

    def __init__(
        self,
        path_or_rows: str | list[dict],
        tokenizer: Any,
        column_mapping: dict[str, str] | None = None,
        answer_only_loss: bool = True,
        max_length: int = 2048,
        prompt_template: str = "{context}{question} ",
    ):
        if isinstance(path_or_rows, str):
            self.rows = _load_rows(path_or_rows)
        else:
            self.rows = list(path_or_rows)
        self.tok = tokenizer
        self.mapping = column_mapping or {"context": "context", "question": "question",
                                          "answer": "answer"}
        self.answer_only_loss = answer_only_loss
        self.max_length = max_length
        self.prompt_template = prompt_template

    def __len__(self):
        return len(self.rows)

    def __getitem__(self, idx: int) -> dict:
        row = self.rows[idx]
        ctx = row.get(self.mapping.get("context", ""), "")
        q = row.get(self.mapping.get("question", ""), "")
        a = row.get(self.mapping.get("answer", ""), "")
        prompt = self.prompt_template.format(context=ctx, question=q)
        prompt_ids = self.tok.encode(prompt)
        answer_ids = self.tok.encode(str(a))
        eos = getattr(self.tok, "eos_token_id", None)
        if eos is not None:
            answer_ids = answer_ids + [eos]
        ids = (prompt_ids + answer_ids)[: self.max_length + 1]
        input_ids = torch.tensor(ids[:-1], dtype=torch.long)
        labels = torch.tensor(ids[1:], dtype=torch.long)
        if self.answer_only_loss:
            n_prompt = max(0, min(len(prompt_ids) - 1, len(labels)))
            labels[:n_prompt] = IGNORE_INDEX
        return {"input_ids": input_ids, "labels": labels}


class SquadDataset(ColumnMappedTextInstructionDataset):
    """SQuAD-format local json (reference datasets/llm/squad.py)."""

    def __init__(self, path: str, tokenizer: Any, **kw):
        rows = []
        with open(path) as f:
            data = json.load(f)
        for article in data.get("data", []):
            for para in article.get("paragraphs", []):
                for qa in para.get("qas", []):
                    if qa.get("answers"):
                        rows.append({
                            "context": para.get("context", ""),
                            "question": qa.get("question", ""),
                            "answer": qa["answers"][0]["text"],
                        })
        super().__init__(rows, tokenizer,
                         prompt_template="Context: {context} Question: {question} Answer: ",
                         **kw)


class ChatDataset(Dataset):
    """messages-format chat rows; loss on assistant turns only
    (reference datasets/llm/chat_dataset.py)."""

    def __init__(self, path_or_rows: str | list[dict], tokenizer: Any,
                 max_length: int = 2048):
        self.rows = _load_rows(path_or_rows) if isinstance(path_or_rows, str) else list(path_or_rows)
        self.tok = tokenizer
        self.max_length = max_length

    def __len__(self):
        return len(self.rows)

    def __getitem__(self, idx: int) -> dict:
        msgs = self.rows[idx]["messages"]
        ids: list[int] = []
        mask: list[bool] = []    # True where loss applies
        for m in msgs:
            turn = f"<|{m['role']}|>{m['content']}"
            turn_ids = self.tok.encode(turn)
            eos = getattr(self.tok, "eos_token_id", None)
            if eos is not None:
                turn_ids = turn_ids + [eos]
            ids.extend(turn_ids)
            mask.extend([m["role"] == "assistant"] * len(turn_ids))
        ids = ids[: self.max_length + 1]
        mask = mask[: self.max_length + 1]
        input_ids = torch.tensor(ids[:-1], dtype=torch.long)
        labels = torch.tensor(ids[1:], dtype=torch.long)
        lm = torch.tensor(mask[1:], dtype=torch.bool)
        labels[~lm] = IGNORE_INDEX
        return {"input_ids": input_ids, "labels": labels}


def _load_rows(path: str) -> list[dict]:
    if path.endswith(".jsonl"):
        with open(path) as f:
            return [json.loads(line) for line in f if line.strip()]
    with open(path) as f:
        data = json.load(f)
    if isinstance(data, dict) and "rows" in data:
        return data["rows"]
    return data
