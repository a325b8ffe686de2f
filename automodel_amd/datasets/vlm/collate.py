"""VLM SFT collators: assistant-span label masking + mixed-media batching.

Reference behavior: nemo_automodel/components/datasets/vlm/collate_fns.py
— ``build_labels`` (:138) re-tokenizes each assistant message and pattern-
matches it into the encoded conversation (with the lstrip retry for BPE
leading-whitespace drift and stop-token absorption), and
``build_labels_from_markers`` (:336) scans for the chat template's fixed
assistant-marker token ids instead (robust to BPE context sensitivity).
``vlm_chat_collate`` mirrors the per-family collators' shared tail: right-
pad ids/labels, stack same-shaped pixel tensors, concat patchified ones,
and carry grid metadata (image_grid_thw) through.
"""

from __future__ import annotations

from typing import Any, Sequence

import torch


def find_pattern_indices(encoded: torch.Tensor, pattern: torch.Tensor,
                         search_start: int = 0,
                         allow_first_token_mismatch: bool = False
                         ) -> tuple[int, int]:
    """First (start, end) of ``pattern`` in ``encoded`` at/after
    ``search_start``; (-1, -1) if absent. ``allow_first_token_mismatch``
    tolerates BPE drift on the leading token."""
    n, m = len(encoded), len(pattern)
    for s in range(search_start, n - m + 1):
        window = encoded[s:s + m]
        if allow_first_token_mismatch:
            if torch.equal(window[1:], pattern[1:]):
                return s, s + m
        elif torch.equal(window, pattern):
            return s, s + m
    return -1, -1


def build_labels(input_ids: torch.Tensor,
                 conversations: Sequence[Sequence[dict[str, Any]]],
                 tokenizer, stop_tokens: Sequence[str] = ("</s>",)
                 ) -> torch.Tensor:
    """[B, S] labels: -100 everywhere except assistant responses (+ the
    immediately following stop token)."""

    def _text(message):
        c = message.get("content", "")
        if isinstance(c, str):
            return c
        return "".join(p.get("text", "") for p in c
                       if isinstance(p, dict) and p.get("type") == "text")

    out = []
    for encoded, conv in zip(input_ids, conversations):
        labels = torch.full_like(encoded, -100)
        start = 0
        for message in conv:
            if message.get("role") != "assistant":
                continue
            text = _text(message)
            if not text:
                continue
            toks = torch.as_tensor(tokenizer.encode(text,
                                                    add_special_tokens=False),
                                   device=encoded.device)
            a, b = find_pattern_indices(encoded, toks, start)
            if a < 0 and text != text.lstrip():
                toks = torch.as_tensor(
                    tokenizer.encode(text.lstrip(), add_special_tokens=False),
                    device=encoded.device)
                a, b = find_pattern_indices(encoded, toks, start)
            if a < 0:
                break
            if b < len(encoded):
                nxt = tokenizer.decode([int(encoded[b])])
                if nxt.strip() in stop_tokens:
                    b += 1
            labels[a:b] = encoded[a:b]
            start = b
        out.append(labels)
    return torch.stack(out)


def build_labels_from_markers(input_ids: torch.Tensor,
                              assistant_marker: Sequence[int],
                              stop_token_id: int) -> torch.Tensor:
    """Marker-scan label builder: everything from just after each assistant
    marker through (incl.) the next stop token is supervised."""
    marker = torch.as_tensor(list(assistant_marker))
    out = []
    for encoded in input_ids:
        labels = torch.full_like(encoded, -100)
        pos = 0
        while True:
            a, b = find_pattern_indices(encoded, marker.to(encoded.device), pos)
            if a < 0:
                break
            stops = (encoded[b:] == stop_token_id).nonzero()
            end = b + int(stops[0]) + 1 if len(stops) else len(encoded)
            labels[b:end] = encoded[b:end]
            pos = end
        out.append(labels)
    return torch.stack(out)


def vlm_chat_collate(examples: list[dict], pad_id: int = 0,
                     label_pad: int = -100) -> dict:
    """Right-pad input_ids/labels; stack same-shape pixel_values, concat
    ragged/patchified ones along dim 0; carry image_grid_thw."""
    S = max(len(e["input_ids"]) for e in examples)
    ids, labels, attn = [], [], []
    for e in examples:
        t = torch.as_tensor(e["input_ids"])
        l = torch.as_tensor(e["labels"]) if "labels" in e else t.clone()
        pad = S - len(t)
        ids.append(torch.cat([t, t.new_full((pad,), pad_id)]))
        labels.append(torch.cat([l, l.new_full((pad,), label_pad)]))
        attn.append(torch.cat([torch.ones(len(t), dtype=torch.long),
                               torch.zeros(pad, dtype=torch.long)]))
    batch = {"input_ids": torch.stack(ids), "labels": torch.stack(labels),
             "attention_mask": torch.stack(attn)}
    pvs = [torch.as_tensor(e["pixel_values"]) for e in examples
           if e.get("pixel_values") is not None]
    if pvs:
        same = all(p.shape == pvs[0].shape for p in pvs)
        batch["pixel_values"] = (torch.stack(pvs) if same and pvs[0].dim() == 3
                                 else torch.cat(pvs, dim=0))
    grids = [torch.as_tensor(e["image_grid_thw"]) for e in examples
             if e.get("image_grid_thw") is not None]
    if grids:
        batch["image_grid_thw"] = torch.cat(
            [g if g.dim() == 2 else g.unsqueeze(0) for g in grids], dim=0)
    return batch
