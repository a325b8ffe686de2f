"""Tool-call parsing and generation-based accuracy evaluation for agent SFT.

Reference behavior: nemo_automodel/components/eval/tool_call_parser.py
(wrapper-specific parsers tried most-specific-first, positional pred↔gt
alignment, six [0,1] metrics) and eval/tool_call_evaluator.py:49-423
(ToolCallAccuracyEvaluator: renders prompts through the chat template,
generates greedily, parses and aggregates; rank-sharded sampling with the
caller all-reducing counts). The implementation here is independent: a
bracket-balanced JSON scanner shared by all dialect parsers, and the
evaluator rides automodel_amd.utils.generation.generate (KV-less greedy).

Supported wrapper dialects (all public chat-template formats):
  * qwen:    <tool_call>{...}</tool_call>
  * mistral: [TOOL_CALLS] [{...}, ...]
  * llama:   bare {"name": ..., "parameters"/"arguments": {...}} JSON
  * harmony: <|channel|>commentary to=functions.NAME ... <|message|>{args}
"""

from __future__ import annotations

import json
import re
from dataclasses import dataclass, field
from typing import Any, Iterator

import torch


@dataclass
class ToolCall:
    name: str | None
    arguments: dict = field(default_factory=dict)
    valid_json: bool = False
    raw: str = ""


def _scan_json_objects(text: str) -> Iterator[str]:
    """Yield every top-level balanced {...} span (string-literal aware)."""
    depth = 0
    start = -1
    in_str = False
    esc = False
    for i, ch in enumerate(text):
        if in_str:
            if esc:
                esc = False
            elif ch == "\\":
                esc = True
            elif ch == '"':
                in_str = False
            continue
        if ch == '"':
            in_str = True
        elif ch == "{":
            if depth == 0:
                start = i
            depth += 1
        elif ch == "}":
            if depth > 0:
                depth -= 1
                if depth == 0:
                    yield text[start : i + 1]
    return


def _args_to_dict(val: Any) -> tuple[dict, bool]:
    if isinstance(val, dict):
        return val, True
    if isinstance(val, str):
        try:
            d = json.loads(val)
        except json.JSONDecodeError:
            return {}, False
        return (d, True) if isinstance(d, dict) else ({}, False)
    return {}, val is None


def _call_from_obj(obj: Any, raw: str) -> ToolCall | None:
    if not isinstance(obj, dict):
        return None
    if "function" in obj and isinstance(obj["function"], dict):
        obj = obj["function"]  # OpenAI-style {"type":"function","function":{...}}
    name = obj.get("name")
    if name is None:
        return None
    args_val = obj.get("arguments", obj.get("parameters", {}))
    args, ok = _args_to_dict(args_val)
    return ToolCall(name=name, arguments=args, valid_json=ok, raw=raw)


def _calls_from_spans(spans: Iterator[str] | list[str]) -> list[ToolCall]:
    out = []
    for span in spans:
        try:
            obj = json.loads(span)
        except json.JSONDecodeError:
            continue
        call = _call_from_obj(obj, span)
        if call is not None:
            out.append(call)
    return out


def _parse_qwen(text: str) -> list[ToolCall]:
    spans = re.findall(r"<tool_call>\s*(.*?)\s*</tool_call>", text, re.DOTALL)
    return _calls_from_spans(spans)


def _parse_mistral(text: str) -> list[ToolCall]:
    m = re.search(r"\[TOOL_CALLS\]\s*", text)
    if not m:
        return []
    return _calls_from_spans(_scan_json_objects(text[m.end():]))


def _parse_harmony(text: str) -> list[ToolCall]:
    out = []
    for m in re.finditer(
        r"to=(?:functions\.)?([\w.\-]+).*?<\|message\|>(.*?)(?:<\|call\|>|<\|end\|>|$)",
        text, re.DOTALL,
    ):
        name, body = m.group(1), m.group(2).strip()
        args, ok = ({}, False)
        for span in _scan_json_objects(body):
            args, ok = _args_to_dict(span)
            break
        out.append(ToolCall(name=name, arguments=args, valid_json=ok, raw=m.group(0)))
    return out


def _parse_generic(text: str) -> list[ToolCall]:
    return _calls_from_spans(_scan_json_objects(text))


def parse_tool_calls(text: str) -> list[ToolCall]:
    """Parse every tool call from decoded model output; dialect wrappers are
    tried most-specific-first, generic JSON scan as the fallback."""
    if not text:
        return []
    for parser in (_parse_harmony, _parse_qwen, _parse_mistral):
        calls = parser(text)
        if calls:
            return calls
    return _parse_generic(text)


METRIC_KEYS = (
    "has_call",
    "name_correct",
    "args_json_valid",
    "args_field_recall",
    "args_field_precision",
    "args_exact_match",
)


def score_tool_calls(pred: list[ToolCall], gt: list[dict]) -> dict[str, float]:
    """Positional alignment: pred[i] vs gt[i]; missing preds score zero at
    their position (a model emitting one of two parallel calls is penalized);
    extra preds are ignored. All values in [0,1]."""
    if not gt:
        return {k: 0.0 for k in METRIC_KEYS}
    sums = dict.fromkeys(METRIC_KEYS, 0.0)
    for i, g in enumerate(gt):
        p = pred[i] if i < len(pred) else None
        if p is None:
            continue
        g_args, _ = _args_to_dict(g.get("arguments", {}))
        sums["has_call"] += 1.0
        sums["name_correct"] += float(p.name == g.get("name"))
        sums["args_json_valid"] += float(p.valid_json)
        pk = set(p.arguments) if p.valid_json else set()
        gk = set(g_args)
        sums["args_field_recall"] += (len(pk & gk) / len(gk)) if gk else float(not pk)
        sums["args_field_precision"] += (len(pk & gk) / len(pk)) if pk else float(not gk)
        sums["args_exact_match"] += float(p.valid_json and p.arguments == g_args)
    n = len(gt)
    return {k: v / n for k, v in sums.items()}


class ToolCallEvaluator:
    """Generation-based tool-call accuracy over a JSONL sample file.

    Each sample: {"prompt": str | "prompt_ids": [int], "gt_tool_calls":
    [{"name":..., "arguments": {...}}]}. With a tokenizer, "prompt" is
    encoded (chat templating is the dataset-prep step's job — samples are
    stored already rendered). sample_shard=(rank, world) strides samples so
    ranks split the work; the caller all-reduces (sum) the returned metric
    sums and _count.
    """

    def __init__(self, path: str | None = None, samples: list[dict] | None = None,
                 max_new_tokens: int = 64, sample_shard: tuple[int, int] | None = None,
                 metric_prefix: str = "tool_call"):
        assert (path is None) != (samples is None), "give path XOR samples"
        if path is not None:
            samples = []
            with open(path) as f:
                for line in f:
                    line = line.strip()
                    if line:
                        samples.append(json.loads(line))
        self.samples = samples
        self.max_new_tokens = max_new_tokens
        self.sample_shard = sample_shard
        self.metric_prefix = metric_prefix

    def _my_samples(self) -> list[dict]:
        if self.sample_shard is None:
            return self.samples
        rank, world = self.sample_shard
        return self.samples[rank::world]

    @torch.no_grad()
    def evaluate(self, model, tokenizer=None) -> dict[str, float]:
        from automodel_amd.utils.generation import generate

        try:
            device = next(model.parameters()).device
        except StopIteration:
            device = torch.device("cpu")
        sums = dict.fromkeys(METRIC_KEYS, 0.0)
        n = 0
        for sample in self._my_samples():
            if "prompt_ids" in sample:
                ids = sample["prompt_ids"]
            else:
                assert tokenizer is not None, "tokenizer needed for text prompts"
                ids = tokenizer.encode(sample["prompt"])
            input_ids = torch.tensor([ids], dtype=torch.long, device=device)
            eos = getattr(tokenizer, "eos_token_id", None) if tokenizer else None
            out = generate(model, input_ids, max_new_tokens=self.max_new_tokens,
                           eos_token_id=eos)
            new = out[0, input_ids.shape[1]:].tolist()
            text = tokenizer.decode(new) if tokenizer else sample.get("decode_fn", str)(new)
            metrics = score_tool_calls(parse_tool_calls(text), sample["gt_tool_calls"])
            for k in METRIC_KEYS:
                sums[k] += metrics[k]
            n += 1
        out_metrics = {f"{self.metric_prefix}/{k}": (sums[k] / n if n else 0.0)
                       for k in METRIC_KEYS}
        out_metrics[f"{self.metric_prefix}/_count"] = float(n)
        return out_metrics
