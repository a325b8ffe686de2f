"""Tool-call parser dialects, positional scoring, and the evaluator loop."""

import json

import torch
import torch.nn as nn

from automodel_amd.eval.tool_calling import (
    ToolCallEvaluator,
    parse_tool_calls,
    score_tool_calls,
)


def test_parse_qwen_style():
    text = ('thinking...\n<tool_call>\n{"name": "get_weather", '
            '"arguments": {"city": "SF"}}\n</tool_call>')
    calls = parse_tool_calls(text)
    assert len(calls) == 1
    assert calls[0].name == "get_weather"
    assert calls[0].arguments == {"city": "SF"} and calls[0].valid_json


def test_parse_mistral_style_multiple():
    text = '[TOOL_CALLS] [{"name": "a", "arguments": {}}, {"name": "b", "arguments": {"x": 1}}]'
    calls = parse_tool_calls(text)
    assert [c.name for c in calls] == ["a", "b"]
    assert calls[1].arguments == {"x": 1}


def test_parse_harmony_style():
    text = ('<|channel|>commentary to=functions.search <|constrain|>json'
            '<|message|>{"query": "rocm"}<|call|>')
    calls = parse_tool_calls(text)
    assert len(calls) == 1 and calls[0].name == "search"
    assert calls[0].arguments == {"query": "rocm"}


def test_parse_generic_and_openai_nesting():
    text = 'Sure: {"type": "function", "function": {"name": "f", "arguments": "{\\"k\\": 2}"}}'
    calls = parse_tool_calls(text)
    assert calls[0].name == "f" and calls[0].arguments == {"k": 2}


def test_parse_ignores_nested_braces_in_strings():
    text = '{"name": "f", "arguments": {"s": "has { brace"}}'
    calls = parse_tool_calls(text)
    assert len(calls) == 1 and calls[0].arguments == {"s": "has { brace"}


def test_parse_empty():
    assert parse_tool_calls("") == []
    assert parse_tool_calls("no calls here") == []


def test_score_exact_and_partial():
    pred = parse_tool_calls('<tool_call>{"name": "f", "arguments": {"a": 1, "b": 2}}</tool_call>')
    gt = [{"name": "f", "arguments": {"a": 1, "b": 2}}]
    m = score_tool_calls(pred, gt)
    assert m["name_correct"] == 1.0 and m["args_exact_match"] == 1.0

    gt2 = [{"name": "f", "arguments": {"a": 1, "c": 3}}]
    m2 = score_tool_calls(pred, gt2)
    assert m2["args_field_recall"] == 0.5 and m2["args_field_precision"] == 0.5
    assert m2["args_exact_match"] == 0.0


def test_score_missing_parallel_call_penalized():
    pred = parse_tool_calls('<tool_call>{"name": "f", "arguments": {}}</tool_call>')
    gt = [{"name": "f", "arguments": {}}, {"name": "g", "arguments": {}}]
    m = score_tool_calls(pred, gt)
    assert m["has_call"] == 0.5 and m["name_correct"] == 0.5


def test_score_string_gt_arguments():
    pred = parse_tool_calls('{"name": "f", "arguments": {"x": 1}}')
    m = score_tool_calls(pred, [{"name": "f", "arguments": '{"x": 1}'}])
    assert m["args_exact_match"] == 1.0


class _EchoLM(nn.Module):
    """Tiny LM that deterministically emits a fixed token sequence."""

    def __init__(self, script):
        super().__init__()
        self.script = script
        self.dummy = nn.Parameter(torch.zeros(1))

    def forward(self, ids):
        t = ids.shape[1]
        logits = torch.zeros(ids.shape[0], t, 300)
        step = t - self.prompt_len
        nxt = self.script[min(step, len(self.script) - 1)]
        logits[:, -1, nxt] = 10.0
        return logits


class _CharTok:
    eos_token_id = 0

    def encode(self, s):
        return [ord(c) for c in s]

    def decode(self, ids):
        return "".join(chr(i) for i in ids if i > 0)


def test_evaluator_end_to_end(tmp_path):
    tok = _CharTok()
    payload = '{"name": "f", "arguments": {}}'
    model = _EchoLM([ord(c) for c in payload] + [0])
    sample = {"prompt": "call f", "gt_tool_calls": [{"name": "f", "arguments": {}}]}
    p = tmp_path / "eval.jsonl"
    p.write_text(json.dumps(sample) + "\n")
    ev = ToolCallEvaluator(path=str(p), max_new_tokens=len(payload) + 1)
    model.prompt_len = len(sample["prompt"])
    out = ev.evaluate(model, tok)
    assert out["tool_call/_count"] == 1.0
    assert out["tool_call/name_correct"] == 1.0
    assert out["tool_call/args_exact_match"] == 1.0


def test_evaluator_shard_strides():
    samples = [{"prompt": "p", "gt_tool_calls": []} for _ in range(5)]
    ev = ToolCallEvaluator(samples=samples, sample_shard=(1, 2))
    assert len(ev._my_samples()) == 2
