"""Property-based tests (hypothesis) for core invariants: NF4 round-trip,
QAT grid, packing, rope tables, tool-call scoring, GEMV padding table."""

import pytest

hypothesis = pytest.importorskip("hypothesis")
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

COMMON = dict(max_examples=25, deadline=None)


@given(rows=st.integers(1, 8), blocks=st.integers(1, 6),
       scale=st.floats(0.01, 100.0), seed=st.integers(0, 2**16))
@settings(**COMMON)
def test_nf4_roundtrip_bounded_error(rows, blocks, scale, seed):
    from automodel_amd.quantization.nf4 import dequantize_nf4, quantize_nf4

    torch.manual_seed(seed)
    w = torch.randn(rows, blocks * 64) * scale
    packed, absmax = quantize_nf4(w, 64)
    deq = dequantize_nf4(packed, absmax, w.shape, 64)
    # error bounded by half the widest codebook gap per block
    # (largest NF4 gap is -1.0 -> -0.696 = 0.304)
    bound = absmax.reshape(rows, blocks, 1) * 0.304 / 2 + 1e-6
    err = (deq - w).abs().reshape(rows, blocks, 64)
    assert bool((err <= bound).all())


@given(n=st.integers(1, 6), bits=st.sampled_from([2, 3, 4, 8]),
       seed=st.integers(0, 2**16))
@settings(**COMMON)
def test_qat_values_on_grid(n, bits, seed):
    from automodel_amd.quantization.qat import fake_quant_per_group

    torch.manual_seed(seed)
    w = torch.randn(n, 32)
    q = fake_quant_per_group(w, n_bits=bits, group_size=32)
    qmax = 2 ** (bits - 1) - 1
    scale = w.abs().amax(-1, keepdim=True).clamp_min(1e-8) / qmax
    k = q / scale
    torch.testing.assert_close(k, k.round(), atol=1e-3, rtol=1e-3)


@given(lens=st.lists(st.integers(1, 40), min_size=1, max_size=8),
       seed=st.integers(0, 2**16))
@settings(**COMMON)
def test_block_causal_mask_structure(lens, seed):
    from automodel_amd.datasets.llm.packed_sequence import block_causal_mask

    cu = torch.tensor([0] + list(torch.tensor(lens).cumsum(0)), dtype=torch.int32)
    T = int(cu[-1])
    mask = block_causal_mask(cu, T)
    doc = torch.bucketize(torch.arange(T), cu[1:-1], right=True)
    i = torch.arange(T)
    expected = (doc[:, None] == doc[None, :]) & (i[None, :] <= i[:, None])
    assert torch.equal(mask.reshape(T, T).bool(), expected)


@given(seq=st.integers(2, 64), dim=st.sampled_from([8, 16, 32, 64]),
       base=st.floats(100.0, 1e6))
@settings(**COMMON)
def test_rope_tables_norm_preserving(seq, dim, base):
    """Rotary application preserves the L2 norm of every (pair) subspace."""
    from automodel_amd.ops.rope import apply_rope_ref, build_rope_cache

    cos, sin = build_rope_cache(dim, seq, base)
    assert cos.shape == (seq, dim)
    torch.testing.assert_close(cos[0], torch.ones(dim))  # position 0 = identity
    q = torch.randn(1, seq, 2, dim)
    qo, _ = apply_rope_ref(q, q, cos, sin)
    torch.testing.assert_close(qo.norm(dim=-1), q.norm(dim=-1),
                               atol=1e-4, rtol=1e-4)


@given(n_pred=st.integers(0, 5), n_gt=st.integers(0, 5),
       seed=st.integers(0, 999))
@settings(**COMMON)
def test_tool_call_scores_in_unit_interval(n_pred, n_gt, seed):
    import random

    from automodel_amd.eval.tool_calling import METRIC_KEYS, ToolCall, score_tool_calls

    rng = random.Random(seed)
    pred = [ToolCall(name=rng.choice("abc"), arguments={"k": rng.randint(0, 2)},
                     valid_json=True) for _ in range(n_pred)]
    gt = [{"name": rng.choice("abc"), "arguments": {"k": rng.randint(0, 2)}}
          for _ in range(n_gt)]
    m = score_tool_calls(pred, gt)
    assert set(m) == set(METRIC_KEYS)
    assert all(0.0 <= v <= 1.0 for v in m.values())


@given(b=st.integers(1, 16))
@settings(**COMMON)
def test_gemv_pad_table_targets_compiled_sizes(b):
    from automodel_amd.serving.decode_linear import _PAD_TO

    tgt = _PAD_TO.get(b, b)
    assert tgt >= b and tgt in (1, 2, 3, 4, 8, 16)
