"""CPU checks of the op reference implementations (the HIP kernels are
parity-tested against these same references in test_gpu_kernels.py)."""

import pytest
import torch

from automodel_amd.ops.attention import attention_ref, flash_attention
from automodel_amd.ops.rms_norm import rms_norm_ref
from automodel_amd.ops.rope import apply_rope_ref, build_rope_cache
from automodel_amd.ops.swiglu import swiglu_ref
from automodel_amd.loss.linear_ce import fused_linear_cross_entropy
from automodel_amd.loss.masked_ce import MaskedCrossEntropy

torch.manual_seed(0)


def test_rms_norm_ref_matches_manual():
    x = torch.randn(4, 64)
    w = torch.randn(64)
    y = rms_norm_ref(x, w, 1e-6)
    expected = x / (x.pow(2).mean(-1, keepdim=True) + 1e-6).sqrt() * w
    assert torch.allclose(y, expected, atol=1e-5)


def test_rope_ref_matches_hf_convention():
    B, S, H, D = 2, 16, 4, 32
    q = torch.randn(B, S, H, D)
    k = torch.randn(B, S, 2, D)
    cos, sin = build_rope_cache(D, S, base=10000.0)
    qo, ko = apply_rope_ref(q, k, cos, sin)
    # position 0: cos=1, sin=0 -> identity
    assert torch.allclose(qo[:, 0], q[:, 0], atol=1e-5)
    # norm preservation per pair
    assert torch.allclose(qo.norm(dim=-1), q.norm(dim=-1), atol=1e-4)


def test_rope_llama3_scaling_changes_long_wavelengths():
    cos_a, _ = build_rope_cache(64, 32, base=500000.0)
    cos_b, _ = build_rope_cache(
        64, 32, base=500000.0,
        scaling={"rope_type": "llama3", "factor": 8.0, "low_freq_factor": 1.0,
                 "high_freq_factor": 4.0, "original_max_position_embeddings": 8192},
    )
    assert not torch.allclose(cos_a, cos_b)


def test_attention_ref_vs_sdpa():
    B, S, Hq, Hk, D = 2, 32, 4, 2, 16
    q = torch.randn(B, S, Hq, D)
    k = torch.randn(B, S, Hk, D)
    v = torch.randn(B, S, Hk, D)
    o_ref = attention_ref(q, k, v, causal=True)
    o_sdpa = flash_attention(q, k, v, causal=True, backend="sdpa")
    assert torch.allclose(o_ref, o_sdpa, atol=1e-4)


def test_swiglu_ref():
    g, u = torch.randn(8, 16), torch.randn(8, 16)
    assert torch.allclose(swiglu_ref(g, u), torch.nn.functional.silu(g) * u)


def test_chunked_linear_ce_matches_dense():
    T, H, V = 64, 32, 100
    hidden = torch.randn(T, H, requires_grad=True)
    weight = torch.randn(V, H, requires_grad=True)
    labels = torch.randint(0, V, (T,))
    labels[5] = -100
    loss = fused_linear_cross_entropy(hidden, weight, labels, backend="chunked", chunk_size=10)
    loss.backward()

    h2 = hidden.detach().clone().requires_grad_(True)
    w2 = weight.detach().clone().requires_grad_(True)
    dense = torch.nn.functional.cross_entropy(h2 @ w2.t(), labels,
                                              ignore_index=-100, reduction="sum")
    dense.backward()
    assert torch.allclose(loss, dense, rtol=1e-5)
    assert torch.allclose(hidden.grad, h2.grad, atol=1e-5)
    assert torch.allclose(weight.grad, w2.grad, atol=1e-4)


def test_masked_ce_sum_semantics():
    logits = torch.randn(4, 8, 50)
    labels = torch.randint(0, 50, (4, 8))
    labels[0, :4] = -100
    loss = MaskedCrossEntropy()(logits, labels)
    ref = torch.nn.functional.cross_entropy(
        logits.reshape(-1, 50).float(), labels.reshape(-1),
        ignore_index=-100, reduction="sum")
    assert torch.allclose(loss, ref)
