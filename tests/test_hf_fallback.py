"""Generic HF-transformers fallback: an architecture NOT in the native
registry trains end to end (reference auto_model.py:380-643)."""

import os

import pytest
import torch

transformers = pytest.importorskip("transformers")

XGLM_CFG = dict(
    model_type="xglm", vocab_size=128, d_model=32, ffn_dim=64,
    num_layers=2, attention_heads=2, max_position_embeddings=64,
)


def test_fallback_builds_unregistered_arch():
    from automodel_amd.models.hf_fallback import HFFallbackForCausalLM
    from automodel_amd.models.registry import build_model

    m = build_model(config=dict(XGLM_CFG), architecture="XGLMForCausalLM",
                    dtype="float32", meta_init=False)
    assert isinstance(m, HFFallbackForCausalLM)
    ids = torch.randint(0, 128, (2, 16))
    logits = m(ids)
    assert logits.shape == (2, 16, 128)
    loss = m(ids, labels=ids.clone())
    assert loss.dim() == 0 and torch.isfinite(loss)
    loss.backward()


def test_fallback_state_dict_hf_keys(tmp_path):
    """Adapter keeps the exact HF key layout through consolidated export."""
    from safetensors.torch import load_file

    from automodel_amd.checkpoint.checkpointing import Checkpointer
    from automodel_amd.models.registry import build_model

    m = build_model(config=dict(XGLM_CFG), architecture="XGLMForCausalLM",
                    dtype="float32", meta_init=False)
    hf_keys = set(m.hf.state_dict().keys())
    adapted = set(m.state_dict_adapter.to_hf(m.state_dict()).keys())
    assert hf_keys == adapted
    out = tmp_path / "hf"
    Checkpointer(checkpoint_dir=str(tmp_path)).export_hf_safetensors(m, str(out))
    sd = {}
    for f in os.listdir(out):
        if f.endswith(".safetensors"):
            sd.update(load_file(str(out / f)))
    # exported keys are HF keys (modulo tied-alias dropping)
    assert sd.keys() <= hf_keys, sorted(sd.keys() - hf_keys)[:5]


def test_fallback_recipe_end_to_end(tmp_path):
    """train_ft drives the fallback two steps with checkpointing."""
    from automodel_amd.config.loader import ConfigNode
    from automodel_amd.recipes.llm.train_ft import (
        TrainFinetuneRecipeForNextTokenPrediction,
    )

    cfg = ConfigNode({
        "seed": 0,
        "model": {"architecture": "XGLMForCausalLM",
                  "config": dict(XGLM_CFG), "dtype": "float32"},
        "optimizer": {"lr": 1e-3},
        "step_scheduler": {"max_steps": 2},
        "dataloader": {"dataset": {"kind": "mock", "num_samples": 4,
                                   "seq_len": 16, "vocab_size": 128},
                       "batch_size": 2},
        "output_dir": str(tmp_path),
    })
    r = TrainFinetuneRecipeForNextTokenPrediction(cfg)
    r.setup()
    r.run_train_validation_loop()


def test_fallback_fsdp_layer_detection():
    """Generic decoder-stack detection finds the transformer blocks."""
    from automodel_amd.models.registry import build_model
    from automodel_amd.parallel.fsdp import detect_decoder_layers

    m = build_model(config=dict(XGLM_CFG), architecture="XGLMForCausalLM",
                    dtype="float32", meta_init=False)
    layers = detect_decoder_layers(m)
    assert len(layers) == 2
    assert all(type(l).__name__ == "XGLMDecoderLayer" for l in layers)


def test_kernel_patch_ladder_preserves_logits():
    """apply_kernel_patches swaps HF RMSNorms + silu MLPs for the native ops
    without changing logits, reusing the SAME weight Parameters."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.hf_patches import apply_kernel_patches
    from automodel_amd.ops.rms_norm import RMSNorm

    cfg = transformers.LlamaConfig(
        vocab_size=128, hidden_size=32, intermediate_size=48,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, attn_implementation="eager")
    torch.manual_seed(0)
    m = transformers.LlamaForCausalLM(cfg).eval()
    ids = torch.randint(0, 128, (2, 12))
    with torch.no_grad():
        ref = m(ids).logits
    params_before = {n: p for n, p in m.named_parameters()}
    counts = apply_kernel_patches(m)
    assert counts["rms_norm"] >= 5 and counts["swiglu_mlp"] == 2, counts
    assert any(isinstance(mod, RMSNorm) for mod in m.modules())
    with torch.no_grad():
        out = m(ids).logits
    torch.testing.assert_close(out, ref, atol=2e-5, rtol=2e-5)
    # same Parameter objects (optimizer/checkpoint identity preserved)
    for n, p in m.named_parameters():
        assert params_before[n] is p, n


def test_fallback_build_applies_patches():
    pytest.importorskip("transformers")
    from automodel_amd.models.hf_fallback import build_hf_fallback
    from automodel_amd.ops.rms_norm import RMSNorm

    m = build_hf_fallback(
        config=dict(model_type="llama", vocab_size=96, hidden_size=32,
                    intermediate_size=48, num_hidden_layers=2,
                    num_attention_heads=4, num_key_value_heads=2,
                    max_position_embeddings=64),
        architecture="LlamaForCausalLM", dtype="float32")
    assert any(isinstance(mod, RMSNorm) for mod in m.modules())
    ids = torch.randint(0, 96, (1, 8))
    labels = ids.clone()
    loss = m(ids, labels=labels)
    assert torch.isfinite(loss)
