"""Gemma-2 family: logits parity against HF transformers on a tiny random
config, plus architecture-specific behaviors (softcap, sliding window,
(1+w) norm, embed scaling)."""

import pytest
import torch

from automodel_amd.models.gemma.model import GemmaConfig, GemmaForCausalLM

TINY = dict(vocab_size=300, hidden_size=64, intermediate_size=128,
            num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
            head_dim=16, max_position_embeddings=128, rope_theta=10000.0,
            sliding_window=8, query_pre_attn_scalar=16.0,
            attn_logit_softcapping=50.0, final_logit_softcapping=30.0)


def _mine():
    torch.manual_seed(0)
    m = GemmaForCausalLM(TINY)
    m.init_weights(device="cpu")
    return m.eval()


def test_gemma_hf_logits_parity():
    transformers = pytest.importorskip("transformers")
    hf_cfg = transformers.Gemma2Config(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, max_position_embeddings=128, rope_theta=10000.0,
        sliding_window=8, query_pre_attn_scalar=16,
        attn_logit_softcapping=50.0, final_logit_softcapping=30.0,
        attn_implementation="eager",
    )
    torch.manual_seed(1)
    hf = transformers.Gemma2ForCausalLM(hf_cfg).eval()
    mine = GemmaForCausalLM(TINY).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing  # only rope buffers

    ids = torch.randint(0, 300, (2, 24))
    with torch.no_grad():
        ref = hf(ids).logits
        out = mine(ids)
    torch.testing.assert_close(out, ref, atol=2e-4, rtol=2e-4)


def test_gemma_sliding_window_masks_far_tokens():
    m = _mine()
    ids = torch.randint(0, 300, (1, 32))
    with torch.no_grad():
        base = m(ids)
        # perturbing a token >window before the last position must NOT change
        # the last logits when ALL layers are sliding (even layers here are;
        # odd/global layers do see it, so compare through a single layer)
        layer = m.model.layers[0]          # layer 0: sliding (window 8)
        x = torch.randn(1, 32, 64)
        cos, sin = m.model.rope_cos[:32], m.model.rope_sin[:32]
        y0 = layer(x, cos, sin)[0, -1]
        x2 = x.clone()
        x2[0, 5] += 10.0                   # pos 5 is 26 back from pos 31 > 8
        y1 = layer(x2, cos, sin)[0, -1]
        torch.testing.assert_close(y0, y1)
        y2 = layer(x2, cos, sin)[0, 6]     # pos 6 sees pos 5 (distance 1)
        assert not torch.allclose(layer(x, cos, sin)[0, 6], y2)


def test_gemma_softcap_bounds_logit_influence():
    cfg = dict(TINY, attn_logit_softcapping=None, final_logit_softcapping=5.0)
    torch.manual_seed(0)
    m = GemmaForCausalLM(cfg)
    m.init_weights(device="cpu")
    with torch.no_grad():
        out = m(torch.randint(0, 300, (1, 8)))
    assert out.abs().max() <= 5.0 + 1e-5  # tanh cap


def test_gemma_norm_zero_init_is_identity_scale():
    from automodel_amd.models.gemma.model import GemmaRMSNorm

    n = GemmaRMSNorm(16, 1e-6, "torch")
    x = torch.randn(4, 16)
    y = n(x)
    rms = torch.rsqrt(x.float().pow(2).mean(-1, keepdim=True) + 1e-6)
    torch.testing.assert_close(y, (x.float() * rms).to(x.dtype), atol=1e-5, rtol=1e-5)


def test_gemma_generate_and_train_step():
    from automodel_amd.loss.masked_ce import MaskedCrossEntropy
    from automodel_amd.utils.generation import generate

    m = _mine()
    ids = torch.randint(0, 300, (1, 8))
    out = generate(m, ids, max_new_tokens=4)
    assert out.shape == (1, 12)
    m.loss_fn = lambda h, w, l: MaskedCrossEntropy()(h @ w.t(), l)
    loss = m(ids, labels=ids.clone())
    loss.backward()
    assert m.model.layers[0].mlp.gate_proj.weight.grad is not None
    assert m.lm_head.weight is m.model.embed_tokens.weight  # tied


def test_gemma3_hf_logits_parity():
    """Gemma-3: qk-norm, explicit layer_types, dual-frequency rope
    (local theta on sliding layers, global theta on full layers)."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.gemma.model import Gemma3ForCausalLM

    hf_cfg = transformers.Gemma3TextConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=7, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, max_position_embeddings=128, rms_norm_eps=1e-6,
        sliding_window=8, query_pre_attn_scalar=16,
        attn_implementation="eager", tie_word_embeddings=False,
    )
    torch.manual_seed(13)
    hf = transformers.Gemma3ForCausalLM(hf_cfg).eval()
    mine = Gemma3ForCausalLM(Gemma3ForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    assert mine.config.qk_norm and mine.config.layer_types is not None
    assert mine.config.rope_local_base_freq == 10000.0
    assert mine.config.rope_theta == 1000000.0
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 24))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_gemma1_hf_logits_parity():
    """Gemma-1: two-norm pre-norm layers, no softcapping/sliding —
    the same module in post_norms=False mode."""
    transformers = pytest.importorskip("transformers")
    torch.manual_seed(37)
    hf_cfg = transformers.GemmaConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, max_position_embeddings=128, rope_theta=10000.0,
        rms_norm_eps=1e-6, attn_implementation="eager",
        tie_word_embeddings=False)
    hf = transformers.GemmaForCausalLM(hf_cfg).eval()
    d = hf_cfg.to_dict()
    d["architectures"] = ["GemmaForCausalLM"]
    mine = GemmaForCausalLM(GemmaForCausalLM.config_from_hf(d)).eval()
    assert not mine.config.post_norms
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_gemma3_vlm_hf_logits_parity():
    """Gemma-3 multimodal: SigLIP tower, avg-pool soft-token projector,
    image-block bidirectional attention — text and text+image paths."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.gemma.vlm import Gemma3ForConditionalGeneration

    torch.manual_seed(60)
    hf_cfg = transformers.Gemma3Config(
        text_config=dict(vocab_size=300, hidden_size=64, intermediate_size=128,
                         num_hidden_layers=3, num_attention_heads=4,
                         num_key_value_heads=2, head_dim=16,
                         max_position_embeddings=128, sliding_window=8,
                         query_pre_attn_scalar=16, rms_norm_eps=1e-6,
                         tie_word_embeddings=False),
        vision_config=dict(hidden_size=32, intermediate_size=64,
                           num_hidden_layers=2, num_attention_heads=2,
                           image_size=16, patch_size=4),
        mm_tokens_per_image=4, image_token_id=299)
    hf = transformers.Gemma3ForConditionalGeneration(hf_cfg).eval()
    mine = Gemma3ForConditionalGeneration(
        Gemma3ForConditionalGeneration.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing

    ids = torch.randint(0, 290, (2, 14))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(input_ids=ids).logits,
                                   atol=2e-4, rtol=2e-4)
    pix = torch.randn(1, 3, 16, 16)
    seq = torch.cat([torch.randint(0, 290, (1, 3)), torch.full((1, 4), 299),
                     torch.randint(0, 290, (1, 5))], dim=1)
    with torch.no_grad():
        ref = hf(input_ids=seq, pixel_values=pix,
                 token_type_ids=(seq == 299).int()).logits
        out = mine(seq, pixel_values=pix)
    torch.testing.assert_close(out, ref, atol=2e-4, rtol=2e-4)
    # image tokens see each other bidirectionally (HF zero-inits the
    # projector, so randomize it first to make pixels matter)
    with torch.no_grad():
        torch.nn.init.normal_(
            mine.model.multi_modal_projector.mm_input_projection_weight, std=0.1)
        base = mine(seq, pixel_values=pix)
        out2 = mine(seq, pixel_values=pix + 10.0)
    assert not torch.allclose(base[0, 3], out2[0, 3])


def test_gemma_recipe_end_to_end(tmp_path):
    """train_ft drives the gemma family (softcapped CE path) two steps."""
    from automodel_amd.config.loader import ConfigNode
    from automodel_amd.recipes.llm.train_ft import (
        TrainFinetuneRecipeForNextTokenPrediction,
    )

    cfg = ConfigNode({
        "seed": 0,
        "model": {"architecture": "Gemma2ForCausalLM",
                  "config": dict(TINY, vocab_size=256),
                  "dtype": "float32"},
        "loss_fn": {"backend": "chunked", "chunk_size": 64},
        "optimizer": {"lr": 1e-3},
        "step_scheduler": {"max_steps": 2},
        "dataloader": {"dataset": {"kind": "mock", "num_samples": 4,
                                   "seq_len": 24, "vocab_size": 256},
                       "batch_size": 2},
        "output_dir": str(tmp_path),
    })
    r = TrainFinetuneRecipeForNextTokenPrediction(cfg)
    r.setup()
    r.run_train_validation_loop()
