"""Tiny-Llama CPU tests: forward shape, loss path, training-step sanity,
meta init, HF state-dict key parity."""

import pytest
import torch

from automodel_amd.loss.linear_ce import FusedLinearCrossEntropy
from automodel_amd.models.common.backend import BackendConfig
from automodel_amd.models.llama.model import LlamaConfig, LlamaForCausalLM
from automodel_amd.models.registry import build_model
from automodel_amd.optim.adamw import FusedAdamW

TINY = dict(
    vocab_size=256, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
    num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=128,
)


def make_model(**over):
    cfg = LlamaConfig(**{**TINY, **over})
    m = LlamaForCausalLM(cfg, backend=BackendConfig().for_cpu())
    m.init_weights()
    return m


def test_forward_shapes():
    m = make_model()
    ids = torch.randint(0, 256, (2, 16))
    logits = m(ids)
    assert logits.shape == (2, 16, 256)
    hidden = m(ids, return_hidden=True)
    assert hidden.shape == (2, 16, 64)


def test_loss_path_and_backward():
    m = make_model()
    m.loss_fn = FusedLinearCrossEntropy(backend="chunked", chunk_size=8)
    ids = torch.randint(0, 256, (2, 17))
    loss = m(ids[:, :-1], labels=ids[:, 1:].contiguous())
    assert loss.dim() == 0 and torch.isfinite(loss)
    loss.backward()
    assert m.model.layers[0].self_attn.q_proj.weight.grad is not None
    assert m.lm_head.weight.grad is not None


def test_train_steps_reduce_loss():
    torch.manual_seed(0)
    m = make_model()
    m.loss_fn = FusedLinearCrossEntropy(backend="chunked", chunk_size=32)
    opt = FusedAdamW(m.parameters(), lr=1e-3, weight_decay=0.0)
    ids = torch.randint(0, 256, (4, 33))
    inp, lab = ids[:, :-1], ids[:, 1:].contiguous()
    losses = []
    for _ in range(10):
        loss = m(inp, labels=lab) / lab.numel()
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(float(loss))
    assert losses[-1] < losses[0] * 0.9, losses


def test_meta_init_then_materialize():
    with torch.device("meta"):
        m = LlamaForCausalLM(LlamaConfig(**TINY))
    assert next(m.parameters()).is_meta
    m.init_weights(device="cpu")
    assert not next(m.parameters()).is_meta
    ids = torch.randint(0, 256, (1, 8))
    assert m(ids).shape == (1, 8, 256)


def test_hf_state_dict_key_parity():
    """Keys must match the HF llama layout exactly (identity adapter)."""
    m = make_model()
    keys = set(m.state_dict().keys())
    expected_samples = {
        "model.embed_tokens.weight",
        "model.layers.0.self_attn.q_proj.weight",
        "model.layers.0.self_attn.k_proj.weight",
        "model.layers.0.self_attn.v_proj.weight",
        "model.layers.0.self_attn.o_proj.weight",
        "model.layers.1.mlp.gate_proj.weight",
        "model.layers.1.mlp.up_proj.weight",
        "model.layers.1.mlp.down_proj.weight",
        "model.layers.0.input_layernorm.weight",
        "model.layers.0.post_attention_layernorm.weight",
        "model.norm.weight",
        "lm_head.weight",
    }
    assert expected_samples <= keys
    assert not any("rope_cos" in k for k in keys), "rope buffers must be non-persistent"


def test_registry_build_model():
    m = build_model(config=TINY, architecture="LlamaForCausalLM", meta_init=True,
                    dtype="float32")
    assert next(m.parameters()).is_meta
    m.init_weights(device="cpu")
    assert m.num_parameters() > 0


def test_tie_word_embeddings():
    m = make_model(tie_word_embeddings=True)
    assert m.lm_head.weight is m.model.embed_tokens.weight
    # only counted once
    untied = make_model(tie_word_embeddings=False)
    assert m.num_parameters() < untied.num_parameters()


def test_fused_projections_match_unfused():
    """fused qkv/gate_up forward == separate projections (same weights via
    the fused state-dict adapter)."""
    torch.manual_seed(0)
    base = make_model()
    cfg_f = LlamaConfig(**{**TINY}, fused_qkv=True, fused_gate_up=True)
    fused = LlamaForCausalLM(cfg_f, backend=BackendConfig().for_cpu())
    fused.load_state_dict(fused.state_dict_adapter.from_hf(base.state_dict()),
                          strict=False)
    ids = torch.randint(0, 256, (2, 16))
    assert torch.allclose(fused(ids), base(ids), atol=1e-5)
    # roundtrip back to HF keys
    hf = fused.state_dict_adapter.to_hf(fused.state_dict())
    assert "model.layers.0.self_attn.q_proj.weight" in hf
    assert torch.allclose(hf["model.layers.0.mlp.up_proj.weight"],
                          base.state_dict()["model.layers.0.mlp.up_proj.weight"])


def test_backend_resolve_head_dim_guard():
    """head_dim beyond the kernel max downgrades attn to sdpa at build time
    (visible warning); every dim <= 256 stays on the HIP kernels (round-2:
    the flash kernels cover 64/96/128/192/256 with pad-to-tile)."""
    import warnings

    from automodel_amd.models.common.backend import BackendConfig

    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        b = BackendConfig.resolve(None, "cuda", head_dim=512)
    assert b.attn == "sdpa"
    assert any("head_dim 512" in str(x.message) for x in w)
    for d in (64, 80, 96, 128, 192, 256):
        assert BackendConfig.resolve(None, "cuda", head_dim=d).attn == "hip", d


def test_hf_logits_parity_llama():
    """Bit-level check of the flagship architecture against HF transformers
    (eager): GQA attention, rope, rms norms, swiglu, lm_head."""
    transformers = pytest.importorskip("transformers")
    hf_cfg = transformers.LlamaConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=3, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rope_theta=10000.0,
        attn_implementation="eager", tie_word_embeddings=False,
    )
    torch.manual_seed(3)
    hf = transformers.LlamaForCausalLM(hf_cfg).eval()
    mine = LlamaForCausalLM(dict(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=3, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rope_theta=10000.0, rms_norm_eps=1e-6,
    )).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 24))
    with torch.no_grad():
        ref = hf(ids).logits
        out = mine(ids)
    torch.testing.assert_close(out, ref, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_llama3_rope_scaling():
    """Llama-3.1-style rope scaling parity (the bench model's rope)."""
    transformers = pytest.importorskip("transformers")
    scaling = {"rope_type": "llama3", "factor": 8.0,
               "low_freq_factor": 1.0, "high_freq_factor": 4.0,
               "original_max_position_embeddings": 64}
    hf_cfg = transformers.LlamaConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=256, rope_theta=10000.0,
        rope_scaling=dict(scaling), attn_implementation="eager",
        tie_word_embeddings=False,
    )
    torch.manual_seed(4)
    hf = transformers.LlamaForCausalLM(hf_cfg).eval()
    mine = LlamaForCausalLM(dict(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=256, rope_theta=10000.0, rms_norm_eps=1e-6,
        rope_scaling=dict(scaling),
    )).eval()
    mine.load_state_dict(hf.state_dict(), strict=False)
    ids = torch.randint(0, 300, (1, 100))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_bidirectional_llama_attends_to_future():
    """bidirectional=True (retrieval embedding models): early positions see
    later tokens; causal model's position-0 hidden is input-suffix-invariant."""
    cfg = dict(vocab_size=100, hidden_size=32, intermediate_size=64,
               num_hidden_layers=2, num_attention_heads=2, num_key_value_heads=1,
               max_position_embeddings=64)
    torch.manual_seed(0)
    causal = LlamaForCausalLM(cfg)
    causal.init_weights(device="cpu")
    torch.manual_seed(0)
    bidir = LlamaForCausalLM(dict(cfg, bidirectional=True))
    bidir.init_weights(device="cpu")

    a = torch.randint(0, 100, (1, 8))
    b = a.clone()
    b[0, -1] = (b[0, -1] + 1) % 100  # change only the LAST token
    with torch.no_grad():
        hc_a = causal(a, return_hidden=True)
        hc_b = causal(b, return_hidden=True)
        hb_a = bidir(a, return_hidden=True)
        hb_b = bidir(b, return_hidden=True)
    torch.testing.assert_close(hc_a[:, 0], hc_b[:, 0])      # causal: unchanged
    assert not torch.allclose(hb_a[:, 0], hb_b[:, 0])       # bidir: changed


def test_capabilities_sliding_window_cp_guard():
    from automodel_amd.models.common.capabilities import validate_model_against_mesh

    m = LlamaForCausalLM(dict(vocab_size=100, hidden_size=32, intermediate_size=64,
                              num_hidden_layers=1, num_attention_heads=2,
                              num_key_value_heads=1, max_position_embeddings=64,
                              sliding_window=16))
    probs = validate_model_against_mesh(m, {"cp": 2})
    assert any("sliding-window" in p for p in probs)
    assert not validate_model_against_mesh(m, {"cp": 1})


def test_capabilities_new_families_reject_unsupported_axes():
    from automodel_amd.models.common.capabilities import validate_model_against_mesh
    from automodel_amd.models.gemma.model import GemmaForCausalLM
    from automodel_amd.models.gpt_oss.model import GptOssForCausalLM

    m = GemmaForCausalLM(dict(vocab_size=100, hidden_size=32, intermediate_size=64,
                              num_hidden_layers=1, num_attention_heads=2,
                              num_key_value_heads=1, head_dim=16,
                              max_position_embeddings=64))
    # gemma now HAS a TP plan; PP/CP stay rejected
    assert not any("TP" in p for p in validate_model_against_mesh(m, {"tp": 2}))
    assert any("PP" in p for p in validate_model_against_mesh(m, {"pp": 2}))
    o = GptOssForCausalLM(dict(vocab_size=100, hidden_size=32, intermediate_size=48,
                               num_hidden_layers=2, num_attention_heads=2,
                               num_key_value_heads=1, head_dim=16,
                               num_local_experts=2, num_experts_per_tok=1,
                               max_position_embeddings=64, sliding_window=8))
    assert any("TP" in p for p in validate_model_against_mesh(o, {"tp": 2}))
