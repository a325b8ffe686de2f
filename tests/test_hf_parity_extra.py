"""HF-transformers logits parity for the qk-norm (Qwen3) and MoE
(Qwen3-MoE) families — validates attention qk-norm, the gate's
softmax-topk routing, stacked-expert compute, and the state-dict adapter
end to end against the upstream implementations."""

import pytest
import torch

transformers = pytest.importorskip("transformers")


def test_hf_logits_parity_qwen3():
    from automodel_amd.models.llama.model import LlamaForCausalLM

    hf_cfg = transformers.Qwen3Config(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=3, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, max_position_embeddings=128, rope_theta=10000.0,
        rms_norm_eps=1e-6, attn_implementation="eager",
        tie_word_embeddings=False,
    )
    torch.manual_seed(5)
    hf = transformers.Qwen3ForCausalLM(hf_cfg).eval()
    mine = LlamaForCausalLM(dict(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=3, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, max_position_embeddings=128, rope_theta=10000.0,
        rms_norm_eps=1e-6, qk_norm=True,
    )).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 24))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_qwen3_moe():
    from automodel_amd.models.registry import build_model
    from automodel_amd.moe.state_dict_adapter import MoEStateDictAdapter

    hf_cfg = transformers.Qwen3MoeConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, max_position_embeddings=128, rope_theta=10000.0,
        rms_norm_eps=1e-6, num_experts=8, num_experts_per_tok=2,
        moe_intermediate_size=96, norm_topk_prob=True,
        router_aux_loss_coef=0.0, decoder_sparse_step=1,
        mlp_only_layers=[], attn_implementation="eager",
        tie_word_embeddings=False,
    )
    torch.manual_seed(6)
    hf = transformers.Qwen3MoeForCausalLM(hf_cfg).eval()
    mine = build_model(
        config=dict(
            vocab_size=300, hidden_size=64, intermediate_size=128,
            num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
            head_dim=16, max_position_embeddings=128, rope_theta=10000.0,
            rms_norm_eps=1e-6, qk_norm=True,
            moe=dict(n_routed_experts=8, n_activated_experts=2,
                     moe_intermediate_size=96, norm_topk_prob=True,
                     aux_loss_coeff=0.0),
        ),
        architecture="Qwen3MoeForCausalLM", dtype="float32",
        meta_init=False, device="cpu",
    ).eval()
    adapter = MoEStateDictAdapter(mine.config)
    native_sd = adapter.from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(native_sd, strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k or "expert_bias" in k for k in missing), missing

    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        ref = hf(ids).logits
        out = mine(ids)
    torch.testing.assert_close(out, ref, atol=3e-4, rtol=3e-4)


def test_hf_logits_parity_mixtral():
    from automodel_amd.models.registry import build_model
    from automodel_amd.moe.state_dict_adapter import MoEStateDictAdapter

    hf_cfg = transformers.MixtralConfig(
        vocab_size=300, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, max_position_embeddings=128, rope_theta=10000.0,
        rms_norm_eps=1e-6, num_local_experts=4, num_experts_per_tok=2,
        router_aux_loss_coef=0.0, attn_implementation="eager",
        tie_word_embeddings=False,
    )
    torch.manual_seed(7)
    hf = transformers.MixtralForCausalLM(hf_cfg).eval()
    mine = build_model(
        config=dict(
            vocab_size=300, hidden_size=64, intermediate_size=96,
            num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
            head_dim=16, max_position_embeddings=128, rope_theta=10000.0,
            rms_norm_eps=1e-6, hf_flavor="mixtral",
            moe=dict(n_routed_experts=4, n_activated_experts=2,
                     moe_intermediate_size=96, norm_topk_prob=True,
                     aux_loss_coeff=0.0),
        ),
        architecture="MixtralForCausalLM", dtype="float32",
        meta_init=False, device="cpu",
    ).eval()
    adapter = MoEStateDictAdapter(mine.config)
    native_sd = adapter.from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(native_sd, strict=False)
    assert not unexpected, unexpected
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=3e-4, rtol=3e-4)


def test_hf_logits_parity_deepseek_v3():
    """MLA (q/kv LoRA projections, decoupled interleaved rope), dense-first
    layers, sigmoid gate with e_score_correction_bias, shared expert."""
    from automodel_amd.models.deepseek_v3.model import DeepseekV3ForCausalLM
    from automodel_amd.moe.state_dict_adapter import MoEStateDictAdapter

    torch.manual_seed(9)
    kw = dict(vocab_size=300, hidden_size=64, intermediate_size=96,
              num_hidden_layers=3, num_attention_heads=4,
              first_k_dense_replace=1, q_lora_rank=32, kv_lora_rank=16,
              qk_nope_head_dim=16, qk_rope_head_dim=8, v_head_dim=16,
              max_position_embeddings=128, rope_theta=10000.0, rms_norm_eps=1e-6)
    hf_cfg = transformers.DeepseekV3Config(
        moe_intermediate_size=48, num_key_value_heads=4, n_routed_experts=8,
        n_shared_experts=1, num_experts_per_tok=2, n_group=1, topk_group=1,
        norm_topk_prob=True, routed_scaling_factor=1.0,
        attn_implementation="eager", tie_word_embeddings=False, **kw)
    hf = transformers.DeepseekV3ForCausalLM(hf_cfg).eval()
    mine = DeepseekV3ForCausalLM(dict(
        moe=dict(n_routed_experts=8, n_shared_experts=1, n_activated_experts=2,
                 moe_intermediate_size=48, shared_expert_intermediate_size=48,
                 norm_topk_prob=True, score_func="sigmoid", route_scale=1.0,
                 expert_bias=True), **kw)).eval()
    sd = MoEStateDictAdapter(mine.config).from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=3e-4, rtol=3e-4)


def test_hf_logits_parity_phi3():
    """Phi-3 rides the llama family's fused qkv/gate_up layout unchanged."""
    from automodel_amd.models.llama.model import LlamaForCausalLM

    hf_cfg = transformers.Phi3Config(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rope_theta=10000.0, rms_norm_eps=1e-6,
        attn_implementation="eager", tie_word_embeddings=False,
        pad_token_id=0, eos_token_id=1, bos_token_id=2,
    )
    torch.manual_seed(11)
    hf = transformers.Phi3ForCausalLM(hf_cfg).eval()
    mine = LlamaForCausalLM(
        LlamaForCausalLM.config_from_hf(hf_cfg.to_dict() | {"architectures": ["Phi3ForCausalLM"]})
    ).eval()
    assert mine.config.fused_qkv and mine.config.fused_gate_up
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def _gptoss_pair(rope_params, window=8, seed=12):
    from automodel_amd.models.gpt_oss.model import GptOssForCausalLM

    torch.manual_seed(seed)
    mk = dict(vocab_size=300, hidden_size=64, intermediate_size=96,
              num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
              head_dim=16, num_local_experts=4, num_experts_per_tok=2,
              max_position_embeddings=128, rms_norm_eps=1e-6,
              sliding_window=window)
    hf_cfg = transformers.GptOssConfig(
        rope_parameters=rope_params, attn_implementation="eager",
        tie_word_embeddings=False, **mk)
    hf = transformers.GptOssForCausalLM(hf_cfg).eval()
    mine = GptOssForCausalLM(GptOssForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    return hf, mine


def test_hf_logits_parity_gpt_oss():
    """Attention sinks, sliding window, biased projections, interleaved
    clamped-GLU experts, top-k-first router."""
    hf, mine = _gptoss_pair({"rope_type": "default", "rope_theta": 10000.0})
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_gpt_oss_yarn():
    """YaRN rope branch (build_rope_cache 'yarn') end to end."""
    hf, mine = _gptoss_pair({
        "rope_type": "yarn", "rope_theta": 150000.0, "factor": 32.0,
        "beta_fast": 32.0, "beta_slow": 1.0,
        "original_max_position_embeddings": 4096, "truncate": False,
    })
    ids = torch.randint(0, 300, (1, 24))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_gpt_oss_trains():
    from automodel_amd.loss.masked_ce import MaskedCrossEntropy
    from automodel_amd.models.registry import build_model

    m = build_model(config=dict(
        vocab_size=300, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, num_local_experts=4, num_experts_per_tok=2,
        max_position_embeddings=128, sliding_window=8),
        architecture="GptOssForCausalLM", dtype="float32",
        meta_init=False, device="cpu")
    m.loss_fn = lambda h, w, l: MaskedCrossEntropy()(h @ w.t(), l)
    ids = torch.randint(0, 300, (2, 16))
    loss = m(ids, labels=ids.clone())
    loss.backward()
    assert m.model.layers[0].self_attn.sinks.grad is not None
    assert m.model.layers[0].mlp.experts.gate_up_proj.grad is not None


def test_hf_logits_parity_qwen2_vl():
    """Full Qwen2-VL: ViT tower (conv3d patch embed, 2-axis rotary,
    per-image attention, quick-gelu, 2x2 merger), image splice, and m-rope
    text decoder — text-only AND text+image paths."""
    from automodel_amd.models.qwen2_vl.model import Qwen2VLForConditionalGeneration

    torch.manual_seed(20)
    hf_cfg = transformers.Qwen2VLConfig(
        text_config=dict(vocab_size=300, hidden_size=64, intermediate_size=128,
                         num_hidden_layers=2, num_attention_heads=4,
                         num_key_value_heads=2, max_position_embeddings=256,
                         rope_theta=10000.0, rms_norm_eps=1e-6,
                         rope_scaling={"type": "mrope", "mrope_section": [2, 3, 3]},
                         tie_word_embeddings=False),
        vision_config=dict(embed_dim=32, depth=2, num_heads=2, hidden_size=64,
                           patch_size=4, temporal_patch_size=2,
                           spatial_merge_size=2, in_channels=3, mlp_ratio=4),
        image_token_id=299, vision_start_token_id=298,
    )
    hf = transformers.Qwen2VLForConditionalGeneration(hf_cfg).eval()
    mine = Qwen2VLForConditionalGeneration(
        Qwen2VLForConditionalGeneration.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k or "rot_inv" in k for k in missing), missing

    ids = torch.randint(0, 290, (2, 20))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(input_ids=ids).logits,
                                   atol=2e-4, rtol=2e-4)

    grid = torch.tensor([[1, 4, 4]])
    pix = torch.randn(16, 3 * 2 * 4 * 4)
    seq = torch.cat([torch.randint(0, 290, (1, 3)), torch.tensor([[298]]),
                     torch.full((1, 4), 299), torch.randint(0, 290, (1, 5))], dim=1)
    with torch.no_grad():
        ref = hf(input_ids=seq, pixel_values=pix, image_grid_thw=grid,
                 mm_token_type_ids=(seq == 299).int()).logits
        out = mine(seq, pixel_values=pix, image_grid_thw=grid)
    torch.testing.assert_close(out, ref, atol=2e-4, rtol=2e-4)


def test_qwen2_vl_trains():
    from automodel_amd.loss.masked_ce import MaskedCrossEntropy
    from automodel_amd.models.qwen2_vl.model import Qwen2VLForConditionalGeneration

    m = Qwen2VLForConditionalGeneration(dict(
        text=dict(vocab_size=300, hidden_size=64, intermediate_size=128,
                  num_hidden_layers=2, num_attention_heads=4,
                  num_key_value_heads=2, max_position_embeddings=256,
                  attention_bias=True),
        vision=dict(embed_dim=32, depth=1, num_heads=2, hidden_size=64,
                    patch_size=4, temporal_patch_size=2, spatial_merge_size=2),
        mrope_section=(2, 3, 3), image_token_id=299,
    ))
    m.init_weights(device="cpu")
    m.loss_fn = lambda h, w, l: MaskedCrossEntropy()(h @ w.t(), l)
    seq = torch.cat([torch.randint(0, 290, (1, 4)), torch.full((1, 4), 299),
                     torch.randint(0, 290, (1, 4))], dim=1)
    pix = torch.randn(16, 3 * 2 * 4 * 4)
    loss = m(seq, pixel_values=pix, image_grid_thw=torch.tensor([[1, 4, 4]]),
             labels=seq.clone())
    loss.backward()
    assert m.model.visual.blocks[0].attn.qkv.weight.grad is not None
    assert m.model.language_model.layers[0].self_attn.q_proj.weight.grad is not None


def test_hf_logits_parity_qwen2_moe():
    """Qwen2-MoE: softmax-all routing without topk renorm + sigmoid-GATED
    shared expert (shared_expert_gate) + qkv bias."""
    from automodel_amd.moe.model import MoEForCausalLM, MoEModelConfig
    from automodel_amd.moe.state_dict_adapter import MoEStateDictAdapter

    torch.manual_seed(30)
    hf_cfg = transformers.Qwen2MoeConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rope_theta=10000.0, rms_norm_eps=1e-6,
        num_experts=4, num_experts_per_tok=2, moe_intermediate_size=48,
        shared_expert_intermediate_size=96, norm_topk_prob=False,
        decoder_sparse_step=1, mlp_only_layers=[],
        attn_implementation="eager", tie_word_embeddings=False)
    hf = transformers.Qwen2MoeForCausalLM(hf_cfg).eval()
    d = hf_cfg.to_dict()
    d["architectures"] = ["Qwen2MoeForCausalLM"]
    mine = MoEForCausalLM(MoEModelConfig.from_hf_config(d)).eval()
    assert mine.config.moe.shared_expert_gate
    sd = MoEStateDictAdapter(mine.config).from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=3e-4, rtol=3e-4)


def test_hf_logits_parity_nemotron():
    """Nemotron: LayerNorm-1P (bias, (1+w) gain), squared-ReLU MLP,
    partial rotary (factor 0.5)."""
    from automodel_amd.models.nemotron.model import NemotronForCausalLM

    torch.manual_seed(31)
    hf_cfg = transformers.NemotronConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, attn_implementation="eager",
        tie_word_embeddings=False)
    hf = transformers.NemotronForCausalLM(hf_cfg).eval()
    mine = NemotronForCausalLM(
        NemotronForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)
    # trains
    from automodel_amd.loss.masked_ce import MaskedCrossEntropy

    mine.loss_fn = lambda h, w, l: MaskedCrossEntropy()(h @ w.t(), l)
    loss = mine(ids, labels=ids.clone())
    loss.backward()
    assert mine.model.layers[0].mlp.up_proj.weight.grad is not None


def test_hf_logits_parity_glm4_moe():
    """GLM4-MoE: DeepSeek-style sigmoid+bias routing and shared expert
    under plain GQA with partial rotary (0.5)."""
    from automodel_amd.models.glm4_moe.model import Glm4MoeForCausalLM

    torch.manual_seed(33)
    hf_cfg = transformers.Glm4MoeConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=3, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, attn_implementation="eager",
        tie_word_embeddings=False, n_routed_experts=8, n_shared_experts=1,
        num_experts_per_tok=2, moe_intermediate_size=48,
        first_k_dense_replace=1, norm_topk_prob=True,
        routed_scaling_factor=1.0, head_dim=16)
    hf = transformers.Glm4MoeForCausalLM(hf_cfg).eval()
    mine = Glm4MoeForCausalLM(
        Glm4MoeForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    sd = mine.state_dict_adapter.from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=3e-4, rtol=3e-4)


def test_hf_logits_parity_smollm3():
    """SmolLM3: llama family + NoPE layers (no_rope_layers mask)."""
    from automodel_amd.models.llama.model import LlamaForCausalLM

    torch.manual_seed(35)
    hf_cfg = transformers.SmolLM3Config(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rope_theta=10000.0, rms_norm_eps=1e-6,
        attn_implementation="eager", tie_word_embeddings=False,
        pad_token_id=0, bos_token_id=1, eos_token_id=2)
    hf = transformers.SmolLM3ForCausalLM(hf_cfg).eval()
    d = hf_cfg.to_dict()
    d["architectures"] = ["SmolLM3ForCausalLM"]
    mine = LlamaForCausalLM(LlamaForCausalLM.config_from_hf(d)).eval()
    assert mine.config.no_rope_layers == [1, 1, 1, 0]
    assert not mine.model.layers[3].self_attn.use_rope
    mine.load_state_dict(hf.state_dict(), strict=False)
    ids = torch.randint(0, 300, (2, 20))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_olmoe():
    """OLMoE: full-projection q/k RMSNorm + softmax routing without topk
    renorm, stacked experts."""
    from automodel_amd.moe.model import MoEForCausalLM, MoEModelConfig
    from automodel_amd.moe.state_dict_adapter import MoEStateDictAdapter

    torch.manual_seed(36)
    hf_cfg = transformers.OlmoeConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        num_experts=4, num_experts_per_tok=2, max_position_embeddings=128,
        rope_theta=10000.0, rms_norm_eps=1e-5, attn_implementation="eager",
        tie_word_embeddings=False, pad_token_id=0, eos_token_id=2)
    hf = transformers.OlmoeForCausalLM(hf_cfg).eval()
    d = hf_cfg.to_dict()
    d["architectures"] = ["OlmoeForCausalLM"]
    mine = MoEForCausalLM(MoEModelConfig.from_hf_config(d)).eval()
    assert mine.config.qk_norm_full
    sd = MoEStateDictAdapter(mine.config).from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=3e-4, rtol=3e-4)


def test_hf_logits_parity_mistral_sliding_window():
    """Mistral with a real sliding window (the flagship flash path is
    unaffected: window None keeps the fused kernel)."""
    from automodel_amd.models.llama.model import LlamaForCausalLM

    torch.manual_seed(38)
    hf_cfg = transformers.MistralConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rope_theta=10000.0, rms_norm_eps=1e-6,
        sliding_window=8, attn_implementation="eager", tie_word_embeddings=False)
    hf = transformers.MistralForCausalLM(hf_cfg).eval()
    d = hf_cfg.to_dict()
    d["architectures"] = ["MistralForCausalLM"]
    mine = LlamaForCausalLM(LlamaForCausalLM.config_from_hf(d)).eval()
    assert mine.config.sliding_window == 8
    mine.load_state_dict(hf.state_dict(), strict=False)
    ids = torch.randint(0, 300, (2, 24))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)
    # flagship configs (no window) keep the flash path
    assert LlamaForCausalLM(dict(vocab_size=100, hidden_size=32,
                                 intermediate_size=64, num_hidden_layers=1,
                                 num_attention_heads=2, num_key_value_heads=1,
                                 max_position_embeddings=32)) \
        .config.sliding_window is None


def test_hf_logits_parity_deepseek_v2():
    """DeepSeek-V2 through the V3 class: softmax routing (no bias, no topk
    renorm in the HF V2 router) with the same MLA attention."""
    from automodel_amd.models.deepseek_v3.model import DeepseekV3ForCausalLM
    from automodel_amd.moe.state_dict_adapter import MoEStateDictAdapter

    torch.manual_seed(40)
    hf_cfg = transformers.DeepseekV2Config(
        vocab_size=300, hidden_size=64, intermediate_size=96,
        moe_intermediate_size=48, num_hidden_layers=3, num_attention_heads=4,
        num_key_value_heads=4, n_routed_experts=8, n_shared_experts=1,
        num_experts_per_tok=2, first_k_dense_replace=1, q_lora_rank=32,
        kv_lora_rank=16, qk_nope_head_dim=16, qk_rope_head_dim=8,
        v_head_dim=16, n_group=1, topk_group=1, norm_topk_prob=True,
        routed_scaling_factor=1.0, max_position_embeddings=128,
        rope_theta=10000.0, rms_norm_eps=1e-6, attn_implementation="eager",
        tie_word_embeddings=False)
    hf = transformers.DeepseekV2ForCausalLM(hf_cfg).eval()
    d = hf_cfg.to_dict()
    d["architectures"] = ["DeepseekV2ForCausalLM"]
    mine = DeepseekV3ForCausalLM(DeepseekV3ForCausalLM.config_from_hf(d)).eval()
    assert mine.config.moe.score_func == "softmax"
    sd = MoEStateDictAdapter(mine.config).from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=3e-4, rtol=3e-4)


def test_hf_logits_parity_qwen2_biases():
    """Qwen2: HF hardcodes qkv biases (True) regardless of config — biases
    must load and contribute (randomized here so the test is meaningful)."""
    from automodel_amd.models.llama.model import LlamaForCausalLM

    torch.manual_seed(41)
    hf_cfg = transformers.Qwen2Config(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rope_theta=10000.0, rms_norm_eps=1e-6,
        attn_implementation="eager", tie_word_embeddings=False)
    hf = transformers.Qwen2ForCausalLM(hf_cfg).eval()
    with torch.no_grad():
        for n, p in hf.named_parameters():
            if n.endswith("bias"):
                p.normal_(0, 0.1)
    d = hf_cfg.to_dict()
    d["architectures"] = ["Qwen2ForCausalLM"]
    mine = LlamaForCausalLM(LlamaForCausalLM.config_from_hf(d)).eval()
    assert mine.config.attention_bias
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_ernie4_5():
    """Ernie-4.5: interleaved rope convention (handled by the permutation-
    invariance trick) + rope_parameters theta source + decoupled head_dim."""
    from automodel_amd.models.llama.model import LlamaForCausalLM

    torch.manual_seed(42)
    hf_cfg = transformers.Ernie4_5Config(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, attn_implementation="eager",
        tie_word_embeddings=False)
    hf = transformers.Ernie4_5ForCausalLM(hf_cfg).eval()
    d = hf_cfg.to_dict()
    d["architectures"] = ["Ernie4_5ForCausalLM"]
    mine = LlamaForCausalLM(LlamaForCausalLM.config_from_hf(d)).eval()
    assert mine.config.rope_interleaved
    assert mine.config.rope_theta == 500000.0   # from rope_parameters
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_qwen2_5_vl():
    """Qwen2.5-VL: RMSNorm ViT, biased-SwiGLU MLP, WINDOWED vision attention
    with merge-unit token reorder — text and text+image paths."""
    from automodel_amd.models.qwen2_vl.model import Qwen2_5_VLForConditionalGeneration

    torch.manual_seed(50)
    hf_cfg = transformers.Qwen2_5_VLConfig(
        text_config=dict(vocab_size=300, hidden_size=64, intermediate_size=128,
                         num_hidden_layers=2, num_attention_heads=4,
                         num_key_value_heads=2, max_position_embeddings=256,
                         rope_theta=10000.0, rms_norm_eps=1e-6,
                         rope_scaling={"type": "mrope", "mrope_section": [2, 3, 3]},
                         tie_word_embeddings=False),
        vision_config=dict(hidden_size=32, depth=2, num_heads=2,
                           out_hidden_size=64, patch_size=4,
                           temporal_patch_size=2, spatial_merge_size=2,
                           in_channels=3, intermediate_size=64,
                           window_size=16, fullatt_block_indexes=[1]),
        image_token_id=299, vision_start_token_id=298)
    hf = transformers.Qwen2_5_VLForConditionalGeneration(hf_cfg).eval()
    mine = Qwen2_5_VLForConditionalGeneration(
        Qwen2_5_VLForConditionalGeneration.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected

    ids = torch.randint(0, 290, (2, 20))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(input_ids=ids).logits,
                                   atol=2e-4, rtol=2e-4)
    grid = torch.tensor([[1, 8, 8]])
    pix = torch.randn(64, 3 * 2 * 4 * 4)
    seq = torch.cat([torch.randint(0, 290, (1, 3)), torch.tensor([[298]]),
                     torch.full((1, 16), 299), torch.randint(0, 290, (1, 4))], dim=1)
    with torch.no_grad():
        ref = hf(input_ids=seq, pixel_values=pix, image_grid_thw=grid,
                 mm_token_type_ids=(seq == 299).int()).logits
        out = mine(seq, pixel_values=pix, image_grid_thw=grid)
    torch.testing.assert_close(out, ref, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_llava():
    """LLaVA: CLIP tower (class token, learned positions, pre/post LN,
    quick-GELU), feature layer -2 with class-token drop, GELU projector,
    image splice — text and text+image paths."""
    from automodel_amd.models.llava.model import LlavaForConditionalGeneration

    torch.manual_seed(55)
    hf_cfg = transformers.LlavaConfig(
        text_config=dict(model_type="llama", vocab_size=300, hidden_size=64,
                         intermediate_size=128, num_hidden_layers=2,
                         num_attention_heads=4, num_key_value_heads=2,
                         max_position_embeddings=128, rms_norm_eps=1e-6,
                         rope_theta=10000.0, tie_word_embeddings=False),
        vision_config=dict(model_type="clip_vision_model", hidden_size=32,
                           intermediate_size=64, num_hidden_layers=2,
                           num_attention_heads=2, image_size=16, patch_size=4,
                           projection_dim=32),
        image_token_id=299)
    hf = transformers.LlavaForConditionalGeneration(hf_cfg).eval()
    mine = LlavaForConditionalGeneration(
        LlavaForConditionalGeneration.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert mine.config.image_token_id == 299

    ids = torch.randint(0, 290, (2, 12))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(input_ids=ids).logits,
                                   atol=2e-4, rtol=2e-4)
    pix = torch.randn(1, 3, 16, 16)
    seq = torch.cat([torch.randint(0, 290, (1, 3)), torch.full((1, 16), 299),
                     torch.randint(0, 290, (1, 4))], dim=1)
    with torch.no_grad():
        ref = hf(input_ids=seq, pixel_values=pix).logits
        out = mine(seq, pixel_values=pix)
    torch.testing.assert_close(out, ref, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_mistral3():
    """Mistral-3 VLM: Pixtral tower (RMSNorm blocks, 2-D interleaved-freq
    rotary, bias-free swiglu), unfold patch merger, GELU projector."""
    from automodel_amd.models.mistral3.model import Mistral3ForConditionalGeneration

    torch.manual_seed(70)
    hf_cfg = transformers.Mistral3Config(
        text_config=dict(model_type="mistral", vocab_size=300, hidden_size=64,
                         intermediate_size=128, num_hidden_layers=2,
                         num_attention_heads=4, num_key_value_heads=2,
                         max_position_embeddings=128, rms_norm_eps=1e-6,
                         rope_theta=10000.0, sliding_window=None,
                         tie_word_embeddings=False),
        vision_config=dict(model_type="pixtral", hidden_size=32,
                           intermediate_size=64, num_hidden_layers=2,
                           num_attention_heads=2, image_size=16, patch_size=4,
                           head_dim=16),
        image_token_index=299, spatial_merge_size=2,
        multimodal_projector_bias=False)
    hf = transformers.Mistral3ForConditionalGeneration(hf_cfg).eval()
    mine = Mistral3ForConditionalGeneration(
        Mistral3ForConditionalGeneration.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing

    ids = torch.randint(0, 290, (2, 12))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(input_ids=ids).logits,
                                   atol=2e-4, rtol=2e-4)
    pix = torch.randn(1, 3, 16, 16)
    seq = torch.cat([torch.randint(0, 290, (1, 3)), torch.full((1, 4), 299),
                     torch.randint(0, 290, (1, 4))], dim=1)
    with torch.no_grad():
        ref = hf(input_ids=seq, pixel_values=pix,
                 image_sizes=torch.tensor([[16, 16]])).logits
        out = mine(seq, pixel_values=pix)
    torch.testing.assert_close(out, ref, atol=3e-4, rtol=3e-4)


def test_hf_logits_parity_olmo2():
    """OLMo-2: full-width qk-norm + norms on sublayer OUTPUTS."""
    from automodel_amd.models.llama.model import LlamaForCausalLM

    torch.manual_seed(80)
    hf_cfg = transformers.Olmo2Config(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rope_theta=10000.0, rms_norm_eps=1e-6,
        attn_implementation="eager", tie_word_embeddings=False,
        eos_token_id=2, pad_token_id=0)
    hf = transformers.Olmo2ForCausalLM(hf_cfg).eval()
    d = hf_cfg.to_dict()
    d["architectures"] = ["Olmo2ForCausalLM"]
    mine = LlamaForCausalLM(LlamaForCausalLM.config_from_hf(d)).eval()
    assert mine.config.olmo2_layout and mine.config.qk_norm_full
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_granite():
    """Granite: llama layout + four scalar multipliers (attention scale IS
    the multiplier — no implicit 1/sqrt(d)); tested with NON-neutral values
    so every multiplier is exercised."""
    from automodel_amd.models.llama.model import LlamaForCausalLM

    torch.manual_seed(81)
    hf_cfg = transformers.GraniteConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rope_theta=10000.0, rms_norm_eps=1e-6,
        attention_multiplier=0.2, residual_multiplier=0.7,
        embedding_multiplier=3.0, logits_scaling=2.5,
        attn_implementation="eager", tie_word_embeddings=False)
    hf = transformers.GraniteForCausalLM(hf_cfg).eval()
    d = hf_cfg.to_dict()
    d["architectures"] = ["GraniteForCausalLM"]
    mine = LlamaForCausalLM(LlamaForCausalLM.config_from_hf(d)).eval()
    assert mine.config.attention_multiplier == 0.2
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_granite_moe():
    """GraniteMoe: granite multipliers + topk-then-softmax routing +
    block_sparse_moe stacked-expert key layout."""
    from automodel_amd.moe.model import MoEForCausalLM, MoEModelConfig
    from automodel_amd.moe.state_dict_adapter import MoEStateDictAdapter

    torch.manual_seed(90)
    hf_cfg = transformers.GraniteMoeConfig(
        vocab_size=300, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        num_local_experts=4, num_experts_per_tok=2, max_position_embeddings=128,
        rope_theta=10000.0, rms_norm_eps=1e-6,
        attention_multiplier=0.25, residual_multiplier=0.8,
        embedding_multiplier=2.0, logits_scaling=1.5,
        attn_implementation="eager", tie_word_embeddings=False)
    hf = transformers.GraniteMoeForCausalLM(hf_cfg).eval()
    d = hf_cfg.to_dict()
    d["architectures"] = ["GraniteMoeForCausalLM"]
    mine = MoEForCausalLM(MoEModelConfig.from_hf_config(d)).eval()
    assert mine.config.moe.topk_then_softmax
    sd = MoEStateDictAdapter(mine.config).from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=3e-4, rtol=3e-4)


def test_hf_logits_parity_gpt2():
    """GPT-2: learned absolute positions, Conv1D transposed weights,
    biased LayerNorms, tanh-GELU."""
    from automodel_amd.models.gpt2.model import GPT2LMHeadModel

    torch.manual_seed(91)
    hf_cfg = transformers.GPT2Config(vocab_size=300, n_embd=64, n_layer=2,
                                     n_head=4, n_positions=128,
                                     bos_token_id=0, eos_token_id=1,
                                     tie_word_embeddings=False)
    hf = transformers.GPT2LMHeadModel(hf_cfg).eval()
    mine = GPT2LMHeadModel(GPT2LMHeadModel.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not missing and not unexpected, (missing, unexpected)
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)
    # trains with the framework loss convention
    from automodel_amd.loss.masked_ce import MaskedCrossEntropy

    mine.loss_fn = lambda h, w, l: MaskedCrossEntropy()(h @ w.t(), l)
    loss = mine(ids, labels=ids.clone())
    loss.backward()
    assert mine.transformer.h[0].attn.c_attn.weight.grad is not None


def test_hf_logits_parity_falcon():
    """Falcon (new decoder arch): parallel attention+MLP residual, per-KV-
    group interleaved fused qkv, rotary, biased LNs."""
    from automodel_amd.models.falcon.model import FalconForCausalLM

    torch.manual_seed(92)
    hf_cfg = transformers.FalconConfig(
        vocab_size=300, hidden_size=64, num_hidden_layers=2,
        num_attention_heads=4, new_decoder_architecture=True, num_kv_heads=2,
        max_position_embeddings=128, rope_theta=10000.0,
        bos_token_id=0, eos_token_id=1, tie_word_embeddings=False,
        attn_implementation="eager")
    hf = transformers.FalconForCausalLM(hf_cfg).eval()
    mine = FalconForCausalLM(FalconForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_phi2():
    """Phi-1/2: parallel attention+MLP over ONE shared LN, partial rotary,
    biased everything incl. lm_head."""
    from automodel_amd.models.phi.model import PhiForCausalLM

    torch.manual_seed(93)
    hf_cfg = transformers.PhiConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, partial_rotary_factor=0.5,
        rope_theta=10000.0, bos_token_id=0, eos_token_id=1,
        tie_word_embeddings=False, attn_implementation="eager")
    hf = transformers.PhiForCausalLM(hf_cfg).eval()
    mine = PhiForCausalLM(PhiForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_cohere():
    """Cohere Command-R: parallel residual over one bias-free LayerNorm,
    interleaved rope, logit_scale, tied embeddings."""
    from automodel_amd.models.cohere.model import CohereForCausalLM

    torch.manual_seed(94)
    hf_cfg = transformers.CohereConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rope_theta=10000.0, logit_scale=0.0625,
        bos_token_id=0, eos_token_id=1, attn_implementation="eager")
    hf = transformers.CohereForCausalLM(hf_cfg).eval()
    mine = CohereForCausalLM(CohereForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_starcoder2():
    """StarCoder2: biased LayerNorms + biased projections + plain GELU MLP
    under the llama-shaped pre-norm skeleton."""
    from automodel_amd.models.starcoder2.model import Starcoder2ForCausalLM

    torch.manual_seed(95)
    hf_cfg = transformers.Starcoder2Config(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rope_theta=10000.0,
        bos_token_id=0, eos_token_id=1, tie_word_embeddings=False,
        attn_implementation="eager")
    hf = transformers.Starcoder2ForCausalLM(hf_cfg).eval()
    mine = Starcoder2ForCausalLM(
        Starcoder2ForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_stablelm():
    """StableLM-2: biased LayerNorms + SwiGLU MLP + partial rotary 0.25
    under the llama-shaped pre-norm skeleton."""
    from automodel_amd.models.stablelm.model import StableLmForCausalLM

    torch.manual_seed(96)
    hf_cfg = transformers.StableLmConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        partial_rotary_factor=0.25, use_qkv_bias=True,
        max_position_embeddings=128, rope_theta=10000.0,
        bos_token_id=0, eos_token_id=1, tie_word_embeddings=False,
        attn_implementation="eager")
    hf = transformers.StableLmForCausalLM(hf_cfg).eval()
    mine = StableLmForCausalLM(
        StableLmForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_helium():
    """Helium: llama-shaped with pair-INTERLEAVED rotate_half and explicit
    head_dim — rides the llama family's rope_interleaved path."""
    from automodel_amd.models.llama.model import LlamaForCausalLM

    torch.manual_seed(97)
    hf_cfg = transformers.HeliumConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, max_position_embeddings=128,
        bos_token_id=0, eos_token_id=1, tie_word_embeddings=False,
        attn_implementation="eager")
    d = hf_cfg.to_dict()
    d["architectures"] = ["HeliumForCausalLM"]
    hf = transformers.HeliumForCausalLM(hf_cfg).eval()
    mine = LlamaForCausalLM(LlamaForCausalLM.config_from_hf(d)).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_exaone4():
    """EXAONE-4: POST-norm residual blocks + per-head qk-norm + hybrid
    sliding(rope)/full(NoPE) attention layers."""
    from automodel_amd.models.exaone4.model import Exaone4ForCausalLM

    torch.manual_seed(98)
    hf_cfg = transformers.Exaone4Config(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, max_position_embeddings=128,
        sliding_window=8, sliding_window_pattern=2,
        bos_token_id=0, eos_token_id=1, tie_word_embeddings=False,
        attn_implementation="eager")
    assert "full_attention" in hf_cfg.layer_types, hf_cfg.layer_types
    hf = transformers.Exaone4ForCausalLM(hf_cfg).eval()
    mine = Exaone4ForCausalLM(
        Exaone4ForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 24))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_seed_oss():
    """Seed-OSS: llama-shaped with biased qkv (unbiased o_proj) and explicit
    head_dim — rides the llama family directly."""
    from automodel_amd.models.llama.model import LlamaForCausalLM

    torch.manual_seed(99)
    hf_cfg = transformers.SeedOssConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, max_position_embeddings=128,
        bos_token_id=0, eos_token_id=1, tie_word_embeddings=False,
        attn_implementation="eager")
    d = hf_cfg.to_dict()
    d["architectures"] = ["SeedOssForCausalLM"]
    hf = transformers.SeedOssForCausalLM(hf_cfg).eval()
    mine = LlamaForCausalLM(LlamaForCausalLM.config_from_hf(d)).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


@pytest.mark.parametrize("arch", ["GlmForCausalLM", "Glm4ForCausalLM"])
def test_hf_logits_parity_glm(arch):
    """GLM-4 dense: fused gate_up SwiGLU + biased qkv + partial (0.5)
    pair-interleaved rotary; Glm4 adds sandwich sublayer norms."""
    from automodel_amd.models.glm.model import GlmForCausalLM

    torch.manual_seed(100)
    cfg_cls = transformers.GlmConfig if arch == "GlmForCausalLM" \
        else transformers.Glm4Config
    hf_cfg = cfg_cls(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, max_position_embeddings=128, pad_token_id=0,
        bos_token_id=0, eos_token_id=1, tie_word_embeddings=False,
        attn_implementation="eager")
    hf = getattr(transformers, arch)(hf_cfg).eval()
    d = hf_cfg.to_dict()
    d["architectures"] = [arch]
    mine = GlmForCausalLM(GlmForCausalLM.config_from_hf(d)).eval()
    assert mine.config.sandwich_norms == (arch == "Glm4ForCausalLM")
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_arcee():
    """Arcee/AFM: llama-shaped RMS pre-norm blocks with a GATELESS relu²
    MLP (down(relu(up(x))²))."""
    from automodel_amd.models.arcee.model import ArceeForCausalLM

    torch.manual_seed(101)
    hf_cfg = transformers.ArceeConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, max_position_embeddings=128,
        bos_token_id=0, eos_token_id=1, tie_word_embeddings=False,
        attn_implementation="eager")
    hf = transformers.ArceeForCausalLM(hf_cfg).eval()
    mine = ArceeForCausalLM(
        ArceeForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


@pytest.mark.parametrize("parallel", [True, False])
def test_hf_logits_parity_gpt_neox(parallel):
    """GPT-NeoX/Pythia: per-head fused qkv, partial rotary 0.25, biased
    LayerNorms, parallel (or sequential) attn+MLP residual."""
    from automodel_amd.models.gpt_neox.model import GPTNeoXForCausalLM

    torch.manual_seed(102)
    hf_cfg = transformers.GPTNeoXConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4,
        max_position_embeddings=128, use_parallel_residual=parallel,
        bos_token_id=0, eos_token_id=1, tie_word_embeddings=False,
        attn_implementation="eager")
    hf = transformers.GPTNeoXForCausalLM(hf_cfg).eval()
    mine = GPTNeoXForCausalLM(
        GPTNeoXForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_olmo_v1():
    """OLMo v1: non-parametric fp32 LayerNorms + clip_qkv clamp."""
    from automodel_amd.models.olmo.model import OlmoForCausalLM

    torch.manual_seed(103)
    hf_cfg = transformers.OlmoConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        clip_qkv=0.2, max_position_embeddings=128,
        bos_token_id=0, eos_token_id=1, tie_word_embeddings=False,
        attn_implementation="eager")
    hf = transformers.OlmoForCausalLM(hf_cfg).eval()
    mine = OlmoForCausalLM(OlmoForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_olmo3():
    """OLMo-3: olmo2-style post-norm sublayers + full-width qk-norm +
    hybrid sliding/full attention with per-kind rope tables."""
    from automodel_amd.models.olmo.model import Olmo3ForCausalLM

    torch.manual_seed(104)
    hf_cfg = transformers.Olmo3Config(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, sliding_window=8,
        bos_token_id=0, eos_token_id=1, tie_word_embeddings=False,
        attn_implementation="eager")
    assert "full_attention" in hf_cfg.layer_types
    hf = transformers.Olmo3ForCausalLM(hf_cfg).eval()
    mine = Olmo3ForCausalLM(Olmo3ForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 24))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=3e-4, rtol=2e-4)


@pytest.mark.parametrize("pre_ln", [True, False])
def test_hf_logits_parity_opt(pre_ln):
    """OPT: learned positions (+2 table offset), relu MLP, pre/post-LN."""
    from automodel_amd.models.opt.model import OPTForCausalLM

    torch.manual_seed(105)
    hf_cfg = transformers.OPTConfig(
        vocab_size=300, hidden_size=64, ffn_dim=128, num_hidden_layers=2,
        num_attention_heads=4, max_position_embeddings=128,
        do_layer_norm_before=pre_ln, word_embed_proj_dim=64,
        bos_token_id=0, eos_token_id=1, pad_token_id=2,
        tie_word_embeddings=False, attn_implementation="eager")
    hf = transformers.OPTForCausalLM(hf_cfg).eval()
    mine = OPTForCausalLM(OPTForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected and not missing, (missing, unexpected)
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


@pytest.mark.parametrize("n_head", [4, 6])
def test_hf_logits_parity_bloom(n_head):
    """BLOOM: ALiBi bias + per-head fused qkv + post-embedding LayerNorm.
    n_head=6 exercises the non-power-of-2 slope ladder."""
    from automodel_amd.models.bloom.model import BloomForCausalLM

    torch.manual_seed(106)
    hf_cfg = transformers.BloomConfig(
        vocab_size=300, hidden_size=48 if n_head == 6 else 64, n_layer=2,
        n_head=n_head, bos_token_id=0, eos_token_id=1,
        tie_word_embeddings=False, attn_implementation="eager")
    hf = transformers.BloomForCausalLM(hf_cfg).eval()
    mine = BloomForCausalLM(BloomForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("alibi" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


@pytest.mark.parametrize("n_heads", [4, 6])
def test_hf_logits_parity_mpt(n_heads):
    """MPT: MosaicML ALiBi ladder (odd/even reorder for non-pow2 heads),
    fused Wqkv, bias-free norms/projections, exact GELU."""
    from automodel_amd.models.mpt.model import MptForCausalLM

    torch.manual_seed(107)
    hf_cfg = transformers.MptConfig(
        vocab_size=300, d_model=48 if n_heads == 6 else 64, n_layers=2,
        n_heads=n_heads, tie_word_embeddings=False, attn_implementation="eager")
    hf = transformers.MptForCausalLM(hf_cfg).eval()
    mine = MptForCausalLM(MptForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("alibi" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_persimmon():
    """Persimmon: per-head fused qkv + per-head biased qk LayerNorms before
    rope + partial rotary 0.5 + relu² MLP."""
    from automodel_amd.models.persimmon.model import PersimmonForCausalLM

    torch.manual_seed(108)
    hf_cfg = transformers.PersimmonConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4,
        max_position_embeddings=128, bos_token_id=0, eos_token_id=1,
        tie_word_embeddings=False, attn_implementation="eager")
    hf = transformers.PersimmonForCausalLM(hf_cfg).eval()
    mine = PersimmonForCausalLM(
        PersimmonForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_gptj():
    """GPT-J: parallel attn+MLP over shared ln_1, partial pair-interleaved
    rotary (rotate_every_two), biased lm_head."""
    from automodel_amd.models.gptj.model import GPTJForCausalLM

    torch.manual_seed(109)
    hf_cfg = transformers.GPTJConfig(
        vocab_size=300, n_embd=64, n_layer=2, n_head=4, rotary_dim=8,
        n_positions=128, bos_token_id=0, eos_token_id=1,
        tie_word_embeddings=False, attn_implementation="eager")
    hf = transformers.GPTJForCausalLM(hf_cfg).eval()
    mine = GPTJForCausalLM(GPTJForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_ernie45_moe():
    """Ernie-4.5-MoE: interleaved rope + leading dense layers + softmax
    routing with selection-only correction bias (moe_statics) + stacked
    HF expert tensors + wide shared expert."""
    from automodel_amd.moe.model import MoEForCausalLM, MoEModelConfig
    from automodel_amd.moe.state_dict_adapter import MoEStateDictAdapter

    torch.manual_seed(110)
    hf_cfg = transformers.Ernie4_5_MoeConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        moe_intermediate_size=48, num_hidden_layers=3,
        num_attention_heads=4, num_key_value_heads=2,
        moe_num_experts=4, moe_k=2, moe_num_shared_experts=2,
        moe_layer_start_index=1, max_position_embeddings=128,
        bos_token_id=0, eos_token_id=1, pad_token_id=2,
        tie_word_embeddings=False, attn_implementation="eager")
    hf = transformers.Ernie4_5_MoeForCausalLM(hf_cfg).eval()
    for n, p in hf.named_parameters():
        if "e_score_correction_bias" in n:   # exercise biased selection
            p.data = torch.randn_like(p.data) * 0.5
    d = hf_cfg.to_dict()
    d["architectures"] = ["Ernie4_5_MoeForCausalLM"]
    mc = MoEModelConfig.from_hf_config(d)
    assert mc.rope_interleaved and mc.first_k_dense == 1
    mine = MoEForCausalLM(mc).eval().float()
    adapter = MoEStateDictAdapter(mc)
    missing, unexpected = mine.load_state_dict(
        adapter.from_hf(hf.state_dict()), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=3e-4, rtol=3e-4)
    # to_hf round-trips the stacked-expert + moe_statics layout exactly
    back = adapter.to_hf(mine.state_dict())
    assert set(hf.state_dict().keys()) - set(back.keys()) == set()


def test_hf_logits_parity_dbrx():
    """DBRX: norm_attn_norm blocks, clip-clamped fused Wqkv, softmax router
    with p-norm weight normalization, flattened expert tensors."""
    from automodel_amd.models.dbrx.model import DbrxForCausalLM

    torch.manual_seed(111)
    hf_cfg = transformers.DbrxConfig(
        d_model=64, n_heads=4, n_layers=2, max_seq_len=128, vocab_size=300,
        ffn_config=dict(ffn_hidden_size=48, moe_num_experts=4, moe_top_k=2),
        attn_config=dict(kv_n_heads=2, rope_theta=10000.0, clip_qkv=8.0),
        bos_token_id=0, eos_token_id=1, tie_word_embeddings=False,
        attn_implementation="eager")
    hf = transformers.DbrxForCausalLM(hf_cfg).eval()
    mine = DbrxForCausalLM(DbrxForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=3e-4, rtol=3e-4)


def test_hf_logits_parity_minimax():
    """MiniMax: hybrid lightning (block-decay linear) / full attention
    layers, post-LN weighted residuals, mixtral-class stacked MoE. Seq
    length spans multiple lightning blocks to exercise the KV-state
    recurrence."""
    from automodel_amd.models.minimax.model import MiniMaxForCausalLM

    torch.manual_seed(112)
    hf_cfg = transformers.MiniMaxConfig(
        vocab_size=300, hidden_size=64, intermediate_size=96,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        num_local_experts=4, num_experts_per_tok=2, block_size=8,
        max_position_embeddings=128, bos_token_id=0, eos_token_id=1,
        tie_word_embeddings=False, attn_implementation="eager")
    assert "linear_attention" in hf_cfg.layer_types
    hf = transformers.MiniMaxForCausalLM(hf_cfg).eval()
    mine = MiniMaxForCausalLM(
        MiniMaxForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 20))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=3e-4, rtol=3e-4)


def test_hf_logits_parity_hunyuan_moe():
    """HunYuan-MoE-v1: per-head qk-norm AFTER rope, fp32 router (gate.wg),
    stacked experts + always-on shared MLP."""
    from automodel_amd.models.hunyuan.model import HunYuanMoEV1ForCausalLM

    torch.manual_seed(113)
    hf_cfg = transformers.HunYuanMoEV1Config(
        vocab_size=300, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, num_experts=4, moe_topk=[2, 2], num_shared_expert=[1, 1],
        max_position_embeddings=128, bos_token_id=0, eos_token_id=1,
        tie_word_embeddings=False, attn_implementation="eager")
    hf = transformers.HunYuanMoEV1ForCausalLM(hf_cfg).eval()
    mine = HunYuanMoEV1ForCausalLM(
        HunYuanMoEV1ForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=3e-4, rtol=3e-4)


def test_hf_logits_parity_bitnet():
    """BitNet-b1.58 (bf16 master-weight path): llama skeleton + relu² gate
    + sub-norms before o_proj/down_proj."""
    from automodel_amd.models.bitnet.model import BitNetForCausalLM

    torch.manual_seed(114)
    hf_cfg = transformers.BitNetConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, bos_token_id=0, eos_token_id=1,
        tie_word_embeddings=False, attn_implementation="eager")
    hf = transformers.BitNetForCausalLM(hf_cfg).eval()
    mine = BitNetForCausalLM(BitNetForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_codegen():
    """CodeGen: GPT-J parallel-residual block with mp_num=4-grouped fused
    qkv in [q, v, k] slice order and partial interleaved rotary."""
    from automodel_amd.models.codegen.model import CodeGenForCausalLM

    torch.manual_seed(115)
    hf_cfg = transformers.CodeGenConfig(
        vocab_size=300, n_embd=64, n_layer=2, n_head=8, rotary_dim=4,
        n_positions=128, bos_token_id=0, eos_token_id=1,
        tie_word_embeddings=False, attn_implementation="eager")
    hf = transformers.CodeGenForCausalLM(hf_cfg).eval()
    mine = CodeGenForCausalLM(
        CodeGenForCausalLM.config_from_hf(hf_cfg.to_dict())).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=2e-4, rtol=2e-4)


def test_hf_logits_parity_hunyuan_dense():
    """HunYuan-Dense-V1: llama-shaped + per-head qk RMSNorm AFTER rope."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.registry import build_model

    hf_cfg = transformers.HunYuanDenseV1Config(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, max_position_embeddings=128, rope_theta=10000.0,
        attention_bias=False, tie_word_embeddings=False,
        attn_implementation="eager")
    torch.manual_seed(0)
    hf = transformers.HunYuanDenseV1ForCausalLM(hf_cfg).eval()
    from automodel_amd.models.llama.model import LlamaConfig, LlamaForCausalLM
    cfg = LlamaConfig.from_hf_config(
        {**hf_cfg.to_dict(), "architectures": ["HunYuanDenseV1ForCausalLM"]})
    mine = LlamaForCausalLM(cfg).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    # HF names the qk norms query_layernorm/key_layernorm
    remap = {k.replace("query_layernorm", "q_norm").replace("key_layernorm", "k_norm"): v
             for k, v in hf.state_dict().items()}
    missing, unexpected = mine.load_state_dict(remap, strict=False)
    assert not unexpected, unexpected
    ids = torch.randint(0, 300, (2, 24))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=3e-4, rtol=3e-4)


def test_hf_logits_parity_apertus():
    """Apertus: qk-norm before rope + gate-free xIELU MLP."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.llama.model import LlamaConfig, LlamaForCausalLM

    hf_cfg = transformers.ApertusConfig(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rope_theta=10000.0,
        tie_word_embeddings=False, attn_implementation="eager",
        hidden_act="xielu")
    torch.manual_seed(1)
    hf = transformers.ApertusForCausalLM(hf_cfg).eval().float()
    cfg = LlamaConfig.from_hf_config(
        {**hf_cfg.to_dict(), "architectures": ["ApertusForCausalLM"]})
    mine = LlamaForCausalLM(cfg).eval()
    # Apertus names its pre-norms attention_layernorm / feedforward_layernorm
    # (same pre-norm positions as llama's input/post_attention norms)
    remap = {k.replace("attention_layernorm", "input_layernorm")
              .replace("feedforward_layernorm", "post_attention_layernorm"): v
             for k, v in hf.state_dict().items()}
    missing, unexpected = mine.load_state_dict(remap, strict=False)
    assert not unexpected, unexpected
    assert all(("rope" in k) or ("beta" in k) or ("eps" in k) for k in missing), missing
    ids = torch.randint(0, 300, (2, 24))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=3e-4, rtol=3e-4)


def test_hunyuan_apertus_checkpoint_roundtrip(tmp_path):
    """The rename adapters make real-checkpoint load paths work: save an
    HF-keyed checkpoint, load via build_model(pretrained_path)."""
    import json
    import os

    transformers = pytest.importorskip("transformers")
    from safetensors.torch import save_file

    from automodel_amd.checkpoint.hf_loader import load_hf_weights
    from automodel_amd.models.registry import build_model

    torch.manual_seed(2)
    hf_cfg = transformers.HunYuanDenseV1Config(
        vocab_size=300, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, max_position_embeddings=128, rope_theta=10000.0,
        attention_bias=False, tie_word_embeddings=False,
        attn_implementation="eager")
    hf = transformers.HunYuanDenseV1ForCausalLM(hf_cfg).eval()
    ckpt = tmp_path / "hy"
    os.makedirs(ckpt)
    save_file({k: v.contiguous() for k, v in hf.state_dict().items()},
              str(ckpt / "model.safetensors"))
    with open(ckpt / "config.json", "w") as f:
        json.dump({**hf_cfg.to_dict(),
                   "architectures": ["HunYuanDenseV1ForCausalLM"]}, f)
    m = build_model(pretrained_path=str(ckpt), dtype="float32")
    m.init_weights(device="cpu")
    load_hf_weights(m, str(ckpt))
    ids = torch.randint(0, 300, (2, 16))
    with torch.no_grad():
        torch.testing.assert_close(m.eval()(ids), hf(ids).logits,
                                   atol=3e-4, rtol=3e-4)


def test_hf_logits_parity_llama4():
    """Llama4 text: interleaved MoE/dense layers, input-scaled sigmoid
    routing + shared expert, NoPE layers with temperature tuning, L2
    qk-norm after (interleaved) rope."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.llama4.model import Llama4Config, Llama4ForCausalLM

    hf_cfg = transformers.Llama4TextConfig(
        vocab_size=200, hidden_size=64, intermediate_size=128,
        intermediate_size_mlp=160, num_hidden_layers=4, num_attention_heads=4,
        num_key_value_heads=2, head_dim=16, num_local_experts=4,
        num_experts_per_tok=2, interleave_moe_layer_step=2,
        max_position_embeddings=128, rope_theta=10000.0,
        attn_implementation="eager", tie_word_embeddings=False)
    torch.manual_seed(5)
    hf = transformers.Llama4ForCausalLM(hf_cfg).eval()
    cfg = Llama4Config.from_hf_config(hf_cfg.to_dict())
    assert cfg.no_rope_layers == hf_cfg.no_rope_layers
    mine = Llama4ForCausalLM(cfg).eval()
    sd = mine.state_dict_adapter.from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 200, (2, 24))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=4e-4, rtol=4e-4)
    # export roundtrip through the adapter
    back = mine.state_dict_adapter.to_hf(mine.state_dict())
    for k, v in hf.state_dict().items():
        torch.testing.assert_close(back[k], v, atol=0, rtol=0)


def test_hf_logits_parity_nemotron_h():
    """Nemotron-H hybrid: Mamba2 chunked-SSD mixer + NoPE GQA attention +
    relu^2 MLP layers, pattern-driven; parity vs the HF torch fallback."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.nemotron_h.model import (
        NemotronHConfig,
        NemotronHForCausalLM,
    )

    hf_cfg = transformers.NemotronHConfig(
        vocab_size=200, hidden_size=64, intermediate_size=128,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, mamba_num_heads=4, mamba_head_dim=16, ssm_state_size=8,
        conv_kernel=4, n_groups=2, chunk_size=8,
        hybrid_override_pattern="M*M-", max_position_embeddings=128,
        attn_implementation="eager", tie_word_embeddings=False)
    torch.manual_seed(7)
    hf = transformers.NemotronHForCausalLM(hf_cfg).eval()
    cfg = NemotronHConfig.from_hf_config(hf_cfg.to_dict())
    mine = NemotronHForCausalLM(cfg).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert not missing, missing
    # seq length deliberately NOT a multiple of chunk_size (pad path)
    ids = torch.randint(0, 200, (2, 21))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=5e-4, rtol=5e-4)


def test_hf_logits_parity_bamba():
    """Bamba hybrid: shared Mamba2 mixer (full-dim gated norm) + partial-
    rotary GQA attention layers, per-layer SwiGLU FFN."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.bamba.model import BambaConfig, BambaForCausalLM

    hf_cfg = transformers.BambaConfig(
        vocab_size=200, hidden_size=64, mamba_expand=2, intermediate_size=160,
        num_hidden_layers=3, num_attention_heads=4, num_key_value_heads=2,
        mamba_n_heads=8, mamba_d_head=16, mamba_d_state=8, mamba_n_groups=1,
        mamba_chunk_size=8, attn_layer_indices=[1],
        max_position_embeddings=128, attn_implementation="eager",
        tie_word_embeddings=False)
    torch.manual_seed(9)
    hf = transformers.BambaForCausalLM(hf_cfg).eval()
    cfg = BambaConfig.from_hf_config(hf_cfg.to_dict())
    mine = BambaForCausalLM(cfg).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 200, (2, 21))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=5e-4, rtol=5e-4)


def test_hf_logits_parity_granitemoe_hybrid():
    """GraniteMoeHybrid (Granite 4.0): hybrid mamba/attention mixers +
    granite multipliers + NoPE + topk-then-softmax MoE + shared MLP."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.granitemoe_hybrid.model import (
        GraniteMoeHybridConfig,
        GraniteMoeHybridForCausalLM,
        GraniteMoeHybridStateDictAdapter,
    )

    hf_cfg = transformers.GraniteMoeHybridConfig(
        vocab_size=200, hidden_size=64, intermediate_size=48,
        num_hidden_layers=3, num_attention_heads=4, num_key_value_heads=2,
        num_local_experts=4, num_experts_per_tok=2, shared_intermediate_size=96,
        mamba_n_heads=8, mamba_d_head=16, mamba_d_state=8, mamba_n_groups=1,
        mamba_chunk_size=8, mamba_expand=2,
        layer_types=["mamba", "attention", "mamba"],
        attention_multiplier=0.25, residual_multiplier=0.8,
        embedding_multiplier=2.0, logits_scaling=1.5,
        position_embedding_type=None,
        max_position_embeddings=64, attn_implementation="eager",
        tie_word_embeddings=False)
    torch.manual_seed(10)
    hf = transformers.GraniteMoeHybridForCausalLM(hf_cfg).eval()
    cfg = GraniteMoeHybridConfig.from_hf_config(hf_cfg.to_dict())
    mine = GraniteMoeHybridForCausalLM(cfg).eval()
    sd = GraniteMoeHybridStateDictAdapter().from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 200, (2, 21))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=5e-4, rtol=5e-4)
    # adapter roundtrip
    rt = GraniteMoeHybridStateDictAdapter().to_hf(
        {k: v for k, v in mine.state_dict().items() if "rope" not in k})
    hf_sd = hf.state_dict()
    assert set(rt) == set(hf_sd)
    for k in rt:
        torch.testing.assert_close(rt[k], hf_sd[k])


def test_hf_logits_parity_falcon_h1():
    """FalconH1: parallel mamba+attention per layer, muP multipliers
    (key/attn-in-out/ssm-in-out/mlp/zxbcdt/embedding/lm-head) applied at
    runtime, silu-only mamba gate (mamba_rms_norm=False)."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.falcon_h1.model import (
        FalconH1Config,
        FalconH1ForCausalLM,
    )

    hf_cfg = transformers.FalconH1Config(
        vocab_size=200, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        mamba_n_heads=8, mamba_d_head=16, mamba_d_state=8, mamba_n_groups=2,
        mamba_chunk_size=8, mamba_expand=2, mamba_d_ssm=128,
        max_position_embeddings=64,
        embedding_multiplier=2.0, lm_head_multiplier=0.5, key_multiplier=1.5,
        attention_in_multiplier=0.9, attention_out_multiplier=1.1,
        ssm_in_multiplier=0.8, ssm_out_multiplier=1.2,
        mlp_multipliers=[1.3, 0.7], ssm_multipliers=[0.9, 1.1, 0.8, 1.2, 1.05],
        attn_implementation="eager", tie_word_embeddings=False)
    torch.manual_seed(11)
    hf = transformers.FalconH1ForCausalLM(hf_cfg).eval()
    cfg = FalconH1Config.from_hf_config(hf_cfg.to_dict())
    mine = FalconH1ForCausalLM(cfg).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 200, (2, 21))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=5e-4, rtol=5e-4)


def test_hf_logits_parity_falcon_h1_gated_norm():
    """FalconH1 with mamba_rms_norm=True: group norm BEFORE the silu gate."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.falcon_h1.model import (
        FalconH1Config,
        FalconH1ForCausalLM,
    )

    hf_cfg = transformers.FalconH1Config(
        vocab_size=200, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        mamba_n_heads=8, mamba_d_head=16, mamba_d_state=8, mamba_n_groups=2,
        mamba_chunk_size=8, mamba_expand=2, mamba_d_ssm=128,
        mamba_rms_norm=True, max_position_embeddings=64,
        attn_implementation="eager", tie_word_embeddings=False)
    torch.manual_seed(12)
    hf = transformers.FalconH1ForCausalLM(hf_cfg).eval()
    cfg = FalconH1Config.from_hf_config(hf_cfg.to_dict())
    mine = FalconH1ForCausalLM(cfg).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    ids = torch.randint(0, 200, (2, 17))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=5e-4, rtol=5e-4)


def test_hf_logits_parity_qwen3_next():
    """Qwen3-Next: GatedDeltaNet linear attention (chunked delta rule via
    triangular solve), gated GQA with partial rotary + zero-centered norms,
    softmax-topk MoE with sigmoid-gated shared expert."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.qwen3_next.model import (
        Qwen3NextConfig,
        Qwen3NextForCausalLM,
        Qwen3NextStateDictAdapter,
    )

    hf_cfg = transformers.Qwen3NextConfig(
        vocab_size=200, hidden_size=64, intermediate_size=96,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, linear_num_value_heads=4, linear_num_key_heads=2,
        linear_key_head_dim=8, linear_value_head_dim=16,
        linear_conv_kernel_dim=3, num_experts=4, num_experts_per_tok=2,
        moe_intermediate_size=32, shared_expert_intermediate_size=48,
        decoder_sparse_step=1, mlp_only_layers=[1],
        layer_types=["linear_attention", "full_attention",
                     "linear_attention", "full_attention"],
        max_position_embeddings=128, attn_implementation="eager",
        tie_word_embeddings=False)
    torch.manual_seed(13)
    hf = transformers.Qwen3NextForCausalLM(hf_cfg).eval()
    cfg = Qwen3NextConfig.from_hf_config(hf_cfg.to_dict())
    mine = Qwen3NextForCausalLM(cfg).eval()
    sd = Qwen3NextStateDictAdapter().from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 200, (2, 21))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=5e-4, rtol=5e-4)
    # multi-chunk sequence (chunk_size 64 -> 2 chunks + padding)
    ids_long = torch.randint(0, 200, (1, 120))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids_long), hf(ids_long).logits,
                                   atol=8e-4, rtol=8e-4)
    # adapter roundtrip
    rt = Qwen3NextStateDictAdapter().to_hf(
        {k: v for k, v in mine.state_dict().items() if "rope" not in k})
    hf_sd = hf.state_dict()
    assert set(rt) == set(hf_sd)
    for k in rt:
        torch.testing.assert_close(rt[k], hf_sd[k])


def test_hf_logits_parity_lfm2():
    """LFM2: gated short-conv layers (no activation) + q/k-normed GQA
    attention, auto-adjusted SwiGLU width (w1/w3/w2)."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.lfm2.model import Lfm2Config, Lfm2ForCausalLM

    hf_cfg = transformers.Lfm2Config(
        vocab_size=200, hidden_size=64, intermediate_size=128,
        num_hidden_layers=3, num_attention_heads=4, num_key_value_heads=2,
        conv_L_cache=3, layer_types=["conv", "full_attention", "conv"],
        max_position_embeddings=64, attn_implementation="eager",
        tie_word_embeddings=False)
    torch.manual_seed(14)
    hf = transformers.Lfm2ForCausalLM(hf_cfg).eval()
    cfg = Lfm2Config.from_hf_config(hf_cfg.to_dict())
    mine = Lfm2ForCausalLM(cfg).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 200, (2, 21))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=5e-4, rtol=5e-4)


def test_hf_logits_parity_jamba():
    """Jamba: Mamba-1 selective scan (chunked segsum) with dt/B/C norms,
    NoPE attention, period/offset-scheduled MoE (softmax-topk, no renorm)."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.jamba.model import (
        JambaConfig,
        JambaForCausalLM,
        JambaStateDictAdapter,
    )

    hf_cfg = transformers.JambaConfig(
        vocab_size=200, hidden_size=64, intermediate_size=128,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        mamba_d_state=8, mamba_d_conv=3, mamba_expand=2, mamba_dt_rank=8,
        num_experts=4, num_experts_per_tok=2,
        expert_layer_period=2, expert_layer_offset=1,
        attn_layer_period=4, attn_layer_offset=2,
        max_position_embeddings=64, attn_implementation="eager",
        use_mamba_kernels=False, tie_word_embeddings=False)
    torch.manual_seed(15)
    hf = transformers.JambaForCausalLM(hf_cfg).eval()
    cfg = JambaConfig.from_hf_config(hf_cfg.to_dict())
    mine = JambaForCausalLM(cfg).eval()
    sd = JambaStateDictAdapter().from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert not missing, missing
    ids = torch.randint(0, 200, (2, 37))   # crosses the chunk-16 boundary
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=5e-4, rtol=5e-4)


def test_hf_logits_parity_zamba2():
    """Zamba2: shared transformer blocks over concat(hidden, embeds) with
    per-use gate_up adapters, (head_dim/2)^-0.5 scale, Mamba2 backbone."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.zamba2.model import Zamba2Config, Zamba2ForCausalLM

    hf_cfg = transformers.Zamba2Config(
        vocab_size=200, hidden_size=64, num_hidden_layers=6,
        layers_block_type=["mamba", "mamba", "hybrid", "mamba", "mamba", "hybrid"],
        attention_head_dim=16, num_attention_heads=8, num_key_value_heads=8,
        mamba_d_state=8, mamba_d_conv=4, mamba_expand=2, mamba_ngroups=1,
        mamba_headdim=16, intermediate_size=128, adapter_rank=8,
        num_mem_blocks=2, max_position_embeddings=64,
        attn_implementation="eager", tie_word_embeddings=False)
    torch.manual_seed(16)
    hf = transformers.Zamba2ForCausalLM(hf_cfg).eval()
    cfg = Zamba2Config.from_hf_config(hf_cfg.to_dict())
    mine = Zamba2ForCausalLM(cfg).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert not missing, missing
    ids = torch.randint(0, 200, (2, 21))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=5e-4, rtol=5e-4)


def test_hf_logits_parity_zamba2_rope_shared_adapters():
    """Zamba2 with use_mem_rope + shared attention adapters + ONE mem block
    reused by both hybrid layers (true weight sharing)."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.zamba2.model import Zamba2Config, Zamba2ForCausalLM

    hf_cfg = transformers.Zamba2Config(
        vocab_size=200, hidden_size=64, num_hidden_layers=5,
        layers_block_type=["mamba", "hybrid", "mamba", "hybrid", "mamba"],
        attention_head_dim=16, num_attention_heads=8, num_key_value_heads=8,
        mamba_d_state=8, mamba_d_conv=4, mamba_expand=2, mamba_ngroups=1,
        mamba_headdim=16, intermediate_size=128, adapter_rank=8,
        num_mem_blocks=1, use_mem_rope=True, use_shared_attention_adapter=True,
        max_position_embeddings=64,
        attn_implementation="eager", tie_word_embeddings=False)
    torch.manual_seed(17)
    hf = transformers.Zamba2ForCausalLM(hf_cfg).eval()
    cfg = Zamba2Config.from_hf_config(hf_cfg.to_dict())
    mine = Zamba2ForCausalLM(cfg).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 200, (2, 17))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=5e-4, rtol=5e-4)


def test_zamba2_tie_shared_blocks():
    """tie_shared_blocks: hybrid layers of the same block_id share ONE
    module; per-use adapters survive grafting; forward stays finite."""
    from automodel_amd.models.zamba2.model import Zamba2Config, Zamba2ForCausalLM

    cfg = Zamba2Config(
        vocab_size=100, hidden_size=32, intermediate_size=64,
        num_hidden_layers=5,
        layers_block_type=["mamba", "hybrid", "mamba", "hybrid", "mamba"],
        num_attention_heads=4, num_key_value_heads=4, attention_head_dim=16,
        mamba_d_state=8, mamba_headdim=8, mamba_ngroups=1, adapter_rank=4,
        num_mem_blocks=1, max_position_embeddings=64, tie_word_embeddings=False)
    torch.manual_seed(0)
    m = Zamba2ForCausalLM(cfg)
    m.init_weights()
    l1, l3 = m.model.layers[1], m.model.layers[3]
    assert l1.shared_transformer is not l3.shared_transformer
    # make use-1's adapter distinguishable before tying
    a1 = l3.shared_transformer.feed_forward.gate_up_proj_adapter_list[1]
    assert not isinstance(a1, torch.nn.Identity)
    ties = m.tie_shared_blocks()
    assert ties == 1
    assert l1.shared_transformer is l3.shared_transformer
    assert l1.shared_transformer.feed_forward.gate_up_proj_adapter_list[1] is a1
    ids = torch.randint(0, 100, (1, 12))
    with torch.no_grad():
        out = m(ids)
    assert torch.isfinite(out).all()


@pytest.mark.parametrize("kind", ["mamba", "mamba2", "falcon_mamba"])
def test_hf_logits_parity_pure_mamba(kind):
    """Pure SSM LMs (Mamba / Mamba2 / FalconMamba) on the shared mixers;
    FalconMamba exercises the weightless dt/B/C stabilizer norms."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.mamba_lm.model import (
        FalconMambaForCausalLM,
        Mamba2ForCausalLM,
        MambaForCausalLM,
        MambaLMConfig,
    )

    torch.manual_seed(18)
    if kind == "mamba2":
        hf_cfg = transformers.Mamba2Config(
            vocab_size=200, hidden_size=64, num_hidden_layers=2, state_size=8,
            num_heads=8, head_dim=16, n_groups=1, expand=2, conv_kernel=4,
            chunk_size=8, tie_word_embeddings=False)
        hf = transformers.Mamba2ForCausalLM(hf_cfg).eval()
        mine = Mamba2ForCausalLM(MambaLMConfig.from_hf_config(hf_cfg.to_dict(),
                                                              "mamba2")).eval()
    elif kind == "mamba":
        hf_cfg = transformers.MambaConfig(
            vocab_size=200, hidden_size=64, num_hidden_layers=2, state_size=8,
            expand=2, conv_kernel=4, time_step_rank=8, tie_word_embeddings=False,
            use_mambapy=False)
        hf = transformers.MambaForCausalLM(hf_cfg).eval()
        mine = MambaForCausalLM(MambaLMConfig.from_hf_config(hf_cfg.to_dict(),
                                                             "mamba")).eval()
    else:
        hf_cfg = transformers.FalconMambaConfig(
            vocab_size=200, hidden_size=64, num_hidden_layers=2, state_size=8,
            expand=2, conv_kernel=4, time_step_rank=8, tie_word_embeddings=False)
        hf = transformers.FalconMambaForCausalLM(hf_cfg).eval()
        mine = FalconMambaForCausalLM(
            MambaLMConfig.from_hf_config(hf_cfg.to_dict(), "falcon_mamba")).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert not missing, missing
    ids = torch.randint(0, 200, (2, 21))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=5e-4, rtol=5e-4)


def test_hf_logits_parity_granitemoe_shared():
    """GraniteMoeShared: all-attention granite MoE + shared MLP (rides the
    GraniteMoeHybrid module with rope on)."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.granitemoe_hybrid.model import (
        GraniteMoeHybridConfig,
        GraniteMoeHybridForCausalLM,
        GraniteMoeHybridStateDictAdapter,
    )

    hf_cfg = transformers.GraniteMoeSharedConfig(
        vocab_size=200, hidden_size=64, intermediate_size=48,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        num_local_experts=4, num_experts_per_tok=2, shared_intermediate_size=96,
        attention_multiplier=0.3, residual_multiplier=0.9,
        embedding_multiplier=1.5, logits_scaling=2.0,
        max_position_embeddings=64, attn_implementation="eager",
        tie_word_embeddings=False)
    torch.manual_seed(19)
    hf = transformers.GraniteMoeSharedForCausalLM(hf_cfg).eval()
    d = hf_cfg.to_dict()
    d["architectures"] = ["GraniteMoeSharedForCausalLM"]
    mine = GraniteMoeHybridForCausalLM(
        GraniteMoeHybridConfig.from_hf_config(d)).eval()
    sd = GraniteMoeHybridStateDictAdapter().from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 200, (2, 17))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=5e-4, rtol=5e-4)


def test_hf_logits_parity_recurrent_gemma():
    """RecurrentGemma (Griffin): RG-LRU blocks (block-diagonal gates,
    sqrt(1-a^2) normalization, chunked segsum scan), windowed partial-rotary
    attention, biased GeGLU at intermediate//2, bf16 sqrt(H) normalizer,
    tanh logits soft-cap."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.recurrent_gemma.model import (
        RecurrentGemmaConfig,
        RecurrentGemmaForCausalLM,
    )

    hf_cfg = transformers.RecurrentGemmaConfig(
        vocab_size=200, hidden_size=64, intermediate_size=128,
        num_hidden_layers=3, num_attention_heads=4, num_key_value_heads=1,
        head_dim=16, lru_width=64, conv1d_width=3, attention_window_size=64,
        block_types=("recurrent", "recurrent", "attention"),
        max_position_embeddings=128, attn_implementation="eager",
        tie_word_embeddings=False)
    torch.manual_seed(20)
    hf = transformers.RecurrentGemmaForCausalLM(hf_cfg).eval()
    cfg = RecurrentGemmaConfig.from_hf_config(hf_cfg.to_dict())
    mine = RecurrentGemmaForCausalLM(cfg).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 200, (2, 77))   # crosses the scan chunk boundary
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=5e-4, rtol=5e-4)


def test_hf_logits_parity_phimoe():
    """Phimoe (Phi-3.5-MoE): LayerNorm pre-norms + SparseMixer-v2 routing
    (threshold-banded argmax top-2, eval path)."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.phimoe.model import (
        PhimoeConfig,
        PhimoeForCausalLM,
        PhimoeStateDictAdapter,
    )

    hf_cfg = transformers.PhimoeConfig(
        vocab_size=200, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        num_local_experts=4, num_experts_per_tok=2, attention_bias=True,
        lm_head_bias=True, max_position_embeddings=64,
        attn_implementation="eager", tie_word_embeddings=False)
    torch.manual_seed(21)
    hf = transformers.PhimoeForCausalLM(hf_cfg).eval()
    cfg = PhimoeConfig.from_hf_config(hf_cfg.to_dict())
    mine = PhimoeForCausalLM(cfg).eval()
    sd = PhimoeStateDictAdapter().from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 200, (2, 21))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=5e-4, rtol=5e-4)
    # training path runs and produces finite sparse-mixer gradients
    mine.train()
    loss = mine(ids, labels=ids.clone())
    loss.backward()
    g = mine.model.layers[0].mlp.router.weight.grad
    assert g is not None and torch.isfinite(g).all()


def test_hf_logits_parity_gpt_bigcode():
    """GPTBigCode (StarCoder-1): fused MQA c_attn, learned positions,
    tanh-GELU, biased LayerNorms."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.gpt_bigcode.model import (
        GPTBigCodeConfig,
        GPTBigCodeForCausalLM,
    )

    hf_cfg = transformers.GPTBigCodeConfig(
        vocab_size=200, n_embd=64, n_layer=2, n_head=4, n_positions=64,
        multi_query=True, bos_token_id=0, eos_token_id=0,
        attn_implementation="eager", tie_word_embeddings=False)
    torch.manual_seed(22)
    hf = transformers.GPTBigCodeForCausalLM(hf_cfg).eval()
    cfg = GPTBigCodeConfig.from_hf_config(hf_cfg.to_dict())
    mine = GPTBigCodeForCausalLM(cfg).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert not missing, missing
    ids = torch.randint(0, 200, (2, 21))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=5e-4, rtol=5e-4)


def test_hf_logits_parity_cohere2():
    """Cohere2 (Command-R7B): sliding-window rope layers interleaved with
    NoPE full-attention, parallel residual, logit scale."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.cohere.model import Cohere2ForCausalLM, CohereConfig

    hf_cfg = transformers.Cohere2Config(
        vocab_size=200, hidden_size=64, intermediate_size=96,
        num_hidden_layers=5, num_attention_heads=4, num_key_value_heads=2,
        sliding_window=8, max_position_embeddings=64, eos_token_id=0,
        attn_implementation="eager", tie_word_embeddings=False)
    torch.manual_seed(23)
    hf = transformers.Cohere2ForCausalLM(hf_cfg).eval()
    cfg = CohereConfig.from_hf_config(hf_cfg.to_dict())
    assert "full_attention" in cfg.layer_types
    mine = Cohere2ForCausalLM(cfg).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 200, (2, 21))   # S=21 > window=8 exercises the band
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=5e-4, rtol=5e-4)


def test_hf_logits_parity_minimax_m2():
    """MiniMax-M2: sigmoid routing + aux-free correction bias + top-k
    renorm + full-width q/k norms (generic MoE model flavor)."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.moe.model import MoEForCausalLM, MoEModelConfig
    from automodel_amd.moe.state_dict_adapter import MoEStateDictAdapter

    hf_cfg = transformers.MiniMaxM2Config(
        vocab_size=200, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, num_local_experts=4, num_experts_per_tok=2,
        bos_token_id=0, eos_token_id=0, max_position_embeddings=64,
        attn_implementation="eager", tie_word_embeddings=False)
    torch.manual_seed(24)
    hf = transformers.MiniMaxM2ForCausalLM(hf_cfg).eval()
    # make the correction bias non-trivial so selection-vs-weight asymmetry
    # is exercised
    with torch.no_grad():
        for lyr in hf.model.layers:
            lyr.mlp.e_score_correction_bias.uniform_(-0.5, 0.5)
    d = hf_cfg.to_dict()
    d["architectures"] = ["MiniMaxM2ForCausalLM"]
    cfg = MoEModelConfig.from_hf_config(d)
    assert cfg.moe.score_func == "sigmoid" and cfg.moe.expert_bias
    mine = MoEForCausalLM(cfg).eval()
    sd = MoEStateDictAdapter(cfg).from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    ids = torch.randint(0, 200, (2, 17))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=5e-4, rtol=5e-4)


def test_hf_logits_parity_qwen3_vl():
    """Qwen3-VL: DeepStack ViT (learned pos-table bilinear resample in
    merge order, 2-axis rotary, per-image full attention, postshuffle-norm
    deepstack mergers into early LLM layers) + interleaved 3D MRoPE text.
    Text-only AND image parity."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.qwen3_vl.model import (
        Qwen3VLConfig,
        Qwen3VLForConditionalGeneration,
    )

    tcfg = dict(vocab_size=200, hidden_size=64, intermediate_size=96,
                num_hidden_layers=3, num_attention_heads=4,
                num_key_value_heads=2, head_dim=16,
                rope_scaling={"rope_type": "default", "mrope_section": [4, 2, 2]},
                max_position_embeddings=128)
    vcfg = dict(depth=2, hidden_size=32, intermediate_size=64, num_heads=2,
                patch_size=4, temporal_patch_size=1, spatial_merge_size=2,
                out_hidden_size=64, num_position_embeddings=36,
                deepstack_visual_indexes=[0, 1])
    hf_cfg = transformers.Qwen3VLConfig(
        text_config=tcfg, vision_config=vcfg, image_token_id=3,
        attn_implementation="eager", tie_word_embeddings=False)
    torch.manual_seed(25)
    hf = transformers.Qwen3VLForConditionalGeneration(hf_cfg).eval()
    cfg = Qwen3VLConfig.from_hf_config(hf_cfg.to_dict())
    mine = Qwen3VLForConditionalGeneration(cfg).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert not missing, missing

    # text-only parity
    ids = torch.randint(5, 200, (2, 15))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=8e-4, rtol=8e-4)

    # one 8x8 image -> grid (1, 2, 2) after patch 4: 4 patches -> 1 merged tok
    grid = torch.tensor([[1, 4, 4]])   # 16 patches -> 4 merged tokens
    n_patches = 16
    pixels = torch.randn(n_patches, 3 * 1 * 4 * 4)
    ids = torch.randint(5, 200, (1, 18))
    ids[0, 6:10] = 3                    # 4 merged-image slots
    mm_type = (ids == 3).to(torch.int32)     # text 0, image 1
    with torch.no_grad():
        ref = hf(ids, pixel_values=pixels, image_grid_thw=grid,
                 mm_token_type_ids=mm_type).logits
        out = mine(ids, pixel_values=pixels, image_grid_thw=grid)
    torch.testing.assert_close(out, ref, atol=8e-4, rtol=8e-4)


def test_hf_logits_parity_qwen3_vl_moe():
    """Qwen3-VL-MoE: DeepStack vision + MoE text FFNs (softmax-topk renorm,
    fused experts split by the adapter)."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.qwen3_vl.model import (
        Qwen3VLConfig,
        Qwen3VLMoeForConditionalGeneration,
        Qwen3VLMoeStateDictAdapter,
    )

    tcfg = dict(vocab_size=200, hidden_size=64, intermediate_size=96,
                num_hidden_layers=2, num_attention_heads=4,
                num_key_value_heads=2, head_dim=16,
                num_experts=4, num_experts_per_tok=2, moe_intermediate_size=32,
                rope_scaling={"rope_type": "default", "mrope_section": [4, 2, 2]},
                max_position_embeddings=128)
    vcfg = dict(depth=2, hidden_size=32, intermediate_size=64, num_heads=2,
                patch_size=4, temporal_patch_size=1, spatial_merge_size=2,
                out_hidden_size=64, num_position_embeddings=36,
                deepstack_visual_indexes=[0, 1])
    hf_cfg = transformers.Qwen3VLMoeConfig(
        text_config=tcfg, vision_config=vcfg, image_token_id=3,
        attn_implementation="eager", tie_word_embeddings=False)
    torch.manual_seed(26)
    hf = transformers.Qwen3VLMoeForConditionalGeneration(hf_cfg).eval()
    cfg = Qwen3VLConfig.from_hf_config(hf_cfg.to_dict())
    assert cfg.text.num_experts == 4
    mine = Qwen3VLMoeForConditionalGeneration(cfg).eval()
    sd = Qwen3VLMoeStateDictAdapter().from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert not missing, missing
    grid = torch.tensor([[1, 4, 4]])
    pixels = torch.randn(16, 3 * 1 * 4 * 4)
    ids = torch.randint(5, 200, (1, 18))
    ids[0, 6:10] = 3
    mm_type = (ids == 3).to(torch.int32)
    with torch.no_grad():
        ref = hf(ids, pixel_values=pixels, image_grid_thw=grid,
                 mm_token_type_ids=mm_type).logits
        out = mine(ids, pixel_values=pixels, image_grid_thw=grid)
    torch.testing.assert_close(out, ref, atol=8e-4, rtol=8e-4)


def test_hf_logits_parity_glm4v():
    """GLM-4V: EVA-style ViT (grid_sample bicubic pos resample, RMS norms,
    SwiGLU vision MLP, Conv2d merge downsample, GELU/SwiGLU merger) + GLM-4
    text (sandwich norms, interleaved partial rope) on chunked 3D MRoPE."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.glm4v.model import (
        Glm4vConfig,
        Glm4vForConditionalGeneration,
    )

    tcfg = dict(vocab_size=200, hidden_size=64, intermediate_size=96,
                num_hidden_layers=2, num_attention_heads=4,
                num_key_value_heads=2,
                rope_scaling={"rope_type": "default",
                              "partial_rotary_factor": 0.5,
                              "mrope_section": [2, 1, 1]},
                max_position_embeddings=128)
    vcfg = dict(depth=2, hidden_size=32, intermediate_size=64, num_heads=2,
                patch_size=4, temporal_patch_size=1, spatial_merge_size=2,
                out_hidden_size=64, image_size=16)
    hf_cfg = transformers.Glm4vConfig(
        text_config=tcfg, vision_config=vcfg, image_token_id=3,
        attn_implementation="eager", tie_word_embeddings=False)
    torch.manual_seed(27)
    hf = transformers.Glm4vForConditionalGeneration(hf_cfg).eval()
    cfg = Glm4vConfig.from_hf_config(hf_cfg.to_dict())
    mine = Glm4vForConditionalGeneration(cfg).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert not missing, missing

    ids = torch.randint(5, 200, (2, 15))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=8e-4, rtol=8e-4)

    grid = torch.tensor([[1, 4, 4]])
    pixels = torch.randn(16, 3 * 1 * 4 * 4)
    ids = torch.randint(5, 200, (1, 18))
    ids[0, 6:10] = 3
    mm_type = (ids == 3).to(torch.int32)
    with torch.no_grad():
        ref = hf(ids, pixel_values=pixels, image_grid_thw=grid,
                 mm_token_type_ids=mm_type).logits
        out = mine(ids, pixel_values=pixels, image_grid_thw=grid)
    torch.testing.assert_close(out, ref, atol=8e-4, rtol=8e-4)


def test_hf_logits_parity_qwen2_audio():
    """Qwen2-Audio: Whisper-style encoder (conv+stride2, pre-LN, avg-pool 2)
    + linear projector splicing audio embeddings at audio_token_id."""
    transformers = pytest.importorskip("transformers")
    from automodel_amd.models.qwen2_audio.model import (
        Qwen2AudioConfig,
        Qwen2AudioForConditionalGeneration,
    )

    acfg = dict(d_model=32, encoder_layers=2, encoder_attention_heads=2,
                encoder_ffn_dim=48, num_mel_bins=16, max_source_positions=10)
    tcfg = dict(model_type="qwen2", vocab_size=120, hidden_size=64,
                intermediate_size=96, num_hidden_layers=2,
                num_attention_heads=4, num_key_value_heads=2,
                max_position_embeddings=128)
    hf_cfg = transformers.Qwen2AudioConfig(
        audio_config=acfg, text_config=tcfg, audio_token_id=3,
        attn_implementation="eager")
    torch.manual_seed(28)
    hf = transformers.Qwen2AudioForConditionalGeneration(hf_cfg).eval()
    cfg = Qwen2AudioConfig.from_hf_config(hf_cfg.to_dict())
    mine = Qwen2AudioForConditionalGeneration(cfg).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing

    # mel input padded to max_source_positions * 2 = 20 -> conv2 stride 2 ->
    # 10 frames -> avg pool 2 -> 5 audio tokens
    feats = torch.randn(1, 16, 20)
    ids = torch.randint(5, 120, (1, 14))
    ids[0, 3:8] = 3
    fmask = torch.ones(1, 20, dtype=torch.long)
    with torch.no_grad():
        ref = hf(ids, input_features=feats, feature_attention_mask=fmask).logits
        out = mine(ids, input_features=feats)
    torch.testing.assert_close(out, ref, atol=8e-4, rtol=8e-4)
    # text-only
    ids = torch.randint(5, 120, (2, 11))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits, atol=8e-4, rtol=8e-4)


def test_hf_logits_parity_qwen2_5_omni_thinker():
    """Qwen2.5-Omni thinker: windowed Whisper-style audio encoder (chunked
    attention, sinusoidal positions, per-sample pooling), the Qwen2.5-VL
    windowed ViT with separate q/k/v, and TMRoPE position assignment —
    text-only, +audio, and +audio+image paths."""
    from automodel_amd.models.qwen2_5_omni.model import (
        Qwen2_5OmniThinkerConfig,
        Qwen2_5OmniThinkerForConditionalGeneration,
    )

    torch.manual_seed(60)
    hf_cfg = transformers.Qwen2_5OmniThinkerConfig(
        audio_config=dict(d_model=32, encoder_layers=2,
                          encoder_attention_heads=2, encoder_ffn_dim=48,
                          num_mel_bins=16, max_source_positions=16,
                          n_window=4, output_dim=64),
        vision_config=dict(hidden_size=32, depth=2, num_heads=2,
                           out_hidden_size=64, patch_size=4,
                           temporal_patch_size=2, spatial_merge_size=2,
                           in_channels=3, intermediate_size=64,
                           window_size=16, fullatt_block_indexes=[1]),
        text_config=dict(vocab_size=300, hidden_size=64, intermediate_size=128,
                         num_hidden_layers=2, num_attention_heads=4,
                         num_key_value_heads=2, max_position_embeddings=256,
                         rms_norm_eps=1e-6,
                         rope_parameters={"rope_type": "default",
                                          "rope_theta": 10000.0,
                                          "mrope_section": [2, 3, 3]},
                         tie_word_embeddings=False),
        audio_token_index=3, image_token_index=4, video_token_index=5,
        vision_start_token_id=298, audio_start_token_id=297,
        attn_implementation="eager")
    hf = transformers.Qwen2_5OmniThinkerForConditionalGeneration(hf_cfg).eval()
    cfg = Qwen2_5OmniThinkerConfig.from_hf_config(hf_cfg.to_dict())
    assert cfg.audio_token_id == 3 and cfg.image_token_id == 4
    mine = Qwen2_5OmniThinkerForConditionalGeneration(cfg).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k or "rot_inv" in k or "positional_embedding" in k
               for k in missing), missing

    # ---- text-only
    ids = torch.randint(6, 290, (2, 15))
    with torch.no_grad():
        ref = hf(input_ids=ids, attention_mask=torch.ones_like(ids)).logits
        torch.testing.assert_close(mine(ids), ref, atol=8e-4, rtol=8e-4)

    # ---- audio: 20 mel frames -> chunks [8,8,4] -> 10 post-conv -> 5 tokens
    feats = torch.randn(1, 16, 20)
    fmask = torch.ones(1, 20, dtype=torch.long)
    seq = torch.cat([torch.randint(6, 290, (1, 3)), torch.tensor([[297]]),
                     torch.full((1, 5), 3), torch.randint(6, 290, (1, 4))], dim=1)
    with torch.no_grad():
        ref = hf(input_ids=seq, input_features=feats,
                 feature_attention_mask=fmask,
                 attention_mask=torch.ones_like(seq)).logits
        out = mine(seq, input_features=feats, feature_attention_mask=fmask)
    torch.testing.assert_close(out, ref, atol=8e-4, rtol=8e-4)

    # ---- audio + image in one sequence (TMRoPE interleaved assignment)
    grid = torch.tensor([[1, 8, 8]])
    pix = torch.randn(64, 3 * 2 * 4 * 4)
    seq = torch.cat([torch.randint(6, 290, (1, 2)), torch.tensor([[297]]),
                     torch.full((1, 5), 3), torch.randint(6, 290, (1, 2)),
                     torch.tensor([[298]]), torch.full((1, 16), 4),
                     torch.randint(6, 290, (1, 3))], dim=1)
    with torch.no_grad():
        ref = hf(input_ids=seq, input_features=feats,
                 feature_attention_mask=fmask, pixel_values=pix,
                 image_grid_thw=grid,
                 attention_mask=torch.ones_like(seq)).logits
        out = mine(seq, input_features=feats, feature_attention_mask=fmask,
                   pixel_values=pix, image_grid_thw=grid)
    torch.testing.assert_close(out, ref, atol=8e-4, rtol=8e-4)


def test_hf_logits_parity_qwen3_omni_moe_thinker():
    """Qwen3-Omni-MoE thinker: AuT audio encoder (3x stride-2 Conv2d
    downsample, conv_out fold, merged inference windows), DeepStack ViT with
    omni merger naming, MoE text under interleaved MRoPE, float TMRoPE —
    text-only, +audio, +audio+image paths."""
    from automodel_amd.models.qwen3_omni_moe.model import (
        Qwen3OmniMoeConfig,
        Qwen3OmniMoeThinkerForConditionalGeneration,
    )
    from automodel_amd.models.qwen3_vl.model import Qwen3VLMoeStateDictAdapter

    torch.manual_seed(61)
    tcfg = dict(vocab_size=300, hidden_size=64, intermediate_size=96,
                num_hidden_layers=2, num_attention_heads=4,
                num_key_value_heads=2, head_dim=16,
                num_experts=4, num_experts_per_tok=2, moe_intermediate_size=32,
                rope_scaling={"rope_type": "default", "mrope_section": [4, 2, 2]},
                max_position_embeddings=256, tie_word_embeddings=False)
    vcfg = dict(depth=2, hidden_size=32, intermediate_size=64, num_heads=2,
                patch_size=4, temporal_patch_size=1, spatial_merge_size=2,
                out_hidden_size=64, num_position_embeddings=36,
                deepstack_visual_indexes=[0, 1])
    # n_window must stay 50: HF hardcodes 13 tokens per full chunk
    acfg = dict(d_model=32, encoder_layers=2, encoder_attention_heads=2,
                encoder_ffn_dim=48, num_mel_bins=16, max_source_positions=16,
                n_window=50, n_window_infer=200, conv_chunksize=2,
                downsample_hidden_size=8, output_dim=64)
    hf_cfg = transformers.Qwen3OmniMoeThinkerConfig(
        audio_config=acfg, vision_config=vcfg, text_config=tcfg,
        audio_token_id=3, image_token_id=4, video_token_id=5,
        vision_start_token_id=298, audio_start_token_id=297,
        attn_implementation="eager")
    hf = transformers.Qwen3OmniMoeThinkerForConditionalGeneration(hf_cfg).eval()
    cfg = Qwen3OmniMoeConfig.from_hf_config(hf_cfg.to_dict())
    assert cfg.text.num_experts == 4 and cfg.audio.n_window == 50
    mine = Qwen3OmniMoeThinkerForConditionalGeneration(cfg).eval()
    sd = Qwen3VLMoeStateDictAdapter().from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k or "rot_inv" in k or "inv_freq" in k
               or "positional_embedding" in k for k in missing), missing

    # ---- text-only
    ids = torch.randint(6, 290, (2, 12))
    with torch.no_grad():
        ref = hf(input_ids=ids, attention_mask=torch.ones_like(ids)).logits
        torch.testing.assert_close(mine(ids), ref, atol=8e-4, rtol=8e-4)

    # ---- audio only: 230 frames -> chunks [100,100,30] -> 13+13+4 = 30
    # audio tokens, attention windows of 26 (n_window_infer=200 -> 2 chunks)
    feats = torch.randn(1, 16, 230)
    fmask = torch.ones(1, 230, dtype=torch.long)
    seq = torch.cat([torch.randint(6, 290, (1, 3)), torch.tensor([[297]]),
                     torch.full((1, 30), 3), torch.randint(6, 290, (1, 4))], dim=1)
    with torch.no_grad():
        ref = hf(input_ids=seq, input_features=feats,
                 feature_attention_mask=fmask,
                 attention_mask=torch.ones_like(seq)).logits
        out = mine(seq, input_features=feats, feature_attention_mask=fmask)
    torch.testing.assert_close(out, ref, atol=8e-4, rtol=8e-4)

    # ---- audio + image (float TMRoPE + deepstack injection)
    grid = torch.tensor([[1, 4, 4]])
    pix = torch.randn(16, 3 * 1 * 4 * 4)
    seq = torch.cat([torch.randint(6, 290, (1, 2)), torch.tensor([[297]]),
                     torch.full((1, 30), 3), torch.randint(6, 290, (1, 2)),
                     torch.tensor([[298]]), torch.full((1, 4), 4),
                     torch.randint(6, 290, (1, 3))], dim=1)
    with torch.no_grad():
        ref = hf(input_ids=seq, input_features=feats,
                 feature_attention_mask=fmask, pixel_values=pix,
                 image_grid_thw=grid,
                 attention_mask=torch.ones_like(seq)).logits
        out = mine(seq, input_features=feats, feature_attention_mask=fmask,
                   pixel_values=pix, image_grid_thw=grid)
    torch.testing.assert_close(out, ref, atol=8e-4, rtol=8e-4)


def test_hf_logits_parity_llava_onevision():
    """LLaVA-OneVision: SigLIP tower (no head), anyres grid re-assembly with
    unpad + image_newline columns, video bilinear pooling, qwen2 decoder —
    text, anyres-image, and video paths."""
    from automodel_amd.models.llava_onevision.model import (
        LlavaOnevisionConfig,
        LlavaOnevisionForConditionalGeneration,
    )

    torch.manual_seed(62)
    hf_cfg = transformers.LlavaOnevisionConfig(
        vision_config=dict(model_type="siglip_vision_model", hidden_size=32,
                           intermediate_size=48, num_hidden_layers=2,
                           num_attention_heads=2, image_size=8, patch_size=4,
                           vision_use_head=False),
        text_config=dict(model_type="qwen2", vocab_size=300, hidden_size=64,
                         intermediate_size=96, num_hidden_layers=2,
                         num_attention_heads=4, num_key_value_heads=2,
                         max_position_embeddings=256, rope_theta=10000.0,
                         tie_word_embeddings=False),
        image_grid_pinpoints=[[8, 8], [8, 16], [16, 8], [16, 16]],
        image_token_index=3, video_token_index=4,
        vision_feature_layer=-1, vision_feature_select_strategy="full",
        vision_aspect_ratio="anyres_max_9", attn_implementation="eager")
    hf = transformers.LlavaOnevisionForConditionalGeneration(hf_cfg).eval()
    cfg = LlavaOnevisionConfig.from_hf_config(hf_cfg.to_dict())
    assert cfg.image_token_id == 3 and cfg.vision_feature_layer == -1
    mine = LlavaOnevisionForConditionalGeneration(cfg).eval()
    missing, unexpected = mine.load_state_dict(hf.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing

    # ---- text-only
    ids = torch.randint(6, 290, (2, 11))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(input_ids=ids).logits,
                                   atol=8e-4, rtol=8e-4)

    # ---- anyres image: 12x16 -> best (16,16) -> 2x2 grid + base = 5 crops,
    # base 4 tokens + 4x(4+1 newline) = 24 image tokens
    pix = torch.randn(1, 5, 3, 8, 8)
    sizes = torch.tensor([[12, 16]])
    seq = torch.cat([torch.randint(6, 290, (1, 3)), torch.full((1, 24), 3),
                     torch.randint(6, 290, (1, 4))], dim=1)
    with torch.no_grad():
        ref = hf(input_ids=seq, pixel_values=pix, image_sizes=sizes).logits
        out = mine(seq, pixel_values=pix, image_sizes=sizes)
    torch.testing.assert_close(out, ref, atol=8e-4, rtol=8e-4)

    # ---- video: 2 frames -> pooled 1 token/frame + 1 newline = 3 tokens
    vid = torch.randn(1, 2, 3, 8, 8)
    seq = torch.cat([torch.randint(6, 290, (1, 2)), torch.full((1, 3), 4),
                     torch.randint(6, 290, (1, 3))], dim=1)
    with torch.no_grad():
        ref = hf(input_ids=seq, pixel_values_videos=vid).logits
        out = mine(seq, pixel_values_videos=vid)
    torch.testing.assert_close(out, ref, atol=8e-4, rtol=8e-4)


def test_hf_logits_parity_glm4v_moe():
    """GLM-4.5V: GLM-4V ViT + GLM4-MoE text (sigmoid+bias routing, group
    top-k, shared expert, dense-first, plain pre/post norms, contiguous-half
    partial rotary) on chunked 3D MRoPE — text and image paths."""
    from automodel_amd.models.glm4v_moe.model import (
        Glm4vMoeConfig,
        Glm4vMoeForConditionalGeneration,
    )

    torch.manual_seed(63)
    tcfg = dict(vocab_size=200, hidden_size=64, intermediate_size=96,
                num_hidden_layers=3, num_attention_heads=4,
                num_key_value_heads=2, head_dim=16,
                n_routed_experts=8, n_shared_experts=1, num_experts_per_tok=2,
                moe_intermediate_size=48, first_k_dense_replace=1,
                norm_topk_prob=True, routed_scaling_factor=1.0,
                attention_bias=True,
                rope_scaling={"rope_type": "default",
                              "partial_rotary_factor": 0.5,
                              "mrope_section": [2, 1, 1]},
                max_position_embeddings=128, tie_word_embeddings=False)
    vcfg = dict(depth=2, hidden_size=32, intermediate_size=64, num_heads=2,
                patch_size=4, temporal_patch_size=1, spatial_merge_size=2,
                out_hidden_size=64, image_size=16)
    hf_cfg = transformers.Glm4vMoeConfig(
        text_config=tcfg, vision_config=vcfg, image_token_id=3,
        attn_implementation="eager")
    hf = transformers.Glm4vMoeForConditionalGeneration(hf_cfg).eval()
    cfg = Glm4vMoeConfig.from_hf_config(hf_cfg.to_dict())
    assert cfg.text.moe.n_routed_experts == 8 and cfg.text.attention_bias
    mine = Glm4vMoeForConditionalGeneration(cfg).eval()
    sd = mine.state_dict_adapter.from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k or "inv_freq" in k for k in missing), missing

    ids = torch.randint(5, 200, (2, 15))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits,
                                   atol=8e-4, rtol=8e-4)
    grid = torch.tensor([[1, 4, 4]])
    pixels = torch.randn(16, 3 * 1 * 4 * 4)
    ids = torch.randint(5, 200, (1, 18))
    ids[0, 6:10] = 3
    with torch.no_grad():
        ref = hf(ids, pixel_values=pixels, image_grid_thw=grid,
                 mm_token_type_ids=(ids == 3).int()).logits
        out = mine(ids, pixel_values=pixels, image_grid_thw=grid)
    torch.testing.assert_close(out, ref, atol=8e-4, rtol=8e-4)


def test_hf_logits_parity_lfm2_moe():
    """LFM2-MoE: the conv/attention hybrid with sigmoid + aux-free-bias
    routed MoE FFNs after num_dense_layers dense layers."""
    from automodel_amd.models.lfm2.model import Lfm2MoeConfig, Lfm2MoeForCausalLM

    torch.manual_seed(64)
    hf_cfg = transformers.Lfm2MoeConfig(
        vocab_size=200, hidden_size=64, intermediate_size=96,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        conv_L_cache=3,
        layer_types=["conv", "full_attention", "conv", "full_attention"],
        num_experts=8, num_experts_per_tok=2, moe_intermediate_size=32,
        num_dense_layers=1, use_expert_bias=True, norm_topk_prob=True,
        routed_scaling_factor=1.0, max_position_embeddings=64,
        attn_implementation="eager", tie_word_embeddings=False)
    hf = transformers.Lfm2MoeForCausalLM(hf_cfg).eval()
    # exercise non-zero aux-free bias routing too
    with torch.no_grad():
        for layer in hf.model.layers[1:]:
            layer.feed_forward.expert_bias.uniform_(-0.05, 0.05)
    cfg = Lfm2MoeConfig.from_hf_config(hf_cfg.to_dict())
    assert cfg.num_experts == 8 and cfg.num_dense_layers == 1
    mine = Lfm2MoeForCausalLM(cfg).eval()
    sd = mine.state_dict_adapter().from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 200, (2, 21))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits,
                                   atol=5e-4, rtol=5e-4)


def test_hf_logits_parity_dots1():
    """dots.llm1: DeepSeek-style sigmoid + aux-free-bias routing (shared
    expert, dense-first) on qwen3-style per-head qk-norm attention —
    rides the generic MoE model."""
    from automodel_amd.moe.model import MoEForCausalLM, MoEModelConfig

    torch.manual_seed(65)
    hf_cfg = transformers.Dots1Config(
        vocab_size=200, hidden_size=64, intermediate_size=96,
        num_hidden_layers=3, num_attention_heads=4, num_key_value_heads=2,
        n_routed_experts=8, n_shared_experts=1, num_experts_per_tok=2,
        moe_intermediate_size=32, first_k_dense_replace=1,
        norm_topk_prob=True, routed_scaling_factor=1.0,
        n_group=1, topk_group=1, max_position_embeddings=64,
        rope_theta=10000.0, attn_implementation="eager",
        tie_word_embeddings=False)
    hf = transformers.Dots1ForCausalLM(hf_cfg).eval()
    with torch.no_grad():   # exercise non-zero aux-free routing bias
        for layer in hf.model.layers[1:]:
            layer.mlp.gate.e_score_correction_bias.uniform_(-0.05, 0.05)
    d = hf_cfg.to_dict()
    d["architectures"] = ["Dots1ForCausalLM"]
    cfg = MoEModelConfig.from_hf_config(d)
    assert cfg.qk_norm and cfg.first_k_dense == 1
    assert cfg.moe.score_func == "sigmoid" and cfg.moe.n_shared_experts == 1
    mine = MoEForCausalLM(cfg).eval()
    sd = mine.state_dict_adapter.from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 200, (2, 17))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits,
                                   atol=5e-4, rtol=5e-4)


def test_hf_logits_parity_flex_olmo():
    """FlexOlmo: OLMo-2 POST-norm layout (post_attention/post_feedforward
    norms, full-width q/k norms) with softmax-topk MoE FFNs."""
    from automodel_amd.moe.model import MoEForCausalLM, MoEModelConfig

    torch.manual_seed(66)
    hf_cfg = transformers.FlexOlmoConfig(
        vocab_size=200, hidden_size=64, intermediate_size=96,
        num_hidden_layers=3, num_attention_heads=4, num_key_value_heads=2,
        num_experts=6, num_experts_per_tok=2, norm_topk_prob=False,
        max_position_embeddings=64, rope_theta=10000.0,
        pad_token_id=0, eos_token_id=1,
        attn_implementation="eager", tie_word_embeddings=False)
    hf = transformers.FlexOlmoForCausalLM(hf_cfg).eval()
    d = hf_cfg.to_dict()
    d["architectures"] = ["FlexOlmoForCausalLM"]
    cfg = MoEModelConfig.from_hf_config(d)
    assert cfg.olmo2_layout and cfg.qk_norm_full
    mine = MoEForCausalLM(cfg).eval()
    sd = mine.state_dict_adapter.from_hf(hf.state_dict())
    missing, unexpected = mine.load_state_dict(sd, strict=False)
    assert not unexpected, unexpected
    assert all("rope" in k for k in missing), missing
    ids = torch.randint(0, 200, (2, 19))
    with torch.no_grad():
        torch.testing.assert_close(mine(ids), hf(ids).logits,
                                   atol=5e-4, rtol=5e-4)
