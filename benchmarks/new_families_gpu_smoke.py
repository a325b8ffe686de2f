"""One-forward GPU smoke for the round-2 final-session families (bf16)."""
import torch

def main():
    dev = "cuda"
    results = {}
    # Qwen2.5-Omni thinker
    from automodel_amd.models.qwen2_5_omni.model import (
        Qwen2_5OmniThinkerConfig, Qwen2_5OmniThinkerForConditionalGeneration)
    cfg = Qwen2_5OmniThinkerConfig(
        text=dict(vocab_size=320, hidden_size=64, intermediate_size=96,
                  num_hidden_layers=2, num_attention_heads=4,
                  num_key_value_heads=2, attention_bias=True,
                  max_position_embeddings=64, rope_theta=10000.0),
        audio=dict(d_model=32, encoder_layers=1, encoder_attention_heads=2,
                   encoder_ffn_dim=48, num_mel_bins=16, max_source_positions=8,
                   n_window=4, output_dim=64),
        vision=dict(variant="v2_5", qkv_separate=True, embed_dim=32, depth=1,
                    num_heads=2, intermediate_size=48, patch_size=4,
                    temporal_patch_size=2, spatial_merge_size=2,
                    hidden_size=64, window_size=16, fullatt_block_indexes=(0,)),
        mrope_section=(4, 2, 2), audio_token_id=3, image_token_id=4,
        vision_start_token_id=298, audio_start_token_id=297)
    m = Qwen2_5OmniThinkerForConditionalGeneration(cfg).to(dev, torch.bfloat16)
    m.init_weights(device=dev)
    m = m.to(torch.bfloat16)
    ids = torch.cat([torch.randint(6, 290, (1, 2)), torch.tensor([[297]]),
                     torch.full((1, 5), 3), torch.randint(6, 290, (1, 2)),
                     torch.tensor([[298]]), torch.full((1, 16), 4),
                     torch.randint(6, 290, (1, 2))], dim=1).to(dev)
    feats = torch.randn(1, 16, 20, device=dev)
    fmask = torch.ones(1, 20, dtype=torch.long, device=dev)
    pix = torch.randn(64, 3 * 2 * 4 * 4, device=dev)
    out = m(ids, input_features=feats, feature_attention_mask=fmask,
            pixel_values=pix, image_grid_thw=torch.tensor([[1, 8, 8]], device=dev))
    assert torch.isfinite(out.float()).all(); results["qwen2_5_omni"] = "ok"
    del m

    # DeepSeek-V3.2 sparse path (S > topk)
    from automodel_amd.models.deepseek_v32.model import DeepseekV32ForCausalLM
    v32 = DeepseekV32ForCausalLM(dict(
        vocab_size=160, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4,
        q_lora_rank=32, kv_lora_rank=16, qk_nope_head_dim=16,
        qk_rope_head_dim=8, v_head_dim=16, first_k_dense_replace=1,
        max_position_embeddings=64, index_n_heads=2, index_head_dim=8,
        index_topk=6,
        moe=dict(n_routed_experts=4, n_shared_experts=1,
                 n_activated_experts=2, moe_intermediate_size=32,
                 score_func="sigmoid", expert_bias=True, norm_topk_prob=True)))
    v32.init_weights(device=dev)
    v32 = v32.to(torch.bfloat16)
    ids = torch.randint(0, 160, (1, 16), device=dev)
    out = v32(ids)
    assert torch.isfinite(out.float()).all(); results["deepseek_v32_sparse"] = "ok"
    del v32

    # Kimi-Linear hybrid (KDA chunked on GPU)
    from automodel_amd.models.kimi_linear.model import (
        KimiLinearConfig, KimiLinearForCausalLM)
    kl = KimiLinearForCausalLM(KimiLinearConfig(
        vocab_size=120, hidden_size=48, intermediate_size=64,
        num_hidden_layers=4, linear_num_heads=2, linear_head_dim=8,
        linear_lowrank=8, full_attn_interval=4,
        num_attention_heads=2, kv_lora_rank=16, qk_nope_head_dim=8,
        qk_rope_head_dim=4, v_head_dim=8, first_k_dense_replace=1,
        max_position_embeddings=64,
        moe=dict(n_routed_experts=4, n_shared_experts=1,
                 n_activated_experts=2, moe_intermediate_size=16,
                 score_func="sigmoid", expert_bias=True, norm_topk_prob=True,
                 shared_expert_intermediate_size=16)))
    kl.init_weights(device=dev)
    kl = kl.to(torch.bfloat16)
    ids = torch.randint(0, 120, (2, 24), device=dev)
    loss = kl(ids, labels=ids.clone())
    loss.backward()
    assert torch.isfinite(loss.float()); results["kimi_linear_train"] = "ok"
    torch.cuda.synchronize()
    print("GPU_SMOKE_RESULTS", results)

if __name__ == "__main__":
    main()
