"""DSA (DeepSeek sparse attention) numerics: lightning indexer, causal
top-k selection, gathered sparse attention vs the dense reference, and the
DeepSeek-V3.2 model integration (topk >= S reproduces DeepSeek-V3 exactly
on shared weights)."""

import torch
import torch.nn.functional as F

from automodel_amd.ops.sparse_attention import (
    indexer_kl_loss,
    lightning_index_scores,
    sparse_gather_attention,
    topk_causal_indices,
)


def test_topk_causal_indices_respects_causality():
    torch.manual_seed(0)
    scores = torch.randn(2, 9, 9)
    idx, valid = topk_causal_indices(scores, 4)
    assert idx.shape == (2, 9, 4)
    for t in range(9):
        assert (idx[:, t][valid[:, t]] <= t).all()
        assert int(valid[:, t].sum(-1).max()) == min(4, t + 1)


def test_sparse_full_topk_equals_dense():
    torch.manual_seed(1)
    B, S, H, D, Dv = 2, 12, 3, 16, 8
    q = torch.randn(B, S, H, D)
    k = torch.randn(B, S, H, D)
    v = torch.randn(B, S, H, Dv)
    scores = torch.randn(B, S, S)
    idx, valid = topk_causal_indices(scores, S)        # keep everything
    out = sparse_gather_attention(q, k, v, idx, valid)
    ref = F.scaled_dot_product_attention(
        q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
        is_causal=True).transpose(1, 2)
    torch.testing.assert_close(out, ref, atol=1e-5, rtol=1e-5)


def test_sparse_matches_masked_dense_at_small_topk():
    torch.manual_seed(2)
    B, S, H, D = 1, 10, 2, 8
    q = torch.randn(B, S, H, D)
    k = torch.randn(B, S, 1, D)        # MQA shared KV (MLA-style)
    v = torch.randn(B, S, 1, D)
    scores = torch.randn(B, S, S)
    K = 3
    idx, valid = topk_causal_indices(scores, K)
    out = sparse_gather_attention(q, k, v, idx, valid)
    # dense attention with a mask allowing only the selected keys
    allow = torch.zeros(B, S, S, dtype=torch.bool)
    for b in range(B):
        for t in range(S):
            allow[b, t, idx[b, t][valid[b, t]]] = True
    att = torch.einsum("bshd,bkhd->bhsk", q, k.expand(B, S, H, D)) * D ** -0.5
    att = att.masked_fill(~allow[:, None], float("-inf"))
    ref = torch.einsum("bhsk,bkhd->bshd", att.softmax(-1),
                       v.expand(B, S, H, D))
    torch.testing.assert_close(out, ref, atol=1e-5, rtol=1e-5)


def test_indexer_kl_zero_when_matched():
    torch.manual_seed(3)
    S = 8
    scores = torch.randn(1, S, S)
    causal = torch.ones(S, S, dtype=torch.bool).tril()
    probs = scores.masked_fill(~causal, float("-inf")).softmax(-1)
    loss = indexer_kl_loss(scores, probs)
    assert abs(float(loss)) < 1e-5
    # mismatched distributions give positive KL
    loss2 = indexer_kl_loss(torch.randn(1, S, S), probs)
    assert float(loss2) > 0.01


def _tiny_cfg(**kw):
    return dict(
        vocab_size=160, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4,
        q_lora_rank=32, kv_lora_rank=16, qk_nope_head_dim=16,
        qk_rope_head_dim=8, v_head_dim=16, first_k_dense_replace=1,
        max_position_embeddings=64,
        moe=dict(n_routed_experts=4, n_shared_experts=1,
                 n_activated_experts=2, moe_intermediate_size=32,
                 score_func="sigmoid", expert_bias=True, norm_topk_prob=True),
        **kw)


def test_dsv32_dense_fallback_matches_v3():
    """index_topk >= S: the DSA model must reproduce plain DeepSeek-V3."""
    from automodel_amd.models.deepseek_v3.model import DeepseekV3ForCausalLM
    from automodel_amd.models.deepseek_v32.model import DeepseekV32ForCausalLM

    torch.manual_seed(4)
    v3 = DeepseekV3ForCausalLM(_tiny_cfg()).eval()
    v3.init_weights()
    v32 = DeepseekV32ForCausalLM(_tiny_cfg(
        index_n_heads=2, index_head_dim=8, index_topk=64)).eval()
    v32.init_weights()
    missing, unexpected = v32.load_state_dict(v3.state_dict(), strict=False)
    assert not unexpected, unexpected
    assert all("indexer" in k for k in missing), missing
    ids = torch.randint(0, 160, (2, 12))
    with torch.no_grad():
        torch.testing.assert_close(v32(ids), v3(ids), atol=1e-5, rtol=1e-5)


def test_dsv32_sparse_path_and_indexer_loss():
    from automodel_amd.models.deepseek_v32.model import DeepseekV32ForCausalLM

    torch.manual_seed(5)
    model = DeepseekV32ForCausalLM(_tiny_cfg(
        index_n_heads=2, index_head_dim=8, index_topk=6))
    model.init_weights()
    model.set_collect_indexer_loss(True)
    ids = torch.randint(0, 160, (1, 16))       # S=16 > topk=6 -> sparse
    logits = model(ids)
    assert torch.isfinite(logits).all()
    kls = model.indexer_losses()
    assert len(kls) == 2 and all(torch.isfinite(k) for k in kls)
    # indexer receives gradient through the KL objective
    loss = sum(kls)
    loss.backward()
    g = model.model.layers[0].self_attn.indexer.wq.weight.grad
    assert g is not None and torch.isfinite(g).all() and g.abs().sum() > 0
