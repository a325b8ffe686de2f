"""DeepSeek Sparse Attention (DSA) primitives: lightning-indexer scoring,
causal top-k token selection, and gathered sparse attention.

Reference behavior: the reference framework's TileLang sparse-attention
stack (nemo_automodel components/models/deepseek_v4/ and glm_moe_dsa/ —
indexer + top-k sparse attention). Rebuilt
MI355X-native: the selection math is expressed as batched GEMMs + top-k,
which on gfx950 ride hipBLASLt / the in-tree flash kernels; the gathered
attention below is the exact torch reference the HIP path is tested
against (numerics tests compare fp32).

The DSA scheme (public DeepSeek-V3.2 architecture):

  * a lightweight "lightning indexer" scores every (query t, key s) pair:
    ``I[t, s] = sum_j w[t, j] * relu(qI[t, j] . kI[s])`` with ``H_I`` small
    query heads and ONE shared key per token — O(S^2 * d_I) with d_I tiny,
    far cheaper than main attention and fp8-friendly;
  * each query keeps only its ``top_k`` highest-scoring causal keys;
  * main (MLA) attention runs over the selected keys only.

Training-time indexer supervision: KL(main-attention distribution ||
indexer distribution) summed per query (``indexer_kl_loss``).
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

__all__ = [
    "lightning_index_scores",
    "topk_causal_indices",
    "sparse_gather_attention",
    "indexer_kl_loss",
]


def lightning_index_scores(q_index: torch.Tensor, k_index: torch.Tensor,
                           weights: torch.Tensor) -> torch.Tensor:
    """Index scores I[t, s].

    q_index: [B, S, H_I, D_I] per-head indexer queries
    k_index: [B, S, D_I] shared indexer keys
    weights: [B, S, H_I] per-query head weights
    -> [B, S, S] (query, key) scores (no causal mask applied here).
    """
    # relu(q . k): [B, S_q, H, S_k]
    logits = torch.einsum("bqhd,bkd->bqhk", q_index.float(), k_index.float())
    return torch.einsum("bqh,bqhk->bqk", weights.float(), logits.relu())


def topk_causal_indices(scores: torch.Tensor, top_k: int) -> tuple[torch.Tensor, torch.Tensor]:
    """Per-query causal top-k key selection.

    scores: [B, S, S] -> (indices [B, S, K], valid [B, S, K] bool) where
    K = min(top_k, S). Queries with fewer than K causal keys pad with
    duplicate index 0 marked invalid.
    """
    B, S, _ = scores.shape
    K = min(top_k, S)
    causal = torch.ones(S, S, dtype=torch.bool, device=scores.device).tril()
    masked = scores.masked_fill(~causal, float("-inf"))
    vals, idx = masked.topk(K, dim=-1)
    valid = vals > float("-inf")
    idx = torch.where(valid, idx, torch.zeros_like(idx))
    return idx, valid


def sparse_gather_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                            indices: torch.Tensor, valid: torch.Tensor,
                            scale: float | None = None) -> torch.Tensor:
    """Attention restricted to each query's selected keys (torch reference).

    q: [B, S, H, Dqk]   k: [B, S, Hkv, Dqk]   v: [B, S, Hkv, Dv]
    indices/valid: [B, S, K] from ``topk_causal_indices``
    -> [B, S, H, Dv]

    Hkv may be 1 (MQA/MLA shared-KV) or H.
    """
    B, S, H, Dq = q.shape
    K = indices.shape[-1]
    Hkv = k.shape[2]
    scale = Dq ** -0.5 if scale is None else scale
    # gather per-query keys/values: [B, S, K, Hkv, D]
    ie = indices.reshape(B, S * K, 1, 1).expand(-1, -1, Hkv, k.shape[-1])
    kg = k.gather(1, ie).view(B, S, K, Hkv, -1)
    ie = indices.reshape(B, S * K, 1, 1).expand(-1, -1, Hkv, v.shape[-1])
    vg = v.gather(1, ie).view(B, S, K, Hkv, -1)
    if Hkv == 1:
        kg = kg.expand(B, S, K, H, Dq)
        vg = vg.expand(B, S, K, H, vg.shape[-1])
    elif Hkv != H:
        rep = H // Hkv
        kg = kg.repeat_interleave(rep, dim=3)
        vg = vg.repeat_interleave(rep, dim=3)
    att = torch.einsum("bshd,bskhd->bshk", q.float(), kg.float()) * scale
    att = att.masked_fill(~valid[:, :, None, :], float("-inf"))
    p = att.softmax(dim=-1)
    return torch.einsum("bshk,bskhd->bshd", p, vg.float()).to(q.dtype)


def indexer_kl_loss(index_scores: torch.Tensor, attn_scores: torch.Tensor,
                    mask: torch.Tensor | None = None) -> torch.Tensor:
    """KL(main-attention distribution || indexer distribution), mean over
    queries — the DSA indexer's training signal.

    index_scores: [B, S, S] raw indexer scores
    attn_scores:  [B, S, S] head-summed main attention probabilities
                  (detached target)
    mask: [S, S] bool allowed positions (defaults to causal)
    """
    S = index_scores.shape[1]
    if mask is None:
        mask = torch.ones(S, S, dtype=torch.bool,
                          device=index_scores.device).tril()
    neg = torch.finfo(torch.float32).min
    logp = index_scores.float().masked_fill(~mask, neg).log_softmax(-1)
    target = attn_scores.float().masked_fill(~mask, 0.0)
    target = target / target.sum(-1, keepdim=True).clamp_min(1e-20)
    target = target.detach()
    ce = -(target * logp).sum(-1)
    ent = -(target * target.clamp_min(1e-20).log()).sum(-1)
    return (ce - ent).mean()
