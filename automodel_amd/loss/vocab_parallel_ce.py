"""Vocab-parallel cross entropy for TP-sharded lm_head.

Reference behavior: nemo_automodel/components/loss/te_parallel_ce.py:112 +
loss/triton/te_cross_entropy.py (per-rank online-softmax stats, all-reduce of
(m, d, X_y), then CE/grad on each shard). SURVEY §2.9 kernel #2.

Each rank holds a logits shard [T, V/tp]; (m, s) come from the HIP stats
kernel, the three scalars-per-row are exchanged with RCCL all-reduces, and
the backward writes d(logits_shard) in place with the existing ce_bwd kernel
(vocab_offset makes the one-hot land only on the owning rank).
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from automodel_amd.ops._backend import hip_ops

IGNORE_INDEX = -100


def _stats_ref(logits: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    m = logits.float().max(dim=-1).values
    s = torch.exp(logits.float() - m[:, None]).sum(-1)
    return m, s


class _VocabParallelCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits_shard: torch.Tensor, labels: torch.Tensor,
                vocab_offset: int, group):
        T, Vl = logits_shard.shape
        if logits_shard.is_cuda:
            m, s = hip_ops().ce_stats_logits(logits_shard.contiguous())
        else:
            m, s = _stats_ref(logits_shard)

        m_global = m.clone()
        if group is not None:
            dist.all_reduce(m_global, op=dist.ReduceOp.MAX, group=group)
        s_adj = s * torch.exp(m - m_global)
        if group is not None:
            dist.all_reduce(s_adj, group=group)
        lse = m_global + torch.log(s_adj)

        local = (labels >= vocab_offset) & (labels < vocab_offset + Vl) & (labels != IGNORE_INDEX)
        xy = torch.zeros(T, dtype=torch.float32, device=logits_shard.device)
        if local.any():
            rows = local.nonzero(as_tuple=True)[0]
            xy[rows] = logits_shard[rows, labels[rows] - vocab_offset].float()
        if group is not None:
            dist.all_reduce(xy, group=group)
        valid = labels != IGNORE_INDEX
        loss = ((lse - xy) * valid.float()).sum()

        ctx.save_for_backward(logits_shard, labels, lse)
        ctx.vocab_offset = vocab_offset
        ctx.group = group
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits_shard, labels, lse = ctx.saved_tensors
        d = dloss.reshape(1).float().contiguous()
        if logits_shard.is_cuda:
            g = logits_shard.clone()
            hip_ops().ce_bwd_logits(g, labels, lse, d, ctx.vocab_offset)
        else:
            p = torch.exp(logits_shard.float() - lse[:, None])
            local_y = labels - ctx.vocab_offset
            valid = labels != IGNORE_INDEX
            onehot_rows = valid & (local_y >= 0) & (local_y < logits_shard.shape[1])
            rows = onehot_rows.nonzero(as_tuple=True)[0]
            p[rows, local_y[rows]] -= 1.0
            p[~valid] = 0.0
            g = (p * d).to(logits_shard.dtype)
        return g, None, None, None


def vocab_parallel_cross_entropy(
    logits_shard: torch.Tensor,
    labels: torch.Tensor,
    vocab_offset: int = 0,
    group=None,
) -> torch.Tensor:
    """logits_shard [*, V_local] (this rank's vocab slice), labels global ids.
    Returns the token-sum loss (identical on every TP rank)."""
    return _VocabParallelCE.apply(
        logits_shard.reshape(-1, logits_shard.shape[-1]), labels.reshape(-1),
        vocab_offset, group,
    )
