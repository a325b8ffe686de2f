"""Grouped expert compute: stacked weights, token permute, grouped GEMM.

Reference behavior: nemo_automodel/components/moe/experts.py:370-876
(GroupedExperts with loop / torch._grouped_mm paths; GroupedExpertsDeepEP
permute -> grouped GEMM -> activation -> grouped GEMM -> unpermute).

Weights are stacked [E, ...] tensors so EP sharding is a dim-0 DTensor shard
and HF per-expert keys map via the model's state_dict_adapter.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from automodel_amd.ops.swiglu import swiglu


def permute_tokens(x: torch.Tensor, indices: torch.Tensor, n_experts: int):
    """Sort token replicas by expert. Returns (x_perm [T*K, H], sort_idx,
    tokens_per_expert [E])."""
    T, K = indices.shape
    flat = indices.reshape(-1)
    sort_idx = flat.argsort(stable=True)
    counts = torch.bincount(flat, minlength=n_experts)
    x_rep = x.repeat_interleave(K, dim=0)
    return x_rep[sort_idx], sort_idx, counts


def unpermute_tokens(y_perm: torch.Tensor, sort_idx: torch.Tensor,
                     probs: torch.Tensor) -> torch.Tensor:
    """Scatter back and combine top-k with routing probs."""
    T, K = probs.shape
    y = torch.empty_like(y_perm)
    y[sort_idx] = y_perm
    y = y.view(T, K, -1)
    return (y * probs.unsqueeze(-1)).sum(dim=1)


class GroupedExperts(nn.Module):
    def __init__(self, n_experts: int, hidden_size: int, intermediate_size: int,
                 backend: str = "auto"):
        super().__init__()
        self.n_experts = n_experts
        self.hidden_size = hidden_size
        self.intermediate_size = intermediate_size
        self.backend = backend
        self.gate_proj = nn.Parameter(torch.empty(n_experts, intermediate_size, hidden_size))
        self.up_proj = nn.Parameter(torch.empty(n_experts, intermediate_size, hidden_size))
        self.down_proj = nn.Parameter(torch.empty(n_experts, hidden_size, intermediate_size))

    def init_weights(self, std: float = 0.02) -> None:
        for p in (self.gate_proj, self.up_proj, self.down_proj):
            nn.init.normal_(p, std=std)

    @staticmethod
    def project(x_perm: torch.Tensor, w: torch.Tensor, counts) -> torch.Tensor:
        """Grouped projection y[t] = x[t] @ w[expert(t)].T over expert-sorted
        tokens — grouped-GEMM HIP kernel on GPU, per-expert loop on CPU.
        (Also used by peft/lora_experts.py adapters.)"""
        cl = counts.tolist() if torch.is_tensor(counts) else list(counts)
        if (x_perm.is_cuda and x_perm.dtype == torch.bfloat16
                and x_perm.numel() > 0):
            from automodel_amd.ops.grouped_gemm import grouped_linear

            return grouped_linear(x_perm, w, cl)
        outs = []
        start = 0
        for e, n in enumerate(cl):
            if n == 0:
                continue
            outs.append(x_perm[start : start + n] @ w[e].t())
            start += n
        return torch.cat(outs, dim=0) if outs else x_perm[:0]

    def _expert_mlp_loop(self, x_perm: torch.Tensor, counts: torch.Tensor) -> torch.Tensor:
        outs = []
        start = 0
        counts_list = counts.tolist()
        for e, n in enumerate(counts_list):
            if n == 0:
                continue
            xe = x_perm[start : start + n]
            h = swiglu(xe @ self.gate_proj[e].t(), xe @ self.up_proj[e].t())
            outs.append(h @ self.down_proj[e].t())
            start += n
        return torch.cat(outs, dim=0) if outs else x_perm[:0]

    def forward_permuted(self, x_perm: torch.Tensor, counts: torch.Tensor) -> torch.Tensor:
        """Compute experts over tokens already sorted by expert (counts[e] each).

        counts covers THIS module's experts (post-EP-shard slice). On GPU the
        three projections run the in-tree grouped-GEMM HIP kernel.
        """
        use_grouped = (
            x_perm.is_cuda and self.backend in ("auto", "hip_grouped")
            and x_perm.dtype == torch.bfloat16
            and self.intermediate_size % 128 == 0 and self.hidden_size % 128 == 0
            and x_perm.numel() > 0
        )
        if use_grouped:
            from automodel_amd.ops.grouped_gemm import grouped_linear

            cl = counts.tolist() if torch.is_tensor(counts) else list(counts)
            g = grouped_linear(x_perm, self.gate_proj, cl)
            u = grouped_linear(x_perm, self.up_proj, cl)
            h = swiglu(g, u)
            return grouped_linear(h, self.down_proj, cl)
        return self._expert_mlp_loop(x_perm, counts)

    def forward(self, x: torch.Tensor, probs: torch.Tensor, indices: torch.Tensor) -> torch.Tensor:
        x_perm, sort_idx, counts = permute_tokens(x, indices, self.n_experts)
        y_perm = self.forward_permuted(x_perm, counts)
        return unpermute_tokens(y_perm, sort_idx, probs)
