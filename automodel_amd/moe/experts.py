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


class _FusedPermuteGather(torch.autograd.Function):
    """x [T,H] -> x_perm [T*K,H] via the HIP gather kernel (no
    repeat_interleave copy). Backward: per-token sum of replica grads via
    the combine kernel with unit probs."""

    @staticmethod
    def forward(ctx, x, src, pos, K):
        from automodel_amd.ops._backend import hip_ops

        ctx.save_for_backward(pos)
        ctx.K = K
        return hip_ops().permute_gather(x.contiguous(), src)

    @staticmethod
    def backward(ctx, g):
        from automodel_amd.ops._backend import hip_ops

        (pos,) = ctx.saved_tensors
        T = pos.numel() // ctx.K
        ones = torch.ones(T, ctx.K, dtype=torch.float32, device=g.device)
        dx = hip_ops().unpermute_combine(g.contiguous(), pos, ones)
        return dx, None, None, None


def fused_dispatch_combine(x, probs, indices, n_experts, expert_fn):
    """GPU MoE glue: argsort routing, HIP gather, expert_fn over permuted
    rows, HIP combine — one gather + one combine instead of
    repeat_interleave + fancy-index + scatter (reference fused permute/
    unpermute, moe_utils.py:31/115)."""
    T, K = indices.shape
    flat = indices.reshape(-1)
    sort_idx = flat.argsort(stable=True)
    counts = torch.bincount(flat, minlength=n_experts).to(torch.int32)
    src = (sort_idx // K).to(torch.int32)
    inv = torch.empty_like(sort_idx)
    inv[sort_idx] = torch.arange(T * K, device=x.device)
    pos = inv.to(torch.int32).view(T, K)
    perm_probs = probs.reshape(-1).float()[sort_idx]     # prob per perm slot
    x_perm = _FusedPermuteGather.apply(x, src, pos, K)
    y_perm = expert_fn(x_perm, counts)
    # combine with prob routing; dprobs via scatter of per-slot dots
    return _CombineWithProbs.apply(y_perm, pos, probs, src, perm_probs, sort_idx)


class _CombineWithProbs(torch.autograd.Function):
    @staticmethod
    def forward(ctx, y_perm, pos, probs, src, perm_probs, sort_idx):
        from automodel_amd.ops._backend import hip_ops

        out = hip_ops().unpermute_combine(y_perm.contiguous(), pos,
                                          probs.float().contiguous())
        ctx.save_for_backward(y_perm, src, perm_probs, sort_idx)
        ctx.shape_tk = probs.shape
        return out

    @staticmethod
    def backward(ctx, g):
        from automodel_amd.ops._backend import hip_ops

        y_perm, src, perm_probs, sort_idx = ctx.saved_tensors
        g = g.contiguous()
        g_gath = hip_ops().permute_gather(g, src)       # [T*K, H] perm order
        dy = (g_gath.float() * perm_probs.unsqueeze(1)).to(y_perm.dtype)
        s = (g_gath.float() * y_perm.float()).sum(-1)   # [T*K] perm order
        dprobs = torch.empty_like(s)
        dprobs.scatter_(0, sort_idx, s)
        return dy, None, dprobs.view(ctx.shape_tk).to(torch.float32), None, None, None


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
        # opt-in fp8 expert forward (set by quantization.fp8.apply_fp8 /
        # bench --fp8): fp8-e4m3 grouped NT forward with delayed tensorwise
        # scaling, bf16 grouped backward
        self.fp8 = False
        self._fp8_states = None

    def init_weights(self, std: float = 0.02) -> None:
        for p in (self.gate_proj, self.up_proj, self.down_proj):
            nn.init.normal_(p, std=std)

    @staticmethod
    def project(x_perm: torch.Tensor, w: torch.Tensor, counts) -> torch.Tensor:
        """Grouped projection y[t] = x[t] @ w[expert(t)].T over expert-sorted
        tokens — grouped-GEMM HIP kernel on GPU, per-expert loop on CPU.
        (Also used by peft/lora_experts.py adapters.)"""
        if (x_perm.is_cuda and x_perm.dtype == torch.bfloat16
                and x_perm.numel() > 0):
            from automodel_amd.ops.grouped_gemm import grouped_linear

            return grouped_linear(x_perm, w, counts)
        cl = counts.tolist() if torch.is_tensor(counts) else list(counts)
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
            from automodel_amd.ops.grouped_gemm import grouped_linear, make_group_plan

            if torch.is_tensor(counts) and counts.is_cuda:
                # device-side plan shared by all three projections — the
                # routing counts never touch the host;
                # 256-row tiles when every projection width allows
                bm = 256 if (self.intermediate_size % 256 == 0
                             and self.hidden_size % 256 == 0) else 128
                plan = (*make_group_plan(counts, x_perm.shape[0], bm), bm)
                cl = counts
            else:
                plan = None
                cl = counts.tolist() if torch.is_tensor(counts) else list(counts)
            if (self.fp8 and plan is not None
                    and self.intermediate_size % 128 == 0
                    and self.hidden_size % 128 == 0):
                from automodel_amd.ops.grouped_gemm import (
                    Fp8GroupedState,
                    grouped_linear_fp8,
                )

                if self._fp8_states is None:
                    self._fp8_states = {k: Fp8GroupedState()
                                        for k in ("gate", "up", "down")}
                st = self._fp8_states
                from automodel_amd.ops._backend import hip_ops

                xc = x_perm.contiguous()
                # non-differentiable byproducts: cast under no_grad so the
                # fp8 bytes don't drag an autograd-fallback node into BOTH
                # gate and up Function graphs
                with torch.no_grad():
                    sx = st["gate"].x_scale(xc)
                    x8 = hip_ops().fp8_cast(xc, sx, st["gate"].amax_x, False)
                g = grouped_linear_fp8(xc, self.gate_proj, cl, plan,
                                       st["gate"], x8, sx)
                u = grouped_linear_fp8(xc, self.up_proj, cl, plan,
                                       st["up"], x8, sx)
                h = swiglu(g, u)
                return grouped_linear_fp8(h, self.down_proj, cl, plan, st["down"])
            g = grouped_linear(x_perm, self.gate_proj, cl, plan=plan)
            u = grouped_linear(x_perm, self.up_proj, cl, plan=plan)
            h = swiglu(g, u)
            return grouped_linear(h, self.down_proj, cl, plan=plan)
        return self._expert_mlp_loop(x_perm, counts)

    def forward(self, x: torch.Tensor, probs: torch.Tensor, indices: torch.Tensor) -> torch.Tensor:
        if (x.is_cuda and x.dtype == torch.bfloat16 and x.shape[-1] % 8 == 0
                and self.backend in ("auto", "hip_grouped")):
            return fused_dispatch_combine(x, probs, indices, self.n_experts,
                                          self.forward_permuted)
        x_perm, sort_idx, counts = permute_tokens(x, indices, self.n_experts)
        y_perm = self.forward_permuted(x_perm, counts)
        return unpermute_tokens(y_perm, sort_idx, probs)
