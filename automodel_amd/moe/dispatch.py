"""Expert-parallel token dispatchers over RCCL.

Reference behavior (SURVEY §2.4 EP rows, §2.9 #13):
  * AllGatherDispatcher — the reference "torch" dispatcher
    (moe/experts.py:426-578): differentiable all-gather of tokens, local
    expert compute, all-reduce of outputs, narrow to the local slice.
  * AllToAllDispatcher — DeepEP-equivalent dispatch/combine built on RCCL
    all_to_all_single over xGMI (fused_a2a.py:139-331 interface): tokens are
    exchanged pairwise (xGMI is point-to-point: 7 direct links/GPU, so a
    direct a2a beats ring-shaped collectives), computed on the owning rank,
    and combined by the reverse a2a.

Both are autograd-transparent (collectives wrapped in autograd.Functions).
"""

from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn as nn

from automodel_amd.moe.experts import GroupedExperts, permute_tokens, unpermute_tokens


class _AllGather(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        world = dist.get_world_size(group)
        out = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(out, x.contiguous(), group=group)
        ctx.local_T = x.shape[0]
        return torch.cat(out, dim=0)

    @staticmethod
    def backward(ctx, grad):
        rank = dist.get_rank(ctx.group)
        g = grad.contiguous()
        if dist.get_backend(ctx.group) == "gloo":
            # gloo has no reduce_scatter: all-reduce + narrow (tests only)
            dist.all_reduce(g, group=ctx.group)
            return g.narrow(0, ctx.local_T * rank, ctx.local_T).contiguous(), None
        red = g.new_empty(ctx.local_T, *g.shape[1:])
        dist.reduce_scatter_tensor(red, g, group=ctx.group)
        return red, None


class _AllReduceSum(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        # clone: all_reduce is in-place and the input must stay untouched
        # for autograd correctness if the caller reuses it
        x = x.contiguous().clone()
        dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, grad):
        g = grad.contiguous()
        dist.all_reduce(g, group=ctx.group)
        return g, None


class _AllToAllV(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, out_splits, in_splits, group):
        ctx.group = group
        ctx.out_splits, ctx.in_splits = out_splits, in_splits
        out = x.new_empty(sum(out_splits), *x.shape[1:])
        dist.all_to_all_single(out, x.contiguous(), out_splits, in_splits, group=group)
        return out

    @staticmethod
    def backward(ctx, grad):
        g = grad.contiguous()
        out = g.new_empty(sum(ctx.in_splits), *g.shape[1:])
        dist.all_to_all_single(out, g, ctx.in_splits, ctx.out_splits, group=ctx.group)
        return out, None, None, None


class AllGatherDispatcher(nn.Module):
    """Gather all tokens, compute local experts, all-reduce, take local slice."""

    def __init__(self, ep_group, n_experts: int, local_expert_offset: int,
                 n_local_experts: int):
        super().__init__()
        self.group = ep_group
        self.n_experts = n_experts
        self.offset = local_expert_offset
        self.n_local = n_local_experts

    def forward(self, x, probs, indices, experts: GroupedExperts):
        rank = dist.get_rank(self.group)
        T = x.shape[0]
        xg = _AllGather.apply(x, self.group)
        pg = _AllGather.apply(probs, self.group)
        world = dist.get_world_size(self.group)
        ig_list = [torch.empty_like(indices) for _ in range(world)]
        dist.all_gather(ig_list, indices.contiguous(), group=self.group)
        ig = torch.cat(ig_list, dim=0)

        # zero out contributions of non-local experts, compute, combine
        local = (ig >= self.offset) & (ig < self.offset + self.n_local)
        ig_local = torch.where(local, ig - self.offset, torch.zeros_like(ig))
        p_masked = torch.where(local, pg, torch.zeros_like(pg))
        x_perm, sort_idx, counts = permute_tokens(xg, ig_local, self.n_local)
        # tokens routed to masked (non-local) experts still flow through
        # expert 0 but are combined with prob 0 — wasted flops, zero effect.
        y_perm = experts.forward_permuted(x_perm, counts)
        y = unpermute_tokens(y_perm, sort_idx, p_masked)
        y = _AllReduceSum.apply(y, self.group)
        return y.narrow(0, T * rank, T)


class AllToAllDispatcher(nn.Module):
    """Dispatch/combine with all_to_all_v over the EP group (RCCL over xGMI)."""

    def __init__(self, ep_group, n_experts: int, local_expert_offset: int,
                 n_local_experts: int):
        super().__init__()
        self.group = ep_group
        self.n_experts = n_experts
        self.offset = local_expert_offset
        self.n_local = n_local_experts

    def forward(self, x, probs, indices, experts: GroupedExperts):
        world = dist.get_world_size(self.group)
        # sort local token replicas by destination expert (contiguous dest ranks)
        x_perm, sort_idx, counts = permute_tokens(x, indices, self.n_experts)
        send_per_rank = counts.view(world, self.n_local).sum(dim=1)

        counts_g = torch.zeros(world, self.n_experts, dtype=counts.dtype,
                               device=counts.device)
        dist.all_gather_into_tensor(counts_g.view(-1), counts.contiguous(),
                                    group=self.group)
        rank = dist.get_rank(self.group)
        recv_per_rank = counts_g.view(world, world, self.n_local)[:, rank].sum(dim=1)

        send_splits = send_per_rank.tolist()
        recv_splits = recv_per_rank.tolist()
        x_recv = _AllToAllV.apply(x_perm, recv_splits, send_splits, self.group)

        # received tokens are ordered (src rank, expert); regroup by expert
        recv_counts = counts_g.view(world, world, self.n_local)[:, rank]  # [src, e]
        expert_ids = torch.repeat_interleave(
            torch.arange(world * self.n_local, device=x.device) % self.n_local,
            recv_counts.reshape(-1),
        )
        regroup = expert_ids.argsort(stable=True)
        per_expert = recv_counts.sum(dim=0)
        y_local = experts.forward_permuted(x_recv[regroup], per_expert)
        y_back = torch.empty_like(y_local)
        y_back[regroup] = y_local

        y_perm = _AllToAllV.apply(y_back, send_splits, recv_splits, self.group)
        return unpermute_tokens(y_perm, sort_idx, probs)


class PipelinedAllToAllDispatcher(AllToAllDispatcher):
    """Chunk-pipelined a2a dispatcher (reference fused_a2a.py/DeepEP overlap
    buffer behavior): local tokens are split into ``n_chunks``, the routing
    counts for ALL chunks are exchanged in ONE all_gather (single host sync
    for the split lists), then each chunk runs dispatch -> expert compute ->
    combine with known splits. On RCCL the collectives enqueue on the comm
    stream, so chunk i+1's dispatch payload overlaps chunk i's grouped-GEMM
    compute on the default stream; on gloo (CPU tests) the schedule is
    sequential but bitwise-identical.
    """

    def __init__(self, ep_group, n_experts: int, local_expert_offset: int,
                 n_local_experts: int, n_chunks: int = 2):
        super().__init__(ep_group, n_experts, local_expert_offset, n_local_experts)
        self.n_chunks = n_chunks

    def forward(self, x, probs, indices, experts: GroupedExperts):
        world = dist.get_world_size(self.group)
        rank = dist.get_rank(self.group)
        T = x.shape[0]
        nc = min(self.n_chunks, max(1, T))
        bounds = [(T * c // nc, T * (c + 1) // nc) for c in range(nc)]

        # per-chunk local permutation + counts
        perms = []
        all_counts = []
        for lo, hi in bounds:
            xp, si, cnt = permute_tokens(x[lo:hi], indices[lo:hi], self.n_experts)
            perms.append((xp, si))
            all_counts.append(cnt)
        counts = torch.stack(all_counts)                   # [nc, E]

        # ONE exchange + host sync for every chunk's split lists
        counts_g = torch.zeros(world, nc, self.n_experts, dtype=counts.dtype,
                               device=counts.device)
        dist.all_gather_into_tensor(counts_g.view(-1), counts.contiguous().view(-1),
                                    group=self.group)
        counts_host = counts_g.view(world, nc, world, self.n_local).cpu()
        send_host = counts.view(nc, world, self.n_local).sum(dim=2).cpu()

        outs = []
        for c, ((xp, si), (lo, hi)) in enumerate(zip(perms, bounds)):
            send_splits = send_host[c].tolist()
            recv_counts = counts_g.view(world, nc, world, self.n_local)[:, c, rank]
            recv_splits = counts_host[:, c, rank].sum(dim=1).tolist()
            x_recv = _AllToAllV.apply(xp, recv_splits, send_splits, self.group)
            expert_ids = torch.repeat_interleave(
                torch.arange(world * self.n_local, device=x.device) % self.n_local,
                recv_counts.reshape(-1))
            regroup = expert_ids.argsort(stable=True)
            per_expert = recv_counts.sum(dim=0)
            y_local = experts.forward_permuted(x_recv[regroup], per_expert)
            y_back = torch.empty_like(y_local)
            y_back[regroup] = y_local
            y_perm = _AllToAllV.apply(y_back, send_splits, recv_splits, self.group)
            outs.append(unpermute_tokens(y_perm, si, probs[lo:hi]))
        return torch.cat(outs, dim=0)
