"""Grouped GEMM for stacked expert weights: HIP MFMA kernel + fallbacks.

``grouped_linear(x_perm, w, counts)`` computes y[rows of group e] =
x[rows] @ w[e].T with autograd. Forward runs the in-tree CDNA4 kernel
(csrc/moe_kernels.hip grouped_gemm_nt); backward uses per-group hipBLASLt
GEMMs (dx = g @ w[e], dw[e] = g_e^T @ x_e).

Replaces the reference's grouped_gemm / torch._grouped_mm expert backends
(nemo_automodel/components/moe/experts.py:656).
"""

from __future__ import annotations

import torch

from automodel_amd.ops._backend import hip_ops

GG_BM = 128


def _build_tile_map(counts: list[int]) -> tuple[torch.Tensor, torch.Tensor]:
    """Host-side (expert, row0) tile list + int32 group offsets."""
    tiles = []
    offs = [0]
    for e, c in enumerate(counts):
        m0 = offs[-1]
        for t in range(0, c, GG_BM):
            tiles.append((e, m0 + t))
        offs.append(m0 + c)
    tile_map = torch.tensor(tiles, dtype=torch.int32).reshape(-1, 2)
    return tile_map, torch.tensor(offs, dtype=torch.int32)


def _loop_gemm_nt(x, w, counts):
    outs, start = [], 0
    for e, n in enumerate(counts):
        outs.append(x[start : start + n] @ w[e].t())
        start += n
    return torch.cat(outs) if outs else x.new_zeros(0, w.shape[1])


class _GroupedLinear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, counts: tuple):
        ctx.save_for_backward(x, w)
        ctx.counts = counts
        use_hip = (
            x.is_cuda and x.dtype == torch.bfloat16
            and w.shape[2] % 64 == 0 and w.shape[1] % 128 == 0
        )
        ctx.use_hip = use_hip
        if use_hip:
            tile_map, offs = _build_tile_map(list(counts))
            ctx.tile_offs = (tile_map.to(x.device), offs.to(x.device))
            return hip_ops().grouped_gemm_nt(
                x.contiguous(), w.contiguous(),
                ctx.tile_offs[1], ctx.tile_offs[0],
            )
        return _loop_gemm_nt(x, w, counts)

    @staticmethod
    def backward(ctx, g):
        x, w = ctx.saved_tensors
        counts = ctx.counts
        g = g.contiguous()
        # single-kernel grouped backward (the per-expert
        # hipBLASLt loop is launch-bound at 128+ small experts); dw's TN
        # kernel needs N%128 and K%128
        if ctx.use_hip and w.shape[1] % 128 == 0 and w.shape[2] % 128 == 0:
            tile_map, offs = ctx.tile_offs
            ops = hip_ops()
            dx = ops.grouped_gemm_nn(g, w.contiguous(), offs, tile_map)
            dw = ops.grouped_gemm_tn(g, x.contiguous(), offs, w.shape[0])
            return dx, dw, None
        dx = torch.empty_like(x)
        dw = torch.zeros_like(w)
        start = 0
        for e, n in enumerate(counts):
            if n:
                ge = g[start : start + n]
                dx[start : start + n] = ge @ w[e]
                dw[e] = (ge.t() @ x[start : start + n]).to(w.dtype)
            start += n
        return dx, dw, None


def grouped_linear(x_perm: torch.Tensor, w: torch.Tensor, counts) -> torch.Tensor:
    """x_perm [M, K] sorted by group; w [E, N, K]; counts per group (host)."""
    if not isinstance(counts, (tuple, list)):
        counts = counts.tolist()
    return _GroupedLinear.apply(x_perm, w, tuple(int(c) for c in counts))
