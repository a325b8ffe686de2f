"""Grouped GEMM for stacked expert weights: HIP MFMA kernels + fallbacks.

``grouped_linear(x_perm, w, counts)`` computes y[rows of group e] =
x[rows] @ w[e].T with autograd. Forward runs the in-tree CDNA4 kernel
(csrc/moe_kernels.hip grouped_gemm_nt); backward runs the single-kernel
grouped dx (NN) and dw (TN) kernels — no per-expert loop, no atomics.

With ``counts`` as a DEVICE int tensor the (offs, tile_map, n_tiles) plan is
built by a device kernel (build_group_plan) so the hot path never syncs the
routing counts to the host (the round-1 counts.tolist() cost one device sync
per MoE layer per direction).

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


def make_group_plan(counts: torch.Tensor, M: int, bm: int = 128):
    """Device-side plan (offs, tile_map, n_tiles) from on-device counts."""
    return hip_ops().build_group_plan(counts.to(torch.int32), M, bm)


def _loop_gemm_nt(x, w, counts):
    outs, start = [], 0
    for e, n in enumerate(counts):
        outs.append(x[start : start + n] @ w[e].t())
        start += n
    return torch.cat(outs) if outs else x.new_zeros(0, w.shape[1])


class _GroupedLinear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, counts, offs, tile_map, n_tiles, bm):
        ctx.save_for_backward(x, w)
        use_hip = (
            x.is_cuda and x.dtype == torch.bfloat16
            and w.shape[2] % 64 == 0 and w.shape[1] % 128 == 0
        )
        ctx.use_hip = use_hip
        ctx.counts = counts
        ctx.plan = (offs, tile_map, n_tiles)
        ctx.bm = bm
        if use_hip:
            return hip_ops().grouped_gemm_nt(
                x.contiguous(), w.contiguous(), offs, tile_map, n_tiles, bm)
        cl = counts.tolist() if torch.is_tensor(counts) else list(counts)
        return _loop_gemm_nt(x, w, cl)

    @staticmethod
    def backward(ctx, g):
        x, w = ctx.saved_tensors
        g = g.contiguous()
        offs, tile_map, n_tiles = ctx.plan
        # single-kernel grouped backward; dw's TN
        # kernel needs N%128 and K%128
        if ctx.use_hip and w.shape[1] % 128 == 0 and w.shape[2] % 128 == 0:
            ops = hip_ops()
            if ctx.bm != 128:
                # nn rides 128-row tiles; rebuild its plan on device (cheap)
                if torch.is_tensor(ctx.counts) and ctx.counts.is_cuda:
                    offs, tile_map, n_tiles = make_group_plan(ctx.counts, g.shape[0], 128)
                else:
                    cl = list(ctx.counts)
                    tile_map, offs = _build_tile_map([int(c) for c in cl])
                    tile_map, offs, n_tiles = tile_map.to(g.device), offs.to(g.device), None
            dx = ops.grouped_gemm_nn(g, w.contiguous(), offs, tile_map, n_tiles)
            # TN operands pre-transposed (m contiguous) by the tiled HIP
            # transpose — torch's strided transpose measured 6x off roofline
            dw = ops.grouped_gemm_tn(ops.transpose_bf16(g),
                                     ops.transpose_bf16(x.contiguous()),
                                     offs, w.shape[0])
            return dx, dw, None, None, None, None, None
        cl = ctx.counts.tolist() if torch.is_tensor(ctx.counts) else list(ctx.counts)
        dx = torch.empty_like(x)
        dw = torch.zeros_like(w)
        start = 0
        for e, n in enumerate(cl):
            if n:
                ge = g[start : start + n]
                dx[start : start + n] = ge @ w[e]
                dw[e] = (ge.t() @ x[start : start + n]).to(w.dtype)
            start += n
        return dx, dw, None, None, None, None, None


def grouped_linear(x_perm: torch.Tensor, w: torch.Tensor, counts,
                   plan=None) -> torch.Tensor:
    """x_perm [M, K] sorted by group; w [E, N, K]; counts per group.

    ``counts`` may be a DEVICE int tensor (no host sync — plan built on
    device) or a host list/tuple. ``plan`` lets callers share one
    (offs, tile_map, n_tiles) across several projections of the same
    routing."""
    hip_eligible = (x_perm.is_cuda and x_perm.dtype == torch.bfloat16
                    and w.shape[2] % 64 == 0 and w.shape[1] % 128 == 0)
    if plan is not None:
        # caller-provided plan carries its tile height: (offs, tm, nt, bm)
        offs, tile_map, n_tiles, bm = plan
        return _GroupedLinear.apply(x_perm, w, counts, offs, tile_map, n_tiles, bm)
    # 256x256 big-tile forward when the output width allows (+27% measured,
    # benchmarks/gg_micro.py)
    bm = 256 if (hip_eligible and w.shape[1] % 256 == 0
                 and torch.is_tensor(counts) and counts.is_cuda) else 128
    if hip_eligible:
        if torch.is_tensor(counts) and counts.is_cuda:
            offs, tile_map, n_tiles = make_group_plan(counts, x_perm.shape[0], bm)
        else:
            cl = counts.tolist() if torch.is_tensor(counts) else list(counts)
            tile_map, offs = _build_tile_map([int(c) for c in cl])
            tile_map, offs, n_tiles = tile_map.to(x_perm.device), offs.to(x_perm.device), None
            bm = 128
    else:
        offs = tile_map = n_tiles = None
        bm = 128
    return _GroupedLinear.apply(x_perm, w, counts, offs, tile_map, n_tiles, bm)


# ===========================================================================
# fp8 grouped forward (SURVEY §2.5 MXFP8-grouped equivalent): tensorwise
# delayed scaling exactly like quantization/fp8.py Float8Linear — the step's
# activation scale comes from LAST step's recorded amax (margin 1.25x), the
# whole expert stack shares one cached e4m3 cast refreshed when the
# optimizer bumps w._version. Backward stays bf16 (grouped nn/tn kernels)
# from the saved bf16 operands: fp8 pays on the forward's bandwidth (the
# bf16 grouped kernels are memory-wait-bound per profiles/gg PMC), grads
# keep full grouped-bwd accuracy.
# ===========================================================================

E4M3_MAX = 448.0


class Fp8GroupedState:
    """Delayed-scaling state for one grouped projection (x amax window +
    cached expert-stack cast)."""

    MARGIN = 1.25

    def __init__(self):
        self.amax_x = None
        self.amax_x_prev = None
        self._wcache = None
        self._steps = 0

    def _lazy(self, dev):
        if self.amax_x is None:
            self.amax_x = torch.zeros(1, device=dev)
            self.amax_x_prev = torch.zeros(1, device=dev)

    def x_scale(self, x) -> torch.Tensor:
        self._lazy(x.device)
        if self._steps == 0:
            self.amax_x_prev.copy_(
                x.abs().amax().float().clamp(min=1e-12).reshape(1))
        return (E4M3_MAX / (self.amax_x_prev.clamp(min=1e-12) * self.MARGIN)
                ).clamp(max=1e12).reshape(1)

    def cached_w8(self, w):
        self._lazy(w.device)
        ver = w._version
        if self._wcache is not None and self._wcache[0] == ver:
            return self._wcache[1], self._wcache[2]
        # weight changed -> step boundary: roll the delayed window
        self.amax_x_prev.copy_(self.amax_x.clamp(min=1e-12))
        self.amax_x.zero_()
        if self._wcache is not None:
            self._steps += 1
        wd = w.detach().contiguous()
        amax = wd.abs().amax().float().clamp(min=1e-12).reshape(1)
        sw = (E4M3_MAX / amax).clamp(max=1e12)
        dummy = torch.zeros(1, device=w.device)
        w8 = hip_ops().fp8_cast(wd.view(-1, wd.shape[-1]), sw, dummy,
                                False).view(wd.shape)
        inv_w = sw.reciprocal()
        self._wcache = (ver, w8, inv_w)
        return w8, inv_w


class _GroupedLinearFp8(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, counts, offs, tile_map, n_tiles, bm, state,
                x8=None, sx=None):
        ops = hip_ops()
        x = x.contiguous()
        if x8 is None:
            sx = state.x_scale(x)
            x8 = ops.fp8_cast(x, sx, state.amax_x, False)
        w8, inv_w = state.cached_w8(w)
        dequant = sx.reciprocal() * inv_w
        y = ops.grouped_gemm_nt_fp8(x8, w8, offs, tile_map, dequant,
                                    n_tiles, bm)
        ctx.save_for_backward(x, w)
        ctx.use_hip = True
        ctx.counts = counts
        ctx.plan = (offs, tile_map, n_tiles)
        ctx.bm = bm
        return y

    @staticmethod
    def backward(ctx, g):
        dx, dw, *_ = _GroupedLinear.backward(ctx, g)
        return dx, dw, None, None, None, None, None, None, None, None


def grouped_linear_fp8(x_perm: torch.Tensor, w: torch.Tensor, counts,
                       plan, state: Fp8GroupedState, x8=None,
                       sx=None) -> torch.Tensor:
    """fp8-forward grouped linear. Caller guarantees bf16 CUDA operands with
    K%128==0, N%128==0 and a device plan (falls back to bf16 otherwise).
    ``x8``/``sx``: share one activation cast across projections of the same
    input (gate/up)."""
    if (not x_perm.is_cuda or x_perm.dtype != torch.bfloat16
            or w.shape[2] % 128 or w.shape[1] % 128 or plan is None):
        return grouped_linear(x_perm, w, counts, plan)
    offs, tile_map, n_tiles, bm = plan
    return _GroupedLinearFp8.apply(x_perm, w, counts, offs, tile_map,
                                   n_tiles, bm, state, x8, sx)
