"""SGMV multi-adapter LoRA application (serving path).

Reference behavior: the reference serves per-request LoRA adapters with
punica-style SGMV kernels (SURVEY §2.9 "SGMV"). MI355X-native: tokens are
argsorted by adapter id, the grouped-GEMM device plan (256-row tiles) maps
each tile to its adapter, and ONE fused HIP kernel computes
delta = (x @ A[ad]^T) @ B[ad]^T * scale[ad] with the [M, r] intermediate
living in registers/LDS (ops/csrc/lora.hip sgmv_fused_fwd_kernel).
"""

from __future__ import annotations

import torch

from automodel_amd.ops._backend import hip_ops


def sgmv_delta(x: torch.Tensor, A: torch.Tensor, B: torch.Tensor,
               adapter_ids: torch.Tensor,
               scales: torch.Tensor | list[float]) -> torch.Tensor:
    """x [T, H]; A [n, r, H]; B [n, O, r]; adapter_ids [T] int; -> [T, O].

    Tokens may arrive in any order; sorting/unsorting is handled here.
    CPU (or unsupported-shape) path loops per adapter.
    """
    T, H = x.shape
    n, r, _ = A.shape
    O = B.shape[1]
    if not torch.is_tensor(scales):
        scales = torch.tensor(scales, dtype=torch.float32, device=x.device)
    use_hip = (x.is_cuda and x.dtype == torch.bfloat16 and r in (32, 64)
               and H % 64 == 0 and O % 64 == 0)
    if use_hip:
        order = torch.argsort(adapter_ids, stable=True)
        counts = torch.bincount(adapter_ids, minlength=n).to(torch.int32)
        xs = x[order].contiguous()
        offs, tile_map, n_tiles = hip_ops().build_group_plan(counts, T, 256)
        ys = hip_ops().sgmv_fused_fwd(xs, A.contiguous(), B.contiguous(),
                                      scales.float(), offs, tile_map, n_tiles)
        out = torch.empty_like(ys)
        out[order] = ys
        return out
    out = x.new_zeros(T, O)
    for a in range(n):
        m = adapter_ids == a
        if m.any():
            out[m] = ((x[m] @ A[a].t()) @ B[a].t() * float(scales[a])).to(x.dtype)
    return out
