"""Setup-time prewarm: GEMM workspaces, HIP kernel first-launch, RCCL comms.

Reference behavior: nemo_automodel/components/training/prewarm.py
(cuBLAS workspace + Triton autotune + NCCL communicator warmup before the
timed loop; train_ft.py:664-677). On MI355X: hipBLASLt workspace allocation,
first-launch of every in-tree HIP kernel, and one small collective per mesh
group so RCCL communicators exist before step 1.
"""

from __future__ import annotations

import torch
import torch.distributed as dist


def prewarm_gemms(shapes: list[tuple[int, int, int]] | None = None,
                  dtype=torch.bfloat16) -> None:
    if not torch.cuda.is_available():
        return
    shapes = shapes or [(512, 512, 512)]
    for m, n, k in shapes:
        a = torch.randn(m, k, device="cuda", dtype=dtype)
        b = torch.randn(n, k, device="cuda", dtype=dtype)
        (a @ b.t()).sum().item()


def prewarm_collectives(mesh_ctx=None) -> None:
    if not (dist.is_available() and dist.is_initialized()):
        return
    device = "cuda" if torch.cuda.is_available() else "cpu"
    t = torch.ones(1, device=device)
    dist.all_reduce(t)
    if mesh_ctx is not None and getattr(mesh_ctx, "mesh", None) is not None:
        for name in ("dp", "tp", "cp", "pp"):
            try:
                g = mesh_ctx.mesh[name].get_group()
                if dist.get_world_size(g) > 1:
                    dist.all_reduce(t.clone(), group=g)
            except Exception:
                pass
    if torch.cuda.is_available():
        torch.cuda.synchronize()


def prewarm_hip_kernels() -> None:
    """First-launch every in-tree kernel so compilation/caching cost stays
    out of the timed region."""
    if not torch.cuda.is_available():
        return
    from automodel_amd.ops.attention import flash_attention
    from automodel_amd.ops.rms_norm import rms_norm
    from automodel_amd.ops.rope import apply_rope, build_rope_cache
    from automodel_amd.ops.swiglu import swiglu

    x = torch.randn(128, 256, device="cuda", dtype=torch.bfloat16)
    w = torch.ones(256, device="cuda", dtype=torch.bfloat16)
    rms_norm(x, w, 1e-6)
    swiglu(x, x)
    q = torch.randn(1, 128, 2, 128, device="cuda", dtype=torch.bfloat16)
    cos, sin = build_rope_cache(128, 128, device="cuda")
    apply_rope(q, q, cos, sin)
    flash_attention(q, q, q, causal=True)
    torch.cuda.synchronize()
