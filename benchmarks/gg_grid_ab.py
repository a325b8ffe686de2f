"""A/B: big-tile NT grid order (m-first vs n-first) on MoE shapes (gpurun)."""
import os, subprocess, sys

for nfirst in ("0", "1"):
    env = dict(os.environ, AMD_OPS_GG_NFIRST=nfirst)
    code = (
        "import torch, time\n"
        "from automodel_amd.ops._backend import hip_ops\n"
        "ops = hip_ops()\n"
        "E, M, H = 128, 131072, 2048\n"
        "counts = torch.full((E,), M // E, dtype=torch.int32, device='cuda')\n"
        "x = torch.randn(M, H, device='cuda', dtype=torch.bfloat16)\n"
        "for N in (1024, 1536, 2048):\n"
        "    w = torch.randn(E, N, H, device='cuda', dtype=torch.bfloat16) * 0.02\n"
        "    offs, tm, ntl = ops.build_group_plan(counts, M, 256)\n"
        "    fn = lambda: ops.grouped_gemm_nt(x, w, offs, tm, ntl, 256)\n"
        "    for _ in range(3): fn()\n"
        "    torch.cuda.synchronize(); t0 = time.perf_counter()\n"
        "    for _ in range(10): fn()\n"
        "    torch.cuda.synchronize()\n"
        "    dt = (time.perf_counter() - t0) / 10\n"
        "    print(f'N={N}: {dt*1e3:.2f} ms {2*M*N*H/dt/1e12:.0f} TF/s')\n"
    )
    print(f"--- AMD_OPS_GG_NFIRST={nfirst}")
    sys.stdout.flush()
    subprocess.run(["python", "-c", code], env=env, check=True)
