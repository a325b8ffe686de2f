"""SGMV multi-adapter LoRA CPU tests."""

import torch




def test_sgmv_cpu_fallback_matches_loop():
    """CPU path: sgmv_delta equals the explicit per-adapter computation."""
    from automodel_amd.peft.sgmv import sgmv_delta

    torch.manual_seed(0)
    T, H, O, n, r = 23, 16, 12, 3, 4
    x = torch.randn(T, H)
    A = torch.randn(n, r, H) * 0.1
    B = torch.randn(n, O, r) * 0.1
    ids = torch.randint(0, n, (T,))
    y = sgmv_delta(x, A, B, ids, [1.0, 2.0, 0.5])
    for a in range(n):
        m = ids == a
        if m.any():
            ref = (x[m] @ A[a].t()) @ B[a].t() * [1.0, 2.0, 0.5][a]
            torch.testing.assert_close(y[m], ref, atol=1e-5, rtol=1e-5)
