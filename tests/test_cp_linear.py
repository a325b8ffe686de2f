"""Linear-attention CP (state relay): world-2 forward AND backward parity
against the single-rank full-sequence chunked kernels (GatedDeltaNet and
KDA)."""

import torch

from tests.dist_utils import run_distributed


def _make_inputs(seed=0, B=2, S=32, H=2, D=8, channelwise=False):
    g = torch.Generator().manual_seed(seed)
    q = torch.randn(B, S, H, D, generator=g)
    k = torch.randn(B, S, H, D, generator=g)
    v = torch.randn(B, S, H, D, generator=g)
    gate_shape = (B, S, H, D) if channelwise else (B, S, H)
    gate = -torch.rand(*gate_shape, generator=g) * 1.5
    beta = torch.rand(B, S, H, generator=g)
    return q, k, v, gate, beta


def _worker(rank, world, channelwise):
    from automodel_amd.models.kimi_linear.model import kda_chunked
    from automodel_amd.models.qwen3_next.model import gated_delta_rule_chunked
    from automodel_amd.parallel.cp_linear import cp_linear_scan, split_cp_chunk

    kernel = kda_chunked if channelwise else gated_delta_rule_chunked
    q, k, v, g, beta = _make_inputs(channelwise=channelwise)
    # ---- single-rank reference with grads
    qr, kr, vr, gr, br = (t.clone().requires_grad_() for t in (q, k, v, g, beta))
    ref = kernel(qr, kr, vr, gr, br, chunk_size=8)
    ref.square().sum().backward()
    # ---- CP: local chunk with grads
    ql, kl, vl, gl, bl = (split_cp_chunk(t, rank, world).requires_grad_()
                          for t in (q, k, v, g, beta))
    local_kernel = (lambda *a, **kw: kernel(*a, chunk_size=8, **kw))
    out = cp_linear_scan(local_kernel, ql, kl, vl, gl, bl)
    # the global objective sum over ranks of local chunk's ||out||^2
    out.square().sum().backward()
    ref_chunk = split_cp_chunk(ref.detach(), rank, world)
    fwd_err = float((out.detach() - ref_chunk).abs().max())
    bwd_errs = [
        float((a.grad - split_cp_chunk(b.grad, rank, world)).abs().max())
        for a, b in ((ql, qr), (kl, kr), (vl, vr), (gl, gr), (bl, br))
    ]
    return fwd_err, bwd_errs


def test_cp_gated_delta_rule_world2():
    out = run_distributed(_worker, world=2, args=(False,))
    for rank in (0, 1):
        fwd, bwd = out[rank]
        assert fwd < 3e-5, fwd
        assert all(e < 3e-4 for e in bwd), bwd


def test_cp_kda_world2():
    out = run_distributed(_worker, world=2, args=(True,))
    for rank in (0, 1):
        fwd, bwd = out[rank]
        assert fwd < 3e-5, fwd
        assert all(e < 3e-4 for e in bwd), bwd
