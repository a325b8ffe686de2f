"""HIP kernel parity tests vs plain-torch fp32 references (run on MI355X).

Every test compares the CDNA4 kernel against the same reference the CPU suite
validates (SURVEY §4 takeaway (d): kernel-level parity tests are the gate).
"""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _require_ops():
    from automodel_amd.ops._backend import require_ops

    require_ops()


def test_mfma_probe_layout():
    """Validates the assumed A/B/C fragment mappings for mfma_f32_32x32x16_bf16.

    Asymmetric operands per guide §3 (symmetric B would hide a transposed
    C-write)."""
    torch.manual_seed(0)
    a = torch.randn(32, 16, device="cuda").to(torch.bfloat16)
    b = torch.randn(16, 32, device="cuda").to(torch.bfloat16)
    d = torch.ops.amd_ops.mfma_probe(a, b)
    ref = a.float() @ b.float()
    assert torch.allclose(d, ref, atol=2e-2, rtol=1e-2), (d - ref).abs().max()


def test_rms_norm_fwd_bwd_parity():
    from automodel_amd.ops.rms_norm import rms_norm, rms_norm_ref

    torch.manual_seed(0)
    for T, H in [(128, 4096), (64, 2048), (256, 256)]:
        x = torch.randn(T, H, device="cuda", dtype=torch.bfloat16, requires_grad=True)
        w = torch.randn(H, device="cuda", dtype=torch.bfloat16, requires_grad=True)
        y = rms_norm(x, w, 1e-6, backend="hip")
        x2 = x.detach().clone().float().requires_grad_(True)
        w2 = w.detach().clone().float().requires_grad_(True)
        y_ref = rms_norm_ref(x2, w2, 1e-6)
        assert torch.allclose(y.float(), y_ref, atol=3e-2, rtol=3e-2)

        dy = torch.randn_like(y)
        y.backward(dy)
        y_ref.backward(dy.float())
        assert torch.allclose(x.grad.float(), x2.grad, atol=5e-2, rtol=5e-2), \
            (x.grad.float() - x2.grad).abs().max()
        assert torch.allclose(w.grad.float(), w2.grad, atol=0.1, rtol=5e-2), \
            (w.grad.float() - w2.grad).abs().max()


def test_rope_parity():
    from automodel_amd.ops.rope import apply_rope, apply_rope_ref, build_rope_cache

    torch.manual_seed(0)
    B, S, Hq, Hk, D = 2, 64, 4, 2, 128
    q = torch.randn(B, S, Hq, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn(B, S, Hk, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    cos, sin = build_rope_cache(D, S, base=500000.0, device="cuda")
    qo, ko = apply_rope(q, k, cos, sin, backend="hip")
    q2 = q.detach().clone().requires_grad_(True)
    k2 = k.detach().clone().requires_grad_(True)
    qr, kr = apply_rope_ref(q2, k2, cos, sin)
    assert torch.allclose(qo.float(), qr.float(), atol=2e-2, rtol=2e-2)
    assert torch.allclose(ko.float(), kr.float(), atol=2e-2, rtol=2e-2)
    dq, dk = torch.randn_like(qo), torch.randn_like(ko)
    torch.autograd.backward([qo, ko], [dq, dk])
    torch.autograd.backward([qr, kr], [dq, dk])
    assert torch.allclose(q.grad.float(), q2.grad.float(), atol=2e-2, rtol=2e-2)
    assert torch.allclose(k.grad.float(), k2.grad.float(), atol=2e-2, rtol=2e-2)


def test_swiglu_parity():
    from automodel_amd.ops.swiglu import swiglu, swiglu_ref

    g = torch.randn(64, 512, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    u = torch.randn(64, 512, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    y = swiglu(g, u, backend="hip")
    g2 = g.detach().clone().float().requires_grad_(True)
    u2 = u.detach().clone().float().requires_grad_(True)
    y_ref = swiglu_ref(g2, u2)
    assert torch.allclose(y.float(), y_ref, atol=3e-2, rtol=3e-2)
    dy = torch.randn_like(y)
    y.backward(dy)
    y_ref.backward(dy.float())
    assert torch.allclose(g.grad.float(), g2.grad, atol=5e-2, rtol=5e-2)
    assert torch.allclose(u.grad.float(), u2.grad, atol=5e-2, rtol=5e-2)


def test_adamw_hip_matches_torch_fp32():
    """bf16 param + fp32 master HIP step vs torch.optim.AdamW on fp32."""
    from automodel_amd.optim.adamw import FusedAdamW

    torch.manual_seed(0)
    w0 = torch.randn(4096, device="cuda")
    p_hip = torch.nn.Parameter(w0.clone().to(torch.bfloat16))
    p_ref = torch.nn.Parameter(w0.clone())
    opt_hip = FusedAdamW([p_hip], lr=1e-2, betas=(0.9, 0.95), weight_decay=0.1)
    opt_ref = torch.optim.AdamW([p_ref], lr=1e-2, betas=(0.9, 0.95), eps=1e-8,
                                weight_decay=0.1)
    for _ in range(5):
        g = torch.randn(4096, device="cuda")
        p_hip.grad = g.to(torch.bfloat16)
        p_ref.grad = g.clone()
        opt_hip.step()
        opt_ref.step()
    master = opt_hip.state[p_hip]["master"]
    # master follows the fp32 trajectory up to bf16 grad quantization
    assert torch.allclose(master, p_ref.detach(), atol=2e-3, rtol=1e-2), \
        (master - p_ref).abs().max()
    assert torch.allclose(p_hip.float(), p_ref.detach(), atol=2e-2, rtol=2e-2)


def test_flash_attention_fwd_parity():
    from automodel_amd.ops.attention import attention_ref, flash_attention

    torch.manual_seed(0)
    for B, S, Hq, Hk in [(2, 256, 4, 2), (1, 512, 8, 8), (1, 128, 4, 1)]:
        q = torch.randn(B, S, Hq, 128, device="cuda", dtype=torch.bfloat16)
        k = torch.randn(B, S, Hk, 128, device="cuda", dtype=torch.bfloat16)
        v = torch.randn(B, S, Hk, 128, device="cuda", dtype=torch.bfloat16)
        o = flash_attention(q, k, v, causal=True, backend="hip")
        o_ref = attention_ref(q.float(), k.float(), v.float(), causal=True)
        assert torch.allclose(o.float(), o_ref, atol=3e-2, rtol=3e-2), \
            (B, S, Hq, Hk, (o.float() - o_ref).abs().max())


def test_flash_attention_fwd_noncausal():
    from automodel_amd.ops.attention import attention_ref, flash_attention

    q = torch.randn(1, 128, 4, 128, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(1, 128, 2, 128, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(1, 128, 2, 128, device="cuda", dtype=torch.bfloat16)
    o = flash_attention(q, k, v, causal=False, backend="hip")
    o_ref = attention_ref(q.float(), k.float(), v.float(), causal=False)
    assert torch.allclose(o.float(), o_ref, atol=3e-2, rtol=3e-2)


def test_flash_attention_bwd_grads():
    from automodel_amd.ops.attention import attention_ref, flash_attention

    torch.manual_seed(1)
    B, S, Hq, Hk, D = 1, 256, 4, 2, 128
    q = torch.randn(B, S, Hq, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn(B, S, Hk, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    v = torch.randn(B, S, Hk, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    o = flash_attention(q, k, v, causal=True, backend="hip")
    do = torch.randn_like(o)
    o.backward(do)

    q2 = q.detach().clone().float().requires_grad_(True)
    k2 = k.detach().clone().float().requires_grad_(True)
    v2 = v.detach().clone().float().requires_grad_(True)
    o_ref = attention_ref(q2, k2, v2, causal=True)
    o_ref.backward(do.float())
    assert torch.allclose(q.grad.float(), q2.grad, atol=5e-2, rtol=5e-2), \
        (q.grad.float() - q2.grad).abs().max()
    assert torch.allclose(k.grad.float(), k2.grad, atol=5e-2, rtol=5e-2), \
        (k.grad.float() - k2.grad).abs().max()
    assert torch.allclose(v.grad.float(), v2.grad, atol=5e-2, rtol=5e-2), \
        (v.grad.float() - v2.grad).abs().max()


def test_flash_attention_q_start_parity():
    """CP building block: q chunk attending a longer KV prefix with offset."""
    from automodel_amd.ops.attention import attention_ref, flash_attention

    torch.manual_seed(2)
    B, Sq, Skv, Hq, Hk, D = 1, 128, 384, 4, 2, 128
    q = torch.randn(B, Sq, Hq, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn(B, Skv, Hk, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    v = torch.randn(B, Skv, Hk, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    o = flash_attention(q, k, v, causal=True, backend="hip", q_start=256)
    q2 = q.detach().float().requires_grad_(True)
    k2 = k.detach().float().requires_grad_(True)
    v2 = v.detach().float().requires_grad_(True)
    o_ref = attention_ref(q2, k2, v2, causal=True, q_start=256)
    assert torch.allclose(o.float(), o_ref, atol=3e-2, rtol=3e-2), \
        (o.float() - o_ref).abs().max()
    do = torch.randn_like(o)
    o.backward(do)
    o_ref.backward(do.float())
    assert torch.allclose(q.grad.float(), q2.grad, atol=5e-2, rtol=5e-2)
    assert torch.allclose(k.grad.float(), k2.grad, atol=5e-2, rtol=5e-2)
    assert torch.allclose(v.grad.float(), v2.grad, atol=5e-2, rtol=5e-2)


def test_hybrid_linear_ce_parity():
    """hybrid (hipBLASLt GEMM + HIP CE epilogue) vs fp32 dense reference."""
    from automodel_amd.loss.linear_ce import fused_linear_cross_entropy

    torch.manual_seed(0)
    T, H, V = 512, 256, 1024
    hidden = torch.randn(T, H, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    weight = torch.randn(V, H, device="cuda", dtype=torch.bfloat16, requires_grad=True) * 0.05
    weight = weight.detach().requires_grad_(True)
    labels = torch.randint(0, V, (T,), device="cuda")
    labels[7] = -100
    loss = fused_linear_cross_entropy(hidden, weight, labels, backend="hybrid",
                                      chunk_size=128)
    loss.backward()

    h2 = hidden.detach().float().requires_grad_(True)
    w2 = weight.detach().float().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(h2 @ w2.t(), labels,
                                            ignore_index=-100, reduction="sum")
    ref.backward()
    assert torch.allclose(loss, ref, rtol=2e-2), (float(loss), float(ref))
    assert torch.allclose(hidden.grad.float(), h2.grad, atol=5e-2, rtol=5e-2), \
        (hidden.grad.float() - h2.grad).abs().max()
    assert torch.allclose(weight.grad.float(), w2.grad, atol=5e-2, rtol=5e-2), \
        (weight.grad.float() - w2.grad).abs().max()


def test_grouped_gemm_parity():
    """HIP grouped GEMM vs per-group hipBLASLt loop (fwd + bwd)."""
    from automodel_amd.ops.grouped_gemm import _loop_gemm_nt, grouped_linear

    torch.manual_seed(0)
    E, N, K = 4, 256, 128
    counts = [100, 0, 300, 37]
    M = sum(counts)
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    w = torch.randn(E, N, K, device="cuda", dtype=torch.bfloat16, requires_grad=True) * 0.05
    w = w.detach().requires_grad_(True)
    y = grouped_linear(x, w, counts)
    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    y_ref = _loop_gemm_nt(x2, w2, counts)
    assert torch.allclose(y.float(), y_ref.float(), atol=3e-2, rtol=3e-2), \
        (y.float() - y_ref.float()).abs().max()
    g = torch.randn_like(y)
    y.backward(g)
    y_ref.backward(g)
    assert torch.allclose(x.grad.float(), x2.grad.float(), atol=3e-2, rtol=3e-2)
    assert torch.allclose(w.grad.float(), w2.grad.float(), atol=3e-2, rtol=3e-2)


def test_permute_kernels_parity():
    from automodel_amd.ops._backend import hip_ops

    torch.manual_seed(0)
    T, H, K = 64, 256, 2
    x = torch.randn(T, H, device="cuda", dtype=torch.bfloat16)
    src = torch.randint(0, T, (T * K,), device="cuda", dtype=torch.int32)
    y = hip_ops().permute_gather(x, src)
    assert torch.equal(y, x[src.long()])

    yp = torch.randn(T * K, H, device="cuda", dtype=torch.bfloat16)
    pos = torch.randperm(T * K, device="cuda", dtype=torch.int32)
    probs = torch.rand(T, K, device="cuda", dtype=torch.float32)
    out = hip_ops().unpermute_combine(yp, pos, probs)
    ref = (yp[pos.long()].view(T, K, H).float() * probs[:, :, None]).sum(1)
    assert torch.allclose(out.float(), ref, atol=3e-2, rtol=3e-2)


def test_moe_model_gpu_step():
    """Tiny MoE model full train step on GPU (grouped-GEMM path active)."""
    from automodel_amd.loss.linear_ce import FusedLinearCrossEntropy
    from automodel_amd.moe.model import MoEForCausalLM
    from automodel_amd.optim.adamw import FusedAdamW

    torch.manual_seed(0)
    m = MoEForCausalLM(dict(
        vocab_size=512, hidden_size=256, intermediate_size=512,
        num_hidden_layers=2, num_attention_heads=2, num_key_value_heads=2,
        head_dim=128, max_position_embeddings=256,
        moe={"n_routed_experts": 4, "n_activated_experts": 2,
             "moe_intermediate_size": 128, "aux_loss_coeff": 0.01},
    ))
    m = m.to(torch.bfloat16)
    m.loss_fn = FusedLinearCrossEntropy(backend="hybrid")
    m.init_weights(device="cuda")
    opt = FusedAdamW(m.parameters(), lr=1e-3, weight_decay=0.0)
    ids = torch.randint(0, 512, (2, 129), device="cuda")
    losses = []
    for _ in range(6):
        loss = m(ids[:, :-1], labels=ids[:, 1:].contiguous()) / 256
        opt.zero_grad()
        loss.backward()
        opt.step()
        m.update_moe_gate_bias()
        losses.append(float(loss.detach()))
    assert all(math.isfinite(x) for x in losses)
    assert losses[-1] < losses[0], losses


def test_model_train_step_cuda():
    """Tiny model full step on GPU with all HIP backends active."""
    from automodel_amd.loss.linear_ce import FusedLinearCrossEntropy
    from automodel_amd.models.common.backend import BackendConfig
    from automodel_amd.models.llama.model import LlamaConfig, LlamaForCausalLM
    from automodel_amd.optim.adamw import FusedAdamW

    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=512, hidden_size=256, intermediate_size=512,
                      num_hidden_layers=2, num_attention_heads=2,
                      num_key_value_heads=2, head_dim=128,
                      max_position_embeddings=256)
    with torch.device("meta"):
        m = LlamaForCausalLM(cfg, backend=BackendConfig())
    m = m.to(dtype=torch.bfloat16)
    m.loss_fn = FusedLinearCrossEntropy(backend="chunked")
    m.init_weights(device="cuda")
    opt = FusedAdamW(m.parameters(), lr=1e-3, weight_decay=0.0)
    ids = torch.randint(0, 512, (2, 129), device="cuda")
    losses = []
    for _ in range(8):
        loss = m(ids[:, :-1], labels=ids[:, 1:].contiguous()) / (2 * 128)
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert all(math.isfinite(x) for x in losses)
    assert losses[-1] < losses[0] * 0.9, losses


def test_varlen_attention_gpu():
    """Packed varlen path on the HIP kernel (pad-to-128 per doc)."""
    from automodel_amd.ops.attention import attention_ref, flash_attention_varlen

    torch.manual_seed(0)
    H, Hk, D = 4, 2, 128
    cu = torch.tensor([0, 100, 356, 420], dtype=torch.int32)
    T = 420
    q = torch.randn(1, T, H, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn(1, T, Hk, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    v = torch.randn(1, T, Hk, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    out = flash_attention_varlen(q, k, v, cu, backend="hip")
    for a, b in zip(cu[:-1].tolist(), cu[1:].tolist()):
        ref = attention_ref(q[:, a:b].float(), k[:, a:b].float(), v[:, a:b].float(),
                            causal=True)
        assert torch.allclose(out[:, a:b].float(), ref, atol=3e-2, rtol=3e-2), (a, b)
    out.sum().backward()
    assert torch.isfinite(q.grad.float()).all()


def test_fp8_linear_numerics_and_train():
    """Float8Linear vs bf16 linear: loose forward parity + trainable."""
    from automodel_amd.quantization.fp8 import Float8Linear, apply_fp8_to_model

    torch.manual_seed(0)
    lin = torch.nn.Linear(256, 512, bias=False, device="cuda", dtype=torch.bfloat16)
    f8 = Float8Linear.from_linear(lin)
    x = torch.randn(64, 256, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    y8 = f8(x)
    y16 = torch.nn.functional.linear(x.detach(), lin.weight)
    rel = (y8.float() - y16.float()).abs().mean() / y16.float().abs().mean()
    assert rel < 0.1, float(rel)
    y8.sum().backward()
    assert x.grad is not None and lin.weight.grad is not None
    assert torch.isfinite(x.grad.float()).all()

    # model swap: loss still decreases
    from automodel_amd.loss.linear_ce import FusedLinearCrossEntropy
    from automodel_amd.models.llama.model import LlamaConfig, LlamaForCausalLM
    from automodel_amd.optim.adamw import FusedAdamW

    m = LlamaForCausalLM(LlamaConfig(
        vocab_size=512, hidden_size=512, intermediate_size=1024,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=4,
        head_dim=128, max_position_embeddings=256)).to(torch.bfloat16)
    m.loss_fn = FusedLinearCrossEntropy(backend="hybrid")
    m.init_weights(device="cuda")
    n = apply_fp8_to_model(m)
    assert n == 2 * 7
    opt = FusedAdamW(m.parameters(), lr=1e-3, weight_decay=0.0)
    ids = torch.randint(0, 512, (2, 129), device="cuda")
    losses = []
    for _ in range(6):
        loss = m(ids[:, :-1], labels=ids[:, 1:].contiguous()) / 256
        opt.zero_grad(); loss.backward(); opt.step()
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0], losses


def test_soft_ce_hip_parity():
    from automodel_amd.loss.kd_loss import soft_cross_entropy_from_logits

    torch.manual_seed(0)
    T, V = 128, 1024
    s = torch.randn(T, V, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    t = torch.randn(T, V, device="cuda", dtype=torch.bfloat16)
    loss = soft_cross_entropy_from_logits(s, t)
    s2 = s.detach().float().requires_grad_(True)
    ref = -(torch.softmax(t.float(), -1) * torch.log_softmax(s2, -1)).sum()
    assert torch.allclose(loss, ref, rtol=2e-2), (float(loss), float(ref))
    loss.backward()
    ref.backward()
    assert torch.allclose(s.grad.float(), s2.grad, atol=5e-2, rtol=5e-2), \
        (s.grad.float() - s2.grad).abs().max()


def test_hip_graph_capture_replay():
    """Partial hipGraph capture: replay matches eager for a small module."""
    from automodel_amd.utils.hip_graphs import GraphedForward

    torch.manual_seed(0)
    mod = torch.nn.Sequential(
        torch.nn.Linear(64, 64, device="cuda", dtype=torch.bfloat16),
        torch.nn.GELU(),
        torch.nn.Linear(64, 64, device="cuda", dtype=torch.bfloat16),
    ).eval()
    g = GraphedForward(mod, warmup=2)
    x = torch.randn(8, 64, device="cuda", dtype=torch.bfloat16)
    with torch.no_grad():
        for _ in range(3):
            _ = g(x)           # warmup + capture
        y_graph = g(x).clone()  # replay path
        y_eager = mod(x)
    assert torch.allclose(y_graph.float(), y_eager.float(), atol=1e-2)
    # different data through the same graph
    x2 = torch.randn_like(x)
    with torch.no_grad():
        y2 = g(x2).clone()
        assert torch.allclose(y2.float(), mod(x2).float(), atol=1e-2)


def test_hip_graph_training_capture():
    """Training-mode hipGraph capture (make_graphed_callables): fwd+bwd
    replay matches eager gradients."""
    from automodel_amd.utils.hip_graphs import apply_training_graphs

    torch.manual_seed(1)

    class Block(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.fc1 = torch.nn.Linear(64, 128, device="cuda")
            self.fc2 = torch.nn.Linear(128, 64, device="cuda")

        def forward(self, x):
            return self.fc2(torch.nn.functional.gelu(self.fc1(x)))

    model = torch.nn.Sequential(Block(), Block())
    ref = torch.nn.Sequential(Block(), Block())
    ref.load_state_dict(model.state_dict())

    sample = torch.randn(8, 64, device="cuda")
    n = apply_training_graphs(model, ("Block",), sample)
    assert n == 2

    x = torch.randn(8, 64, device="cuda", requires_grad=True)
    y = model(x)
    y.sum().backward()
    xr = x.detach().clone().requires_grad_(True)
    ref(xr).sum().backward()
    assert torch.allclose(y.float(), ref(xr).float(), atol=1e-4)
    assert torch.allclose(x.grad, xr.grad, atol=1e-4)


def test_new_families_gpu_smoke():
    """Gemma-2 (head_dim 256), GPT-OSS (head_dim 64) and Qwen2-VL run
    forward+backward on MI355X bf16 at their REAL head dims — the round-1
    proxies used head_dim 16 which silently rode sdpa;
    these now run the in-tree flash kernels."""
    from automodel_amd.models.gemma.model import GemmaForCausalLM
    from automodel_amd.models.gpt_oss.model import GptOssForCausalLM
    from automodel_amd.models.qwen2_vl.model import Qwen2VLForConditionalGeneration

    def bf16(m):
        return m.to(torch.bfloat16)

    g = GemmaForCausalLM(dict(vocab_size=200, hidden_size=512, intermediate_size=128,
                              num_hidden_layers=2, num_attention_heads=2,
                              num_key_value_heads=1, head_dim=256,
                              max_position_embeddings=256,
                              query_pre_attn_scalar=256.0))
    g.init_weights(device="cuda")
    g = bf16(g)
    ids = torch.randint(0, 200, (2, 144), device="cuda")
    g(ids).float().sum().backward()
    assert g.model.layers[0].mlp.gate_proj.weight.grad is not None

    o = GptOssForCausalLM(dict(vocab_size=200, hidden_size=128, intermediate_size=96,
                               num_hidden_layers=2, num_attention_heads=2,
                               num_key_value_heads=1, head_dim=64,
                               num_local_experts=4, num_experts_per_tok=2,
                               max_position_embeddings=256))
    o.init_weights(device="cuda")
    o = bf16(o)
    o(ids).float().sum().backward()
    assert o.model.layers[0].mlp.experts.gate_up_proj.grad is not None

    v = Qwen2VLForConditionalGeneration(dict(
        text=dict(vocab_size=300, hidden_size=64, intermediate_size=128,
                  num_hidden_layers=2, num_attention_heads=4,
                  num_key_value_heads=2, max_position_embeddings=64,
                  attention_bias=True),
        vision=dict(embed_dim=32, depth=1, num_heads=2, hidden_size=64,
                    patch_size=4, temporal_patch_size=2, spatial_merge_size=2),
        mrope_section=(2, 3, 3), image_token_id=299))
    v.init_weights(device="cuda")
    v = bf16(v)
    seq = torch.cat([torch.randint(0, 290, (1, 4), device="cuda"),
                     torch.full((1, 4), 299, device="cuda"),
                     torch.randint(0, 290, (1, 4), device="cuda")], dim=1)
    pix = torch.randn(16, 3 * 2 * 4 * 4, device="cuda", dtype=torch.bfloat16)
    out = v(seq, pixel_values=pix, image_grid_thw=torch.tensor([[1, 4, 4]]))
    out.float().sum().backward()
    assert v.model.visual.blocks[0].attn.qkv.weight.grad is not None


def test_flash_attention_head_dim_sweep():
    """fwd+bwd parity at every kernel head-dim pair: 64
    (GPT-OSS), 96, 128 (Llama), (192,128) MLA, 192, 256 (Gemma-2/3), plus a
    padded odd dim (80 -> 96)."""
    from automodel_amd.ops.attention import attention_ref, flash_attention

    torch.manual_seed(3)
    cases = [(64, 64), (96, 96), (128, 128), (192, 128), (192, 192),
             (256, 256), (80, 80)]
    B, S, Hq, Hk = 1, 256, 4, 2
    for dqk, dv in cases:
        q = torch.randn(B, S, Hq, dqk, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True)
        k = torch.randn(B, S, Hk, dqk, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True)
        v = torch.randn(B, S, Hk, dv, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True)
        scale = dqk ** -0.5
        o = flash_attention(q, k, v, causal=True, backend="hip", scale=scale)
        q2 = q.detach().float().requires_grad_(True)
        k2 = k.detach().float().requires_grad_(True)
        v2 = v.detach().float().requires_grad_(True)
        o_ref = attention_ref(q2, k2, v2, causal=True, scale=scale)
        assert torch.allclose(o.float(), o_ref, atol=3e-2, rtol=3e-2), \
            (dqk, dv, (o.float() - o_ref).abs().max())
        do = torch.randn_like(o)
        o.backward(do)
        o_ref.backward(do.float())
        for g, g2, name in [(q.grad, q2.grad, "dq"), (k.grad, k2.grad, "dk"),
                            (v.grad, v2.grad, "dv")]:
            assert torch.allclose(g.float(), g2, atol=6e-2, rtol=6e-2), \
                (dqk, dv, name, (g.float() - g2).abs().max())


def test_flash_attention_bwd_deterministic():
    """Two identical backward runs produce BITWISE-equal grads (VERDICT r1
    #10: the r1 dKV kernel used f32 atomicAdd across the GQA group)."""
    from automodel_amd.ops.attention import flash_attention

    torch.manual_seed(4)
    B, S, Hq, Hk, D = 2, 512, 8, 2, 128     # GQA group 4 -> old kernel raced
    q0 = torch.randn(B, S, Hq, D, device="cuda", dtype=torch.bfloat16)
    k0 = torch.randn(B, S, Hk, D, device="cuda", dtype=torch.bfloat16)
    v0 = torch.randn(B, S, Hk, D, device="cuda", dtype=torch.bfloat16)
    do = torch.randn(B, S, Hq, D, device="cuda", dtype=torch.bfloat16)
    grads = []
    for _ in range(2):
        q = q0.clone().requires_grad_(True)
        k = k0.clone().requires_grad_(True)
        v = v0.clone().requires_grad_(True)
        o = flash_attention(q, k, v, causal=True, backend="hip")
        o.backward(do)
        grads.append((q.grad.clone(), k.grad.clone(), v.grad.clone()))
    for a, b, name in zip(grads[0], grads[1], ("dq", "dk", "dv")):
        assert torch.equal(a, b), f"{name} not bitwise deterministic"


def test_varlen_attention_one_launch_bwd():
    """Native varlen (doc bounds in-kernel, ONE launch) fwd+bwd parity vs
    per-document fp32 reference."""
    from automodel_amd.ops.attention import attention_ref, flash_attention_varlen

    torch.manual_seed(5)
    H, Hk, D = 4, 2, 128
    cu = torch.tensor([0, 100, 356, 420, 1024], dtype=torch.int32)
    T = 1024
    q = torch.randn(1, T, H, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    k = torch.randn(1, T, Hk, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    v = torch.randn(1, T, Hk, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    out = flash_attention_varlen(q, k, v, cu, backend="hip")
    do = torch.randn_like(out)
    out.backward(do)

    q2 = q.detach().float().requires_grad_(True)
    k2 = k.detach().float().requires_grad_(True)
    v2 = v.detach().float().requires_grad_(True)
    refs = []
    for a, b in zip(cu[:-1].tolist(), cu[1:].tolist()):
        refs.append(attention_ref(q2[:, a:b], k2[:, a:b], v2[:, a:b], causal=True))
    ref = torch.cat(refs, dim=1)
    assert torch.allclose(out.float(), ref, atol=3e-2, rtol=3e-2), \
        (out.float() - ref).abs().max()
    ref.backward(do.float())
    assert torch.allclose(q.grad.float(), q2.grad, atol=6e-2, rtol=6e-2), \
        (q.grad.float() - q2.grad).abs().max()
    assert torch.allclose(k.grad.float(), k2.grad, atol=6e-2, rtol=6e-2), \
        (k.grad.float() - k2.grad).abs().max()
    assert torch.allclose(v.grad.float(), v2.grad, atol=6e-2, rtol=6e-2), \
        (v.grad.float() - v2.grad).abs().max()


def test_varlen_attention_short_docs_many():
    """Launch-bound regression shape: many short docs, one kernel launch."""
    from automodel_amd.ops.attention import attention_ref, flash_attention_varlen

    torch.manual_seed(6)
    H, Hk, D = 2, 1, 64
    lens = [17, 40, 3, 128, 64, 9, 251, 32] * 2
    cu = torch.tensor([0] + list(torch.tensor(lens).cumsum(0)), dtype=torch.int32)
    T = int(cu[-1])
    q = torch.randn(1, T, H, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(1, T, Hk, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(1, T, Hk, D, device="cuda", dtype=torch.bfloat16)
    out = flash_attention_varlen(q, k, v, cu, backend="hip")
    for a, b in zip(cu[:-1].tolist(), cu[1:].tolist()):
        ref = attention_ref(q[:, a:b].float(), k[:, a:b].float(), v[:, a:b].float(),
                            causal=True)
        assert torch.allclose(out[:, a:b].float(), ref, atol=3e-2, rtol=3e-2), (a, b)


def test_mla_attention_flash_gpu():
    """DeepSeek MLA block runs on the (192,128) flash instantiation and
    matches its own sdpa path."""
    from automodel_amd.models.registry import build_model

    torch.manual_seed(7)
    cfg = dict(vocab_size=256, hidden_size=64, intermediate_size=128,
               num_hidden_layers=1, num_attention_heads=2,
               kv_lora_rank=32, q_lora_rank=0,
               qk_nope_head_dim=128, qk_rope_head_dim=64, v_head_dim=128,
               max_position_embeddings=512,
               moe=dict(n_routed_experts=4, n_activated_experts=2,
                        moe_intermediate_size=48), first_k_dense_replace=1)
    m = build_model(config=cfg, architecture="DeepseekV3ForCausalLM",
                    dtype="bfloat16", meta_init=False).to("cuda")
    ids = torch.randint(0, 256, (1, 256), device="cuda")
    with torch.no_grad():
        logits_hip = m(ids)
    # force the sdpa path for comparison
    for layer in m.model.layers:
        layer.self_attn.backend = layer.self_attn.backend.__class__(attn="sdpa")
    with torch.no_grad():
        logits_sdpa = m(ids)
    assert torch.allclose(logits_hip.float(), logits_sdpa.float(),
                          atol=5e-2, rtol=5e-2), \
        (logits_hip.float() - logits_sdpa.float()).abs().max()


def test_fused_ce_single_kernel_parity():
    """hip_fused (single-pass MFMA stats GEMM forward, no logits tensor) vs
    fp32 dense reference, fwd + bwd."""
    from automodel_amd.loss.linear_ce import fused_linear_cross_entropy

    torch.manual_seed(0)
    for T, H, V in [(512, 256, 1024), (384, 64, 4000), (129, 128, 296)]:
        hidden = torch.randn(T, H, device="cuda", dtype=torch.bfloat16,
                             requires_grad=True)
        weight = (torch.randn(V, H, device="cuda", dtype=torch.bfloat16) * 0.05
                  ).requires_grad_(True)
        labels = torch.randint(0, V, (T,), device="cuda")
        labels[7] = -100
        loss = fused_linear_cross_entropy(hidden, weight, labels,
                                          backend="hip_fused", chunk_size=128)
        loss.backward()

        h2 = hidden.detach().float().requires_grad_(True)
        w2 = weight.detach().float().requires_grad_(True)
        ref = torch.nn.functional.cross_entropy(h2 @ w2.t(), labels,
                                                ignore_index=-100, reduction="sum")
        ref.backward()
        assert torch.allclose(loss, ref, rtol=2e-2), (T, H, V, float(loss), float(ref))
        assert torch.allclose(hidden.grad.float(), h2.grad, atol=5e-2, rtol=5e-2), \
            (T, H, V, (hidden.grad.float() - h2.grad).abs().max())
        assert torch.allclose(weight.grad.float(), w2.grad, atol=5e-2, rtol=5e-2), \
            (T, H, V, (weight.grad.float() - w2.grad).abs().max())


def test_fused_ce_deterministic():
    """Two identical hip_fused runs give bitwise-equal loss and grads."""
    from automodel_amd.loss.linear_ce import fused_linear_cross_entropy

    torch.manual_seed(1)
    T, H, V = 1024, 256, 8192
    h0 = torch.randn(T, H, device="cuda", dtype=torch.bfloat16)
    w0 = torch.randn(V, H, device="cuda", dtype=torch.bfloat16) * 0.05
    y = torch.randint(0, V, (T,), device="cuda")
    outs = []
    for _ in range(2):
        h = h0.clone().requires_grad_(True)
        w = w0.clone().requires_grad_(True)
        loss = fused_linear_cross_entropy(h, w, y, backend="hip_fused",
                                          chunk_size=256)
        loss.backward()
        outs.append((loss.detach().clone(), h.grad.clone(), w.grad.clone()))
    assert torch.equal(outs[0][0], outs[1][0])
    assert torch.equal(outs[0][1], outs[1][1])
    assert torch.equal(outs[0][2], outs[1][2])


def test_blockdiag_cp_varlen_kernel_long_seq():
    """Blockdiag CP on the varlen kernel at S=16k: no O(S^2) mask (VERDICT
    r1 weak #6). World-1 CP group exercises the full gather+q_start+doc
    path; parity vs the one-shot varlen kernel on the same data."""
    import torch.distributed as dist

    from automodel_amd.ops.attention import flash_attention_varlen
    from automodel_amd.parallel.cp import (
        cp_blockdiag_attention,
        disable_cp,
        enable_cp,
    )

    if not dist.is_initialized():
        import os
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29511")
        dist.init_process_group("gloo", rank=0, world_size=1)
    enable_cp(dist.group.WORLD)
    try:
        torch.manual_seed(0)
        T, H, Hk, D = 16384, 2, 1, 128
        cu = torch.tensor([0, 5000, 9000, 16384], dtype=torch.int32)
        q = torch.randn(1, T, H, D, device="cuda", dtype=torch.bfloat16)
        k = torch.randn(1, T, Hk, D, device="cuda", dtype=torch.bfloat16)
        v = torch.randn(1, T, Hk, D, device="cuda", dtype=torch.bfloat16)
        import gc
        gc.collect()
        torch.cuda.empty_cache()
        torch.cuda.reset_peak_memory_stats()
        base = torch.cuda.memory_allocated()
        out = cp_blockdiag_attention(q, k, v, cu)
        peak = torch.cuda.max_memory_allocated() - base
        # dense [S_local, T] bf16 mask alone would be 16384*16384*2 = 512 MB
        assert peak < 400 * 2**20, f"peak delta {peak/2**20:.0f} MB — dense mask is back?"
        ref = flash_attention_varlen(q, k, v, cu, backend="hip")
        assert torch.allclose(out.float(), ref.float(), atol=3e-2, rtol=3e-2), \
            (out.float() - ref.float()).abs().max()
    finally:
        disable_cp()


def test_grouped_gemm_bwd_kernels_parity():
    """Single-kernel grouped backward (dx NN + dw TN) vs per-expert fp32
    loop at DeepSeek-style many-small-expert shapes."""
    from automodel_amd.ops.grouped_gemm import grouped_linear

    torch.manual_seed(0)
    E, N, K = 16, 128, 256
    counts = [37, 0, 128, 5, 64, 200, 1, 99, 31, 17, 0, 256, 77, 3, 40, 70]
    M = sum(counts)
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    w = (torch.randn(E, N, K, device="cuda", dtype=torch.bfloat16) * 0.05
         ).requires_grad_(True)
    y = grouped_linear(x, w, counts)
    dy = torch.randn_like(y)
    y.backward(dy)

    x2 = x.detach().float().requires_grad_(True)
    w2 = w.detach().float().requires_grad_(True)
    outs, start = [], 0
    for e, n in enumerate(counts):
        outs.append(x2[start:start + n] @ w2[e].t())
        start += n
    ref = torch.cat(outs)
    ref.backward(dy.float())
    assert torch.allclose(y.float(), ref, atol=3e-2, rtol=3e-2), \
        (y.float() - ref).abs().max()
    assert torch.allclose(x.grad.float(), x2.grad, atol=5e-2, rtol=5e-2), \
        (x.grad.float() - x2.grad).abs().max()
    assert torch.allclose(w.grad.float(), w2.grad, atol=8e-2, rtol=8e-2), \
        (w.grad.float() - w2.grad).abs().max()


def test_grouped_gemm_bwd_deterministic():
    """Grouped backward is bitwise reproducible (plain stores, m-loop in
    block)."""
    from automodel_amd.ops.grouped_gemm import grouped_linear

    torch.manual_seed(1)
    E, N, K = 8, 256, 128
    counts = [100, 28, 300, 0, 64, 129, 55, 324]
    M = sum(counts)
    x0 = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    w0 = torch.randn(E, N, K, device="cuda", dtype=torch.bfloat16) * 0.05
    dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
    grads = []
    for _ in range(2):
        x = x0.clone().requires_grad_(True)
        w = w0.clone().requires_grad_(True)
        grouped_linear(x, w, counts).backward(dy)
        grads.append((x.grad.clone(), w.grad.clone()))
    assert torch.equal(grads[0][0], grads[1][0])
    assert torch.equal(grads[0][1], grads[1][1])


def test_fused_dispatch_combine_parity():
    """HIP gather/combine MoE glue (no repeat_interleave, no host sync) vs
    the torch permute path: fwd + grads through x, probs, and experts."""
    from automodel_amd.moe.experts import (
        GroupedExperts,
        fused_dispatch_combine,
        permute_tokens,
        unpermute_tokens,
    )

    torch.manual_seed(0)
    T, H, E, K, inter = 512, 256, 8, 2, 128
    x0 = torch.randn(T, H, device="cuda", dtype=torch.bfloat16)
    probs0 = torch.rand(T, K, device="cuda", dtype=torch.float32)
    probs0 = probs0 / probs0.sum(-1, keepdim=True)
    indices = torch.randint(0, E, (T, K), device="cuda")
    experts = GroupedExperts(E, H, inter).to("cuda", torch.bfloat16)
    experts.init_weights()

    x = x0.clone().requires_grad_(True)
    probs = probs0.clone().requires_grad_(True)
    out = fused_dispatch_combine(x, probs, indices, E, experts.forward_permuted)
    gout = torch.randn_like(out)
    out.backward(gout)

    x2 = x0.clone().requires_grad_(True)
    probs2 = probs0.clone().requires_grad_(True)
    x_perm, sort_idx, counts = permute_tokens(x2, indices, E)
    y_perm = experts.forward_permuted(x_perm, counts)
    ref = unpermute_tokens(y_perm, sort_idx, probs2)
    ref.backward(gout)

    assert torch.allclose(out.float(), ref.float(), atol=3e-2, rtol=3e-2), \
        (out.float() - ref.float()).abs().max()
    assert torch.allclose(x.grad.float(), x2.grad.float(), atol=5e-2, rtol=5e-2), \
        (x.grad.float() - x2.grad.float()).abs().max()
    assert torch.allclose(probs.grad, probs2.grad, atol=5e-2, rtol=5e-2), \
        (probs.grad - probs2.grad).abs().max()


def test_adamw_bf16_state_kernel():
    """Master-free bf16-state AdamW kernel follows the fp32 trajectory to
    bf16 quantization."""
    from automodel_amd.optim.adamw import FusedAdamW

    torch.manual_seed(0)
    w0 = torch.randn(8192, device="cuda")
    p_bf = torch.nn.Parameter(w0.clone().to(torch.bfloat16))
    p_ref = torch.nn.Parameter(w0.clone())
    opt_bf = FusedAdamW([p_bf], lr=1e-2, betas=(0.9, 0.95), weight_decay=0.1,
                        state_dtype=torch.bfloat16)
    opt_ref = torch.optim.AdamW([p_ref], lr=1e-2, betas=(0.9, 0.95), eps=1e-8,
                                weight_decay=0.1)
    for _ in range(5):
        g = torch.randn(8192, device="cuda")
        p_bf.grad = g.to(torch.bfloat16)
        p_ref.grad = g.clone()
        opt_bf.step()
        opt_ref.step()
    assert torch.allclose(p_bf.float(), p_ref.detach(), atol=5e-2, rtol=5e-2), \
        (p_bf.float() - p_ref.detach()).abs().max()


def test_lora_fused_kernel_parity():
    """Fused single-kernel LoRA forward (x@A^T@B^T*s) vs the composite
    two-GEMM path, fwd + grads (reference lora_kernel.py:182 equivalent)."""
    from automodel_amd.ops._backend import hip_ops

    torch.manual_seed(0)
    for M, H, O, r in [(512, 256, 512, 32), (1024, 4096, 6144, 64), (300, 128, 64, 32)]:
        x = torch.randn(M, H, device="cuda", dtype=torch.bfloat16)
        A = torch.randn(r, H, device="cuda", dtype=torch.bfloat16) * 0.05
        B = torch.randn(O, r, device="cuda", dtype=torch.bfloat16) * 0.05
        out = hip_ops().lora_fused_fwd(x, A, B, 0.5)
        ref = (x.float() @ A.float().t() @ B.float().t()) * 0.5
        assert torch.allclose(out.float(), ref, atol=3e-2, rtol=3e-2), \
            (M, H, O, r, (out.float() - ref).abs().max())


def test_lora_module_fused_path_grads():
    """LinearLoRA fused path end-to-end: output + adapter grads match the
    composite path."""
    from automodel_amd.peft.lora import LinearLoRA

    torch.manual_seed(1)
    base = torch.nn.Linear(256, 512, bias=False, device="cuda", dtype=torch.bfloat16)
    lora = LinearLoRA(base, dim=32, alpha=64.0).to("cuda", torch.bfloat16)
    with torch.no_grad():
        torch.nn.init.normal_(lora.lora_B.weight, std=0.05)
    x = torch.randn(4, 64, 256, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    y = lora(x)                                 # fused (eval-mode dropout)
    y.float().sum().backward()
    gA, gB, gx = (lora.lora_A.weight.grad.clone(), lora.lora_B.weight.grad.clone(),
                  x.grad.clone())
    lora.zero_grad(); x.grad = None
    x2 = x.detach().requires_grad_(True)
    ref = lora.base(x2) + lora.lora_B(lora.lora_A(x2)) * lora.scale
    ref.float().sum().backward()
    assert torch.allclose(y.float(), ref.float(), atol=3e-2, rtol=3e-2)
    assert torch.allclose(gA.float(), lora.lora_A.weight.grad.float(), atol=5e-2, rtol=5e-2)
    assert torch.allclose(gB.float(), lora.lora_B.weight.grad.float(), atol=5e-2, rtol=5e-2)
    assert torch.allclose(gx.float(), x2.grad.float(), atol=5e-2, rtol=5e-2)


def test_hybrid_and_llama4_gpu_smoke():
    """Round-2 families forward+backward on MI355X bf16: Nemotron-H hybrid
    (Mamba2 chunked scan) and Llama-4 (MoE + NoPE temperature tuning)."""
    from automodel_amd.models.llama4.model import Llama4Config, Llama4ForCausalLM
    from automodel_amd.models.nemotron_h.model import (
        NemotronHConfig,
        NemotronHForCausalLM,
    )

    torch.manual_seed(0)
    nh = NemotronHForCausalLM(NemotronHConfig(
        vocab_size=256, hidden_size=128, intermediate_size=256,
        num_hidden_layers=3, hybrid_override_pattern="M*-",
        num_attention_heads=2, num_key_value_heads=1, head_dim=64,
        mamba_num_heads=4, mamba_head_dim=32, ssm_state_size=16,
        n_groups=2, chunk_size=64))
    nh.init_weights(device="cuda")
    nh = nh.to(torch.bfloat16)
    ids = torch.randint(0, 256, (2, 128), device="cuda")
    nh(ids).float().sum().backward()
    assert nh.model.layers[0].mixer.in_proj.weight.grad is not None

    l4 = Llama4ForCausalLM(Llama4Config(
        vocab_size=256, hidden_size=128, intermediate_size=128,
        intermediate_size_mlp=256, num_hidden_layers=2,
        num_attention_heads=2, num_key_value_heads=1, head_dim=64,
        num_local_experts=4, num_experts_per_tok=2,
        interleave_moe_layer_step=2, max_position_embeddings=256))
    l4.init_weights(device="cuda")
    l4 = l4.to(torch.bfloat16)
    l4(ids).float().sum().backward()
    assert l4.model.layers[1].feed_forward.router.weight.grad is not None


@pytest.mark.gpu
def test_grouped_gemm_fp8_numerics():
    """fp8-e4m3 grouped NT vs fp32 reference: tensorwise-scaled error stays
    within e4m3 budget; both 128 and 256 tiles agree with each other."""
    from automodel_amd.ops._backend import hip_ops
    from automodel_amd.ops.grouped_gemm import make_group_plan

    ops = hip_ops()
    torch.manual_seed(0)
    E, M, H, N = 8, 1024, 256, 512
    counts = torch.full((E,), M // E, dtype=torch.int32, device="cuda")
    x = torch.randn(M, H, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(E, N, H, device="cuda", dtype=torch.bfloat16) * 0.05
    offs, tm, ntl = make_group_plan(counts, M)
    dummy = torch.zeros(1, device="cuda")
    sx = (448.0 / x.abs().amax().float().clamp(min=1e-12)).reshape(1)
    sw = (448.0 / w.abs().amax().float().clamp(min=1e-12)).reshape(1)
    x8 = ops.fp8_cast(x, sx, dummy, False)
    w8 = ops.fp8_cast(w.view(-1, H), sw, dummy, False).view(E, N, H)
    deq = (sx * sw).reciprocal()
    y8 = ops.grouped_gemm_nt_fp8(x8, w8, offs, tm, deq, ntl)
    # fp32 reference per group
    refs = []
    for e in range(E):
        refs.append(x[e * (M // E):(e + 1) * (M // E)].float() @ w[e].float().t())
    ref = torch.cat(refs)
    rel = (y8.float() - ref).abs().max() / ref.abs().max()
    assert float(rel) < 0.06, float(rel)
    # big-tile agreement
    offs_b, tm_b, ntl_b = ops.build_group_plan(counts, M, 256)
    y8b = ops.grouped_gemm_nt_fp8(x8, w8, offs_b, tm_b, deq, ntl_b, 256)
    torch.testing.assert_close(y8b, y8, atol=0, rtol=0)


@pytest.mark.gpu
def test_grouped_linear_fp8_autograd():
    """grouped_linear_fp8: fp8 forward close to bf16, backward = exact bf16
    grouped grads (bwd path shares the bf16 kernels)."""
    from automodel_amd.ops.grouped_gemm import (
        Fp8GroupedState,
        grouped_linear,
        grouped_linear_fp8,
        make_group_plan,
    )

    torch.manual_seed(1)
    E, M, H, N = 4, 512, 256, 256
    counts = torch.full((E,), M // E, dtype=torch.int32, device="cuda")
    x = torch.randn(M, H, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    w = (torch.randn(E, N, H, device="cuda", dtype=torch.bfloat16) * 0.05
         ).requires_grad_()
    plan = (*make_group_plan(counts, M, 128), 128)
    st = Fp8GroupedState()
    y8 = grouped_linear_fp8(x, w, counts, plan, st)
    yb = grouped_linear(x.detach().clone().requires_grad_(), w, counts, plan)
    rel = (y8.float() - yb.float()).abs().max() / yb.float().abs().max()
    assert float(rel) < 0.05, float(rel)
    g = torch.randn_like(y8)
    y8.backward(g)
    assert x.grad is not None and w.grad is not None
    assert torch.isfinite(x.grad).all() and torch.isfinite(w.grad).all()


@pytest.mark.gpu
def test_sgmv_multi_adapter_numerics():
    """SGMV fused kernel vs per-adapter fp32 reference — mixed adapter ids,
    unsorted tokens, per-adapter scales, r in {32, 64}."""
    from automodel_amd.peft.sgmv import sgmv_delta

    torch.manual_seed(0)
    for r in (32, 64):
        T, H, O, n = 777, 256, 192, 5
        x = torch.randn(T, H, device="cuda", dtype=torch.bfloat16)
        A = torch.randn(n, r, H, device="cuda", dtype=torch.bfloat16) * 0.05
        B = torch.randn(n, O, r, device="cuda", dtype=torch.bfloat16) * 0.05
        ids = torch.randint(0, n, (T,), device="cuda")
        scales = [0.5 + 0.25 * a for a in range(n)]
        y = sgmv_delta(x, A, B, ids, scales)
        ref = torch.zeros(T, O, device="cuda")
        for a in range(n):
            m = ids == a
            ref[m] = (x[m].float() @ A[a].float().t()) @ B[a].float().t() * scales[a]
        err = (y.float() - ref).abs().max() / ref.abs().max().clamp(min=1e-6)
        assert float(err) < 0.02, (r, float(err))
