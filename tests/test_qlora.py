"""NF4 quantization + QLoRA: round-trip quality, NF4Linear parity,
adapter training with quantized base, merge, and the HIP dequant kernel."""

import pytest
import torch
import torch.nn as nn

from automodel_amd.quantization.nf4 import (
    NF4Linear,
    dequantize_nf4,
    quantize_linear_modules,
    quantize_nf4,
)


def test_nf4_roundtrip_error():
    torch.manual_seed(0)
    w = torch.randn(256, 128)
    packed, absmax = quantize_nf4(w, block_size=64)
    assert packed.dtype == torch.uint8 and packed.numel() == w.numel() // 2
    assert absmax.numel() == w.numel() // 64
    deq = dequantize_nf4(packed, absmax, w.shape, 64)
    # NF4 with block 64 on gaussian weights: relative error ~ a few percent
    rel = (deq - w).norm() / w.norm()
    assert rel < 0.10, rel.item()
    # per-block max is representable exactly (|max| maps to code +-1.0)
    blocks = w.reshape(-1, 64)
    deq_blocks = deq.reshape(-1, 64)
    amax_idx = blocks.abs().argmax(dim=1)
    rows = torch.arange(blocks.shape[0])
    torch.testing.assert_close(deq_blocks[rows, amax_idx], blocks[rows, amax_idx],
                               atol=1e-5, rtol=1e-5)


def test_nf4_linear_matches_dequant():
    torch.manual_seed(1)
    base = nn.Linear(128, 64)
    q = NF4Linear(base)
    x = torch.randn(4, 128)
    y = q(x)
    w = dequantize_nf4(q.weight_packed, q.weight_absmax, (64, 128), 64)
    torch.testing.assert_close(y, torch.nn.functional.linear(x, w, base.bias))
    # memory: packed buffer is 0.5 byte per weight
    assert q.weight_packed.numel() == 128 * 64 // 2


def test_quantize_linear_modules_skips_lm_head():
    m = nn.ModuleDict({
        "proj": nn.Linear(64, 64),
        "lm_head": nn.Linear(64, 64),
    })
    n = quantize_linear_modules(m)
    assert n == 1
    assert isinstance(m["proj"], NF4Linear)
    assert isinstance(m["lm_head"], nn.Linear)


def test_qlora_apply_and_train():
    from automodel_amd.peft.lora import LinearLoRA, apply_lora_to_linear_modules

    torch.manual_seed(2)

    class Tiny(nn.Module):
        def __init__(self):
            super().__init__()
            self.q_proj = nn.Linear(32, 32)
            self.o_proj = nn.Linear(32, 32)
            self.lm_head = nn.Linear(32, 8)

        def forward(self, x):
            return self.lm_head(self.o_proj(self.q_proj(x)))

    model = Tiny()
    n = apply_lora_to_linear_modules(
        model,
        {"target_modules": ["*q_proj", "*o_proj"], "dim": 4, "alpha": 8,
         "quantize_base": True},
    )
    assert n == 2
    assert isinstance(model.q_proj, LinearLoRA)
    assert isinstance(model.q_proj.base, NF4Linear)
    trainable = [k for k, p in model.named_parameters() if p.requires_grad]
    assert all("lora_" in k for k in trainable) and len(trainable) == 4

    x = torch.randn(3, 32)
    y0 = model(x)
    loss = model(x).pow(2).sum()
    loss.backward()
    assert model.q_proj.lora_A.weight.grad is not None
    with torch.no_grad():
        for p in model.parameters():
            if p.requires_grad and p.grad is not None:
                p -= 0.05 * p.grad
    assert not torch.allclose(model(x), y0)  # adapters actually moved the output


def test_qlora_merge_dequantizes():
    from automodel_amd.peft.lora import LinearLoRA

    torch.manual_seed(3)
    base = nn.Linear(48, 48)
    lora = LinearLoRA(NF4Linear(base), dim=4, alpha=8)
    with torch.no_grad():
        nn.init.normal_(lora.lora_B.weight, std=0.1)
    x = torch.randn(2, 48)
    y_adapter = lora(x)
    merged = lora.merge()
    assert isinstance(merged, nn.Linear)
    torch.testing.assert_close(merged(x), y_adapter, atol=1e-5, rtol=1e-4)


def test_qlora_dora_with_nf4_base():
    from automodel_amd.peft.lora import LinearLoRA

    base = nn.Linear(32, 16)
    lora = LinearLoRA(NF4Linear(base), dim=2, alpha=4, use_dora=True)
    x = torch.randn(2, 32)
    w = lora._base_weight()
    torch.testing.assert_close(
        lora(x), torch.nn.functional.linear(x, w, base.bias), atol=1e-4, rtol=1e-4
    )


@pytest.mark.gpu
def test_nf4_dequant_kernel_parity():
    torch.manual_seed(4)
    for rows, cols in [(256, 128), (512, 96), (64, 64)]:
        w = torch.randn(rows, cols)
        packed, absmax = quantize_nf4(w, 64)
        ref = dequantize_nf4(packed, absmax, (rows, cols), 64)
        from automodel_amd.ops._backend import hip_ops

        out = hip_ops().nf4_dequant(packed.cuda(), absmax.cuda(), 64, rows, cols)
        torch.testing.assert_close(out.cpu().float(), ref, atol=1e-2, rtol=1e-2)


@pytest.mark.gpu
def test_nf4_linear_gpu_forward():
    torch.manual_seed(5)
    base = nn.Linear(256, 128)
    q = NF4Linear(base).cuda()
    x = torch.randn(8, 256, device="cuda", dtype=torch.bfloat16)
    y = q(x)
    w = dequantize_nf4(q.weight_packed.cpu(), q.weight_absmax.cpu(), (128, 256), 64)
    ref = torch.nn.functional.linear(x.cpu().float(), w, base.bias)
    torch.testing.assert_close(y.cpu().float(), ref, atol=5e-2, rtol=5e-2)


def test_decode_linear_cpu_fallthrough():
    """On CPU / under grad, DecodeLinear must behave exactly like Linear."""
    from automodel_amd.serving import DecodeLinear, swap_linears_for_decode

    torch.manual_seed(7)
    m = nn.Sequential(nn.Linear(512, 64), nn.ReLU(), nn.Linear(64, 512))
    ref = nn.Sequential(nn.Linear(512, 64), nn.ReLU(), nn.Linear(64, 512))
    ref.load_state_dict(m.state_dict())
    n = swap_linears_for_decode(m)
    assert n == 2 and isinstance(m[0], DecodeLinear)
    x = torch.randn(3, 512)
    torch.testing.assert_close(m(x), ref(x))
    m(x).sum().backward()  # grad path works


@pytest.mark.gpu
def test_gemv_kernel_parity():
    from automodel_amd.serving import gemv_bf16

    torch.manual_seed(8)
    for B, K, N in [(1, 4096, 1024), (8, 512, 300), (5, 1024, 256), (16, 512, 128)]:
        x = torch.randn(B, K, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
        bias = torch.randn(N, device="cuda", dtype=torch.bfloat16)
        y = gemv_bf16(x, w, bias)
        ref = torch.nn.functional.linear(x.float(), w.float(), bias.float())
        torch.testing.assert_close(y.float(), ref, atol=5e-2, rtol=5e-2)
        y2 = gemv_bf16(x, w, None)
        torch.testing.assert_close(y2.float(), ref - bias.float(), atol=5e-2, rtol=5e-2)


def test_qlora_adapter_dtype_matches_base():
    """Adapters under an NF4 base must inherit the base's compute dtype
    (bf16 base -> bf16 adapters), or decode-dtype GEMMs fail."""
    from automodel_amd.peft.lora import LinearLoRA

    base = nn.Linear(64, 32).to(torch.bfloat16)
    lora = LinearLoRA(NF4Linear(base), dim=4, alpha=8)
    assert lora.lora_A.weight.dtype == torch.bfloat16
    x = torch.randn(2, 64, dtype=torch.bfloat16)
    assert lora(x).dtype == torch.bfloat16
