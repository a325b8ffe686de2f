"""QLoRA 70B-on-one-GPU smoke: build llama3-70b (random init), NF4-quantize
the attention/MLP linears under LoRA adapters, run forward+backward+step.
Demonstrates the 288 GB HBM3E single-GPU finetune story (bf16 70B = 140 GB
full; NF4 base ~35 GB + bf16 embeddings/head + rank-16 adapters)."""

import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    from automodel_amd.models.llama.model import LlamaForCausalLM
    from automodel_amd.peft.lora import apply_lora_to_linear_modules

    assert torch.cuda.is_available()
    t0 = time.perf_counter()
    cfg = dict(vocab_size=128256, hidden_size=8192, intermediate_size=28672,
               num_hidden_layers=80, num_attention_heads=64,
               num_key_value_heads=8, rope_theta=500000.0,
               max_position_embeddings=8192)
    model = LlamaForCausalLM(cfg)
    model.init_weights(device="cuda")
    model = model.to(torch.bfloat16)
    t_init = time.perf_counter() - t0
    full_gb = torch.cuda.memory_allocated() / 2**30

    t0 = time.perf_counter()
    n = apply_lora_to_linear_modules(model, {
        "target_modules": ["*q_proj", "*k_proj", "*v_proj", "*o_proj",
                           "*gate_proj", "*up_proj", "*down_proj"],
        "dim": 16, "alpha": 32, "quantize_base": True,
    })
    torch.cuda.empty_cache()
    t_quant = time.perf_counter() - t0
    quant_gb = torch.cuda.memory_allocated() / 2**30

    from automodel_amd.loss.linear_ce import FusedLinearCrossEntropy

    model.loss_fn = FusedLinearCrossEntropy(backend="hybrid", chunk_size=2048)
    opt = torch.optim.AdamW([p for p in model.parameters() if p.requires_grad],
                            lr=1e-4)
    ids = torch.randint(0, 128256, (1, 512), device="cuda")
    t0 = time.perf_counter()
    loss = model(ids, labels=ids.clone())
    loss.backward()
    opt.step()
    torch.cuda.synchronize()
    t_step = time.perf_counter() - t0
    peak_gb = torch.cuda.max_memory_allocated() / 2**30
    print(json.dumps({
        "model": "llama3-70b (random init)", "mode": "qlora-nf4 rank16",
        "adapted_linears": n, "init_s": round(t_init, 1),
        "quantize_s": round(t_quant, 1), "step_s": round(t_step, 1),
        "mem_full_bf16_gb": round(full_gb, 1),
        "mem_after_nf4_gb": round(quant_gb, 1),
        "mem_peak_gb": round(peak_gb, 1),
        "loss": round(float(loss.detach()), 2),
    }))


if __name__ == "__main__":
    main()
