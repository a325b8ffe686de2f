"""Decode-throughput microbench: KV-cached greedy generation on one MI355X.

Measures prefill time and per-token decode latency / throughput for the
flagship config. Decode is bandwidth-bound (reads every weight once per
token: 16 GB bf16 for 8B -> theoretical floor ~2 ms/token at 8 TB/s);
this bench records how close the stack gets. Run:

    python benchmarks/decode_bench.py --model llama3_8b --batch 8 --new 64
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

CONFIGS = {
    "llama3_8b": dict(vocab_size=128256, hidden_size=4096, intermediate_size=14336,
                      num_hidden_layers=32, num_attention_heads=32,
                      num_key_value_heads=8, rope_theta=500000.0,
                      max_position_embeddings=8192),
    "tiny": dict(vocab_size=1024, hidden_size=256, intermediate_size=512,
                 num_hidden_layers=4, num_attention_heads=4,
                 num_key_value_heads=2, max_position_embeddings=2048),
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3_8b", choices=sorted(CONFIGS))
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--prompt", type=int, default=512)
    ap.add_argument("--new", type=int, default=64)
    ap.add_argument("--graphed", action="store_true",
                    help="decode via generate_graphed (hipGraph-captured step)")
    ap.add_argument("--gemv", action="store_true",
                    help="swap linears for the in-tree decode GEMV kernel")
    args = ap.parse_args()

    from automodel_amd.models.llama.model import LlamaForCausalLM
    from automodel_amd.utils.generation import generate_cached, generate_graphed
    from automodel_amd.utils.kv_cache import KVCache, kv_cache_context

    dev = "cuda" if torch.cuda.is_available() else "cpu"
    dtype = torch.bfloat16 if dev == "cuda" else torch.float32
    torch.manual_seed(0)
    cfg = CONFIGS[args.model]
    model = LlamaForCausalLM(cfg)
    model.init_weights(device=dev)
    model = model.to(dtype).eval()

    ids = torch.randint(0, cfg["vocab_size"], (args.batch, args.prompt), device=dev)

    # warmup (graph/workspace/first-launch)
    generate_cached(model, ids[:, :64], max_new_tokens=4)
    if dev == "cuda":
        torch.cuda.synchronize()

    if args.gemv:
        from automodel_amd.serving import swap_linears_for_decode

        n = swap_linears_for_decode(model)
        print(f"# swapped {n} linears for DecodeLinear", flush=True)

    if args.graphed:
        t0 = time.perf_counter()
        out = generate_graphed(model, ids, max_new_tokens=args.new)
        if dev == "cuda":
            torch.cuda.synchronize()
        total_s = time.perf_counter() - t0
        # marginal-replay cost: two runs differing only in token count —
        # prefill + capture cancel, leaving pure per-token replay time
        t0 = time.perf_counter()
        out = generate_graphed(model, ids, max_new_tokens=64)
        if dev == "cuda":
            torch.cuda.synchronize()
        t64_s = time.perf_counter() - t0
        t0 = time.perf_counter()
        out = generate_graphed(model, ids, max_new_tokens=args.new)
        if dev == "cuda":
            torch.cuda.synchronize()
        total2_s = time.perf_counter() - t0
        replay_ms = (total2_s - t64_s) / max(1, args.new - 64) * 1e3
        print(json.dumps({
            "model": args.model, "batch": args.batch,
            "mode": "graphed+gemv" if args.gemv else "graphed",
            "prompt_len": args.prompt, "new_tokens": args.new,
            "first_run_s": round(total_s, 3),
            "steady_ms_per_token": round(total2_s / args.new * 1e3, 2),
            "replay_ms_per_token": round(replay_ms, 2),
            "replay_decode_tokens_per_s": round(args.batch / replay_ms * 1e3, 1),
            "dtype": str(dtype).split(".")[-1], "device": dev,
        }))
        return

    # prefill timing (no_grad: decode must not build an autograd graph)
    cache = KVCache.for_model(model, args.batch, args.prompt + args.new)
    with torch.no_grad(), kv_cache_context(cache):
        t0 = time.perf_counter()
        cache.begin_forward()
        logits = model(ids)
        if dev == "cuda":
            torch.cuda.synchronize()
        prefill_s = time.perf_counter() - t0
        cache.advance(args.prompt)

        # decode timing
        nxt = logits[:, -1].argmax(-1, keepdim=True)
        t0 = time.perf_counter()
        for _ in range(args.new):
            pos = torch.arange(cache.pos, cache.pos + 1, device=dev).unsqueeze(0)
            cache.begin_forward()
            logits = model(nxt, position_ids=pos)
            cache.advance(1)
            nxt = logits[:, -1].argmax(-1, keepdim=True)
        if dev == "cuda":
            torch.cuda.synchronize()
        decode_s = time.perf_counter() - t0

    out = {
        "model": args.model,
        "batch": args.batch,
        "prompt_len": args.prompt,
        "new_tokens": args.new,
        "prefill_s": round(prefill_s, 4),
        "prefill_tokens_per_s": round(args.batch * args.prompt / prefill_s, 1),
        "decode_ms_per_step": round(decode_s / args.new * 1e3, 2),
        "decode_tokens_per_s": round(args.batch * args.new / decode_s, 1),
        "dtype": str(dtype).split(".")[-1],
        "device": dev,
    }
    print(json.dumps(out))


if __name__ == "__main__":
    main()
