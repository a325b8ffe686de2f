"""Minimal generation server: FastAPI /generate over the native decode stack.

Reference behavior: nemo_automodel/components/speculative/serve_target.py
(serve the target model for decode evaluation; production serving defers to
vLLM/SGLang). This server wraps the in-tree KV-cached / hipGraph decode
(utils/generation.py) behind an OpenAI-ish completion endpoint — enough for
eval loops and demo serving; continuous batching is round-2.

Run:  python -m automodel_amd.serving.server --config model.yaml --port 8000
Test: build_app(model, tokenizer) + fastapi.testclient (CPU-tested).
"""

# NOTE: no `from __future__ import annotations` — pydantic must resolve the
# locally-defined request model's unions at class-creation time.
import argparse
import threading

import torch


def build_app(model, tokenizer, use_cache: bool = True, use_graph: bool = False,
              proposer=None):
    from fastapi import FastAPI
    from pydantic import BaseModel

    from automodel_amd.utils.generation import (
        generate,
        generate_cached,
        generate_graphed,
    )

    app = FastAPI(title="automodel_amd generation server")
    # FastAPI runs sync endpoints in a threadpool; the kv-cache context
    # (utils/kv_cache._ACTIVE) is process-global, so only one generation may
    # be in flight at a time.
    gen_lock = threading.Lock()

    class GenRequest(BaseModel):
        prompt: str | None = None
        prompt_ids: list[int] | None = None
        max_new_tokens: int = 64
        temperature: float = 0.0
        speculative: bool = False      # greedy draft-verify (needs a proposer
                                       # or falls back to the ngram proposer)

    @app.get("/health")
    def health():
        p = next(model.parameters())
        return {"status": "ok", "device": str(p.device), "dtype": str(p.dtype),
                "n_params": sum(x.numel() for x in model.parameters())}

    @app.post("/generate")
    def gen(req: GenRequest):
        assert (req.prompt is None) != (req.prompt_ids is None), \
            "give prompt XOR prompt_ids"
        if req.prompt_ids is not None:
            ids = req.prompt_ids
        else:
            assert tokenizer is not None, "server started without a tokenizer"
            ids = tokenizer.encode(req.prompt)
        dev = next(model.parameters()).device
        x = torch.tensor([ids], dtype=torch.long, device=dev)
        with gen_lock, torch.no_grad():
            if req.speculative and req.temperature == 0.0:
                from automodel_amd.speculative.decode import (
                    NgramProposer,
                    speculative_generate,
                )

                prop = proposer if proposer is not None else NgramProposer()
                out, stats = speculative_generate(model, prop, x,
                                                  req.max_new_tokens)
                new = out[0, len(ids):].tolist()
                resp = {"prompt_len": len(ids), "output_ids": new,
                        "spec_acceptance_rate": stats.acceptance_rate,
                        "spec_tokens_per_call": stats.tokens_per_target_call}
                if tokenizer is not None:
                    resp["text"] = tokenizer.decode(new)
                return resp
            if use_graph and req.temperature == 0.0:
                out = generate_graphed(model, x, req.max_new_tokens)
            elif use_cache:
                out = generate_cached(model, x, req.max_new_tokens,
                                      temperature=req.temperature)
            else:
                out = generate(model, x, req.max_new_tokens,
                               temperature=req.temperature)
        new = out[0, len(ids):].tolist()
        resp = {"prompt_len": len(ids), "output_ids": new}
        if tokenizer is not None:
            resp["text"] = tokenizer.decode(new)
        return resp

    return app


def main(argv=None):
    import uvicorn

    from automodel_amd.config.loader import load_yaml_config
    from automodel_amd.models.registry import build_model

    ap = argparse.ArgumentParser()
    ap.add_argument("--config", required=True, help="model YAML (model: block)")
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--no-cache", action="store_true")
    ap.add_argument("--graph", action="store_true",
                    help="hipGraph-captured decode (greedy only)")
    ap.add_argument("--draft", default=None,
                    help="EAGLE draft checkpoint dir: serve speculative "
                         "decoding (reference serve_target.py role)")
    args = ap.parse_args(argv)

    cfg = load_yaml_config(args.config)
    device = "cuda" if torch.cuda.is_available() else "cpu"
    model = build_model(
        config=cfg.model.config.to_dict(),
        architecture=cfg.model.get("architecture"),
        pretrained_path=cfg.model.get("pretrained_path"),
        dtype=cfg.model.get("dtype", "bfloat16" if device == "cuda" else "float32"),
        meta_init=False, device=device,
    ).eval()
    tokenizer = None
    tok_path = cfg.get_by_dotted("model.tokenizer_path", None)
    if tok_path:
        from automodel_amd.models.auto_tokenizer import build_tokenizer

        tokenizer = build_tokenizer(tok_path)
    proposer = None
    if args.draft:
        from automodel_amd.speculative.decode import EagleProposer
        from automodel_amd.speculative.draft import load_draft

        proposer = EagleProposer(load_draft(args.draft, model, device), model)
    app = build_app(model, tokenizer, use_cache=not args.no_cache,
                    use_graph=args.graph, proposer=proposer)
    uvicorn.run(app, host=args.host, port=args.port)


if __name__ == "__main__":
    main()
