"""Generic PP: virtual stages (interleaved_1f1b), grad accumulation, and
non-llama families (reference pipelining/functional.py:182,597)."""

import torch

from tests.dist_utils import run_distributed

TINY = dict(
    vocab_size=256, hidden_size=64, intermediate_size=128, num_hidden_layers=4,
    num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=128,
)


def _make_model(seed, cls_cfg=None, arch="LlamaForCausalLM"):
    from automodel_amd.models.registry import build_model

    torch.manual_seed(seed)
    return build_model(config=cls_cfg or dict(TINY), architecture=arch,
                       dtype="float32", meta_init=False)


def _make_batch(seed, B=4, S=16, V=256):
    g = torch.Generator().manual_seed(seed)
    ids = torch.randint(0, V, (B, S + 1), generator=g)
    return ids[:, :-1], ids[:, 1:].contiguous()


# ---------------------------------------------------- PP2 x VP2 interleaved
def _pp_vp_fn(rank, world):
    from automodel_amd.loss.linear_ce import FusedLinearCrossEntropy
    from automodel_amd.parallel.mesh import build_mesh
    from automodel_amd.parallel.pp import AutoPipeline, PipelineConfig

    model = _make_model(seed=11)
    inp, lab = _make_batch(seed=12, B=4, S=16)
    loss_fn = FusedLinearCrossEntropy(backend="chunked", chunk_size=64)
    hidden = model(inp, return_hidden=True)
    ref = float(loss_fn(hidden, model.lm_head.weight, lab))

    ctx = build_mesh(dp_shard=1, pp=2, device_type="cpu")
    pipe = AutoPipeline(model, ctx["pp"],
                        PipelineConfig(pp_size=2, schedule="interleaved_1f1b",
                                       microbatches=4, virtual_stages=2),
                        loss_fn=loss_fn, device="cpu")
    # 2 virtual stages per rank, 4 global stages, 1 layer each
    assert len(pipe.stages) == 2
    assert {m.stage_idx for m in pipe.stage_modules} == \
        ({0, 2} if rank == 0 else {1, 3})
    assert all(len(m.layers) == 1 for m in pipe.stage_modules)
    losses = pipe.step(input_ids=inp, target=lab)
    if pipe.is_last:
        total = float(sum(losses))
        assert abs(total - ref) / max(1.0, abs(ref)) < 2e-3, (total, ref)
        return total
    assert any(p.grad is not None for p in pipe.stage_modules.parameters())
    return None


def test_pp2_vp2_interleaved_parity():
    out = run_distributed(_pp_vp_fn, world=2)
    assert out[1] is not None


# ------------------------------------------------------- PP grad-accum parity
def _pp_gacc_fn(rank, world):
    from automodel_amd.loss.linear_ce import FusedLinearCrossEntropy
    from automodel_amd.parallel.mesh import build_mesh
    from automodel_amd.parallel.pp import AutoPipeline, PipelineConfig

    model = _make_model(seed=21)
    ref_model = _make_model(seed=21)
    b1 = _make_batch(seed=22, B=2, S=16)
    b2 = _make_batch(seed=23, B=2, S=16)
    loss_fn = FusedLinearCrossEntropy(backend="chunked", chunk_size=64)

    # reference: two accumulated backwards on the single-rank model
    for inp, lab in (b1, b2):
        hidden = ref_model(inp, return_hidden=True)
        loss_fn(hidden, ref_model.lm_head.weight, lab).backward()

    ctx = build_mesh(dp_shard=1, pp=2, device_type="cpu")
    pipe = AutoPipeline(model, ctx["pp"],
                        PipelineConfig(pp_size=2, schedule="gpipe", microbatches=2),
                        loss_fn=loss_fn, device="cpu")
    for inp, lab in (b1, b2):
        pipe.step(input_ids=inp, target=lab)

    # per-parameter grad parity for this rank's stage slice
    ref_named = dict(ref_model.named_parameters())
    checked = 0
    for mod in ([pipe.stage_module] if not isinstance(pipe.stage_module, torch.nn.ModuleList)
                else pipe.stage_modules):
        lo, hi = mod.layer_range
        for i, layer in enumerate(mod.layers):
            for n, p in layer.named_parameters():
                rp = ref_named[f"model.layers.{lo + i}.{n}"]
                assert p.grad is not None and rp.grad is not None, n
                torch.testing.assert_close(p.grad, rp.grad, atol=2e-4, rtol=2e-4)
                checked += 1
    assert checked > 0
    return checked


def test_pp2_grad_accum_parity():
    out = run_distributed(_pp_gacc_fn, world=2)
    assert all(v > 0 for v in out.values())


# ---------------------------------------------------------- PP on MoE family
def _pp_moe_fn(rank, world):
    from automodel_amd.loss.linear_ce import FusedLinearCrossEntropy
    from automodel_amd.parallel.mesh import build_mesh
    from automodel_amd.parallel.pp import AutoPipeline, PipelineConfig

    cfg = dict(vocab_size=128, hidden_size=32, intermediate_size=64,
               num_hidden_layers=2, num_attention_heads=2, num_key_value_heads=1,
               max_position_embeddings=64,
               moe=dict(n_routed_experts=4, n_activated_experts=2,
                        moe_intermediate_size=48))
    model = _make_model(seed=31, cls_cfg=cfg, arch="Qwen3MoeForCausalLM")
    inp, lab = _make_batch(seed=32, B=2, S=16, V=128)
    loss_fn = FusedLinearCrossEntropy(backend="chunked", chunk_size=64)
    hidden = model(inp, return_hidden=True)
    ref = float(loss_fn(hidden, model.lm_head.weight, lab))

    ctx = build_mesh(dp_shard=1, pp=2, device_type="cpu")
    pipe = AutoPipeline(model, ctx["pp"],
                        PipelineConfig(pp_size=2, schedule="gpipe", microbatches=2),
                        loss_fn=loss_fn, device="cpu")
    losses = pipe.step(input_ids=inp, target=lab)
    if pipe.is_last:
        total = float(sum(losses))
        assert abs(total - ref) / max(1.0, abs(ref)) < 2e-3, (total, ref)
        return total
    return None


def test_pp2_moe_family():
    out = run_distributed(_pp_moe_fn, world=2)
    assert out[1] is not None
