"""Coverage for the smaller runtime modules: DDP manager, prewarm helpers,
autonvtx patching, async checkpoint writer."""

import os
import time

import pytest
import torch
import torch.distributed as dist
import torch.nn as nn

from tests.dist_utils import run_distributed


def _ddp_worker(rank, world):
    from automodel_amd.parallel.ddp import DDPManager

    torch.manual_seed(0)
    model = nn.Linear(8, 8)
    ddp = DDPManager(bucket_cap_mb=1).parallelize(model)
    torch.manual_seed(100 + rank)  # different data per rank
    x = torch.randn(4, 8)
    ddp(x).sum().backward()
    # grads must be identical (averaged) across ranks after backward
    g = model.weight.grad.clone()
    gathered = [torch.empty_like(g) for _ in range(world)]
    dist.all_gather(gathered, g)
    for other in gathered:
        torch.testing.assert_close(g, other)


def test_ddp_manager_grad_sync():
    run_distributed(_ddp_worker, world=2)


def _prewarm_worker(rank, world):
    from automodel_amd.parallel.mesh import build_mesh
    from automodel_amd.training.prewarm import prewarm_collectives, prewarm_gemms

    ctx = build_mesh(device_type="cpu", dp_shard=world)
    prewarm_gemms()          # no-op on CPU
    prewarm_collectives(ctx)  # must run one collective per initialized group


def test_prewarm_collectives_cpu():
    run_distributed(_prewarm_worker, world=2)


def test_autonvtx_cpu_noop_and_unpatch():
    from automodel_amd.utils import autonvtx

    m = nn.Sequential(nn.Linear(4, 4), nn.ReLU())
    out = autonvtx.patch(m)  # CPU: returns module unpatched
    assert out is m
    y = m(torch.randn(2, 4))
    assert y.shape == (2, 4)
    autonvtx.unpatch(m)  # safe even when never patched


def test_async_checkpoint_writer(tmp_path):
    from automodel_amd.checkpoint.async_save import AsyncCheckpointWriter

    w = AsyncCheckpointWriter()
    state = {"w": torch.randn(16), "nested": {"b": torch.ones(2)}, "step": 3}
    written = {}

    def write_fn(staged):
        time.sleep(0.05)
        torch.save(staged, tmp_path / "ckpt.pt")
        written["done"] = True

    w.save_async(state, write_fn)
    assert w.in_flight or written.get("done")
    # mutate source after staging: the write must see the staged copy
    state["w"].fill_(0.0)
    w.wait()
    assert written["done"] and not w.in_flight
    loaded = torch.load(tmp_path / "ckpt.pt")
    assert loaded["step"] == 3
    assert loaded["w"].abs().sum() > 0  # staged before the fill_(0)
    assert torch.equal(loaded["nested"]["b"], torch.ones(2))

    # second save waits for the first implicitly
    w.save_async({"x": torch.zeros(1)}, write_fn)
    w.wait()


def test_generation_server():
    fastapi = pytest.importorskip("fastapi")
    from fastapi.testclient import TestClient

    from automodel_amd.models.llama.model import LlamaForCausalLM
    from automodel_amd.serving.server import build_app
    from automodel_amd.utils.generation import generate

    torch.manual_seed(0)
    model = LlamaForCausalLM(dict(vocab_size=150, hidden_size=32,
                                  intermediate_size=64, num_hidden_layers=2,
                                  num_attention_heads=2, num_key_value_heads=1,
                                  max_position_embeddings=128))
    model.init_weights(device="cpu")
    model.eval()
    app = build_app(model, tokenizer=None, use_cache=True)
    client = TestClient(app)

    r = client.get("/health")
    assert r.status_code == 200 and r.json()["status"] == "ok"

    prompt = [5, 9, 23, 44]
    r = client.post("/generate", json={"prompt_ids": prompt, "max_new_tokens": 8})
    assert r.status_code == 200
    out = r.json()["output_ids"]
    ref = generate(model, torch.tensor([prompt]), max_new_tokens=8)
    assert out == ref[0, 4:].tolist()   # server(KV-cached) == plain greedy

    # speculative endpoint: greedy-equivalence guarantee holds over the
    # ngram fallback proposer, and acceptance stats are reported
    r = client.post("/generate", json={"prompt_ids": prompt,
                                       "max_new_tokens": 8,
                                       "speculative": True})
    assert r.status_code == 200
    body = r.json()
    assert body["output_ids"] == ref[0, 4:].tolist()
    assert 0.0 <= body["spec_acceptance_rate"] <= 1.0
    assert body["spec_tokens_per_call"] >= 1.0


def _sigterm_worker(rank, world):
    """SIGTERM on ONE rank propagates to all via the MAX all-reduce."""
    import os
    import signal

    from automodel_amd.training.signal_handler import DistributedSignalHandler

    with DistributedSignalHandler(signal.SIGUSR1) as h:
        assert not h.signals_received()
        if rank == 0:
            os.kill(os.getpid(), signal.SIGUSR1)
        # all ranks must now see it
        assert h.signals_received()


def test_signal_handler_propagates():
    run_distributed(_sigterm_worker, world=2)


def test_compile_config_regional():
    """Regional torch.compile wraps decoder layers (ROCm-safe aot_eager)."""
    import torch

    from automodel_amd.models.llama.model import LlamaConfig, LlamaForCausalLM
    from automodel_amd.utils.compile import CompileConfig, apply_compile

    m = LlamaForCausalLM(LlamaConfig(
        vocab_size=128, hidden_size=32, intermediate_size=64,
        num_hidden_layers=2, num_attention_heads=2, num_key_value_heads=1,
        max_position_embeddings=64))
    m.init_weights()
    n = apply_compile(m, CompileConfig(enabled=True, backend="eager"))
    assert n == 2
    ids = torch.randint(0, 128, (1, 8))
    assert m(ids).shape == (1, 8, 128)


def _mesh_timeout_fn(rank, world):
    from automodel_amd.parallel.mesh import build_mesh

    ctx = build_mesh(dp_shard=2, device_type="cpu",
                     axis_timeouts={"dp_shard": 7, "tp": 1})
    return ctx.dims["dp_shard"]


def test_mesh_axis_timeouts_cpu():
    """axis_timeouts applies per-axis pg timeout overrides (world 2 gloo)."""
    from tests.dist_utils import run_distributed

    out = run_distributed(_mesh_timeout_fn, world=2)
    assert out[0] == 2
