"""MegatronFSDP-style flat-shard engine: world-2 gloo parity against
single-process AdamW training, grad accumulation, jointly-sharded
optimizer state, and shard-checkpoint resume."""

import torch

from tests.dist_utils import run_distributed


def _tiny_llama(seed=7):
    from automodel_amd.models.llama.model import LlamaForCausalLM

    torch.manual_seed(seed)
    from automodel_amd.loss.linear_ce import FusedLinearCrossEntropy

    m = LlamaForCausalLM(dict(
        vocab_size=96, hidden_size=32, intermediate_size=48,
        num_hidden_layers=2, num_attention_heads=2, num_key_value_heads=2,
        max_position_embeddings=64, rope_theta=10000.0))
    m.init_weights()
    m.loss_fn = FusedLinearCrossEntropy(backend="chunked", chunk_size=32)
    return m


def _data(seed, B, S, vocab=96):
    g = torch.Generator().manual_seed(seed)
    return torch.randint(0, vocab, (B, S), generator=g)


def _reference_losses(steps=4, lr=1e-2):
    """Single-process full-batch AdamW trajectory (sum loss)."""
    model = _tiny_llama()
    opt = torch.optim.AdamW(model.parameters(), lr=lr)
    losses = []
    for s in range(steps):
        ids = _data(100 + s, 4, 12)
        loss = model(ids, labels=ids.clone())
        losses.append(float(loss))
        loss.backward()
        opt.step()
        opt.zero_grad()
    return losses


def _engine_worker(rank, world, steps, lr):
    from automodel_amd.parallel.megatron_fsdp import MegatronFSDPEngine

    model = _tiny_llama()                      # same seed on every rank
    engine = MegatronFSDPEngine(model, lr=lr)
    losses = []
    for s in range(steps):
        ids = _data(100 + s, 4, 12)
        shard = ids[rank * 2:(rank + 1) * 2]   # split the batch across DP
        loss = model(shard, labels=shard.clone())
        local = loss.detach().clone()
        torch.distributed.all_reduce(local)
        losses.append(float(local))
        loss.backward()
        engine.reduce_grads()
        engine.step()
    # optimizer state is sharded jointly: every state tensor is 1/world
    total = engine.consolidated_param_count()
    state_n = 0
    for st in engine.optimizer.state.values():
        state_n += st["exp_avg"].numel()
    return losses, state_n, total


def test_engine_matches_single_process_adamw():
    ref = _reference_losses(steps=4, lr=1e-2)
    out = run_distributed(_engine_worker, world=2, args=(4, 1e-2))
    for rank in (0, 1):
        losses, state_n, total = out[rank]
        assert torch.allclose(torch.tensor(losses), torch.tensor(ref),
                              atol=2e-3, rtol=2e-4), (losses, ref)
        # shards (and optimizer moments) cover ~1/2 of the params each
        assert abs(state_n - total / 2) <= total * 0.02


def _accum_worker(rank, world, lr):
    from automodel_amd.parallel.megatron_fsdp import MegatronFSDPEngine

    model = _tiny_llama()
    engine = MegatronFSDPEngine(model, lr=lr)
    big = _data(200, 8, 10)
    for micro in (big[rank * 4:rank * 4 + 2], big[rank * 4 + 2:rank * 4 + 4]):
        loss = model(micro, labels=micro.clone())
        loss.backward()
        engine.reduce_grads()                  # accumulates into main_grad
    engine.step()
    engine.gather_all()
    return model.lm_head.weight.detach().tolist()


def test_engine_grad_accumulation_matches_full_batch():
    # reference: one full-batch step
    model = _tiny_llama()
    opt = torch.optim.AdamW(model.parameters(), lr=1e-2)
    big = _data(200, 8, 10)
    loss = model(big, labels=big.clone())
    loss.backward()
    opt.step()
    ref = model.lm_head.weight.detach().clone()
    out = run_distributed(_accum_worker, world=2, args=(1e-2,))
    for rank in (0, 1):
        torch.testing.assert_close(torch.tensor(out[rank]), ref,
                                   atol=2e-4, rtol=2e-4)


def _resume_worker(rank, world):
    from automodel_amd.parallel.megatron_fsdp import MegatronFSDPEngine

    model = _tiny_llama()
    engine = MegatronFSDPEngine(model, lr=1e-2)
    ids = _data(300, 2, 8)
    shard = ids[rank:rank + 1]
    loss = model(shard, labels=shard.clone())
    loss.backward()
    engine.reduce_grads()
    engine.step()
    state = engine.shard_state_dict()
    # fresh model+engine, load shard state, one more identical step
    model2 = _tiny_llama(seed=9)
    engine2 = MegatronFSDPEngine(model2, lr=1e-2)
    engine2.load_shard_state_dict(state)
    for eng, mdl in ((engine, model), (engine2, model2)):
        loss = mdl(shard, labels=shard.clone())
        loss.backward()
        eng.reduce_grads()
        eng.step()
    return (engine.buckets[0].shard.data.tolist(),
            engine2.buckets[0].shard.data.tolist())


def test_engine_shard_checkpoint_resume():
    out = run_distributed(_resume_worker, world=2)
    for rank in (0, 1):
        a, b = out[rank]
        torch.testing.assert_close(torch.tensor(a), torch.tensor(b),
                                   atol=1e-6, rtol=1e-6)
