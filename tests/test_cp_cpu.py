"""Context-parallel tests (gloo world 2, sdpa backend): zigzag sharding,
CP attention parity vs dense, full-model CP loss/grad parity."""

import pytest
import torch

from automodel_amd.parallel.cp import shard_batch_cp, zigzag_chunk_ids
from tests.dist_utils import run_distributed


def test_zigzag_shard_batch():
    batch = {"input_ids": torch.arange(16).unsqueeze(0), "labels": torch.arange(16).unsqueeze(0)}
    out0 = shard_batch_cp(batch, rank=0, world=2)
    out1 = shard_batch_cp(batch, rank=1, world=2)
    # rank0: chunks 0 and 3; rank1: chunks 1 and 2 (4 chunks of 4)
    assert out0["input_ids"].tolist() == [[0, 1, 2, 3, 12, 13, 14, 15]]
    assert out1["input_ids"].tolist() == [[4, 5, 6, 7, 8, 9, 10, 11]]
    assert out0["position_ids"].tolist() == [[0, 1, 2, 3, 12, 13, 14, 15]]
    # union covers everything exactly once
    both = torch.cat([out0["input_ids"], out1["input_ids"]], dim=1).sort().values
    assert both.tolist() == [list(range(16))]


def _cp_attn_fn(rank, world):
    import torch.distributed as dist

    from automodel_amd.ops.attention import attention_ref
    from automodel_amd.parallel.cp import cp_flash_attention, enable_cp, disable_cp

    torch.manual_seed(0)
    B, S, Hq, Hk, D = 2, 32, 4, 2, 16
    q = torch.randn(B, S, Hq, D)
    k = torch.randn(B, S, Hk, D)
    v = torch.randn(B, S, Hk, D)
    o_ref = attention_ref(q, k, v, causal=True)

    enable_cp(dist.group.WORLD)
    try:
        C = S // (2 * world)
        g0, g1 = zigzag_chunk_ids(rank, world)
        sel = torch.cat([torch.arange(g0 * C, (g0 + 1) * C),
                         torch.arange(g1 * C, (g1 + 1) * C)])
        ql = q[:, sel].contiguous().requires_grad_(True)
        kl = k[:, sel].contiguous().requires_grad_(True)
        vl = v[:, sel].contiguous().requires_grad_(True)
        o_local = cp_flash_attention(ql, kl, vl, causal=True, backend="sdpa")
        assert torch.allclose(o_local, o_ref[:, sel], atol=1e-4), \
            (o_local - o_ref[:, sel]).abs().max()

        # backward parity: grads vs dense autograd on local shards
        q2 = q.clone().requires_grad_(True)
        k2 = k.clone().requires_grad_(True)
        v2 = v.clone().requires_grad_(True)
        o2 = attention_ref(q2, k2, v2, causal=True)
        do = torch.ones_like(o2)
        o2.backward(do)
        o_local.backward(do[:, sel])
        assert torch.allclose(ql.grad, q2.grad[:, sel], atol=1e-4)
        assert torch.allclose(kl.grad, k2.grad[:, sel], atol=1e-4), \
            (kl.grad - k2.grad[:, sel]).abs().max()
        assert torch.allclose(vl.grad, v2.grad[:, sel], atol=1e-4)
    finally:
        disable_cp()
    return True


def test_cp2_attention_parity():
    run_distributed(_cp_attn_fn, world=2)


def _cp_model_fn(rank, world):
    import torch.distributed as dist

    from automodel_amd.loss.linear_ce import FusedLinearCrossEntropy
    from automodel_amd.models.common.backend import BackendConfig
    from automodel_amd.models.llama.model import LlamaConfig, LlamaForCausalLM
    from automodel_amd.parallel.cp import disable_cp, enable_cp

    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128,
                      num_hidden_layers=2, num_attention_heads=4,
                      num_key_value_heads=2, max_position_embeddings=64)
    torch.manual_seed(3)
    model = LlamaForCausalLM(cfg, backend=BackendConfig().for_cpu())
    model.init_weights()
    model.loss_fn = FusedLinearCrossEntropy(backend="chunked", chunk_size=64)

    g = torch.Generator().manual_seed(11)
    ids = torch.randint(0, 128, (2, 33), generator=g)
    batch = {"input_ids": ids[:, :-1].contiguous(), "labels": ids[:, 1:].contiguous()}

    # dense reference loss (full batch)
    ref_loss = model(batch["input_ids"], labels=batch["labels"])
    ref_loss.backward()
    ref_grad = model.model.layers[0].self_attn.q_proj.weight.grad.clone()
    model.zero_grad()

    from automodel_amd.parallel.cp import shard_batch_cp
    enable_cp(dist.group.WORLD)
    try:
        local = shard_batch_cp(batch, rank, world)
        loss_local = model(local["input_ids"], labels=local["labels"],
                           position_ids=local["position_ids"])
        total = loss_local.detach().clone()
        dist.all_reduce(total)
        assert torch.allclose(total, ref_loss.detach(), rtol=1e-4), \
            (float(total), float(ref_loss))
        loss_local.backward()
        # sum of CP-rank grads == dense grad
        g_sum = model.model.layers[0].self_attn.q_proj.weight.grad.clone()
        dist.all_reduce(g_sum)
        assert torch.allclose(g_sum, ref_grad, atol=1e-4), \
            (g_sum - ref_grad).abs().max()
    finally:
        disable_cp()
    return float(total)


def test_cp2_model_loss_and_grad_parity():
    out = run_distributed(_cp_model_fn, world=2)
    assert abs(out[0] - out[1]) < 1e-4


# -------------------------------------------------- vocab-parallel CE (TP)
def _vp_ce_fn(rank, world):
    import torch.distributed as dist

    from automodel_amd.loss.vocab_parallel_ce import vocab_parallel_cross_entropy

    torch.manual_seed(0)  # same full logits everywhere
    T, V = 32, 64
    logits = torch.randn(T, V)
    labels = torch.randint(0, V, (T,))
    labels[3] = -100
    ref = torch.nn.functional.cross_entropy(logits.float(), labels,
                                            ignore_index=-100, reduction="sum")

    Vl = V // world
    shard = logits[:, rank * Vl : (rank + 1) * Vl].clone().requires_grad_(True)
    loss = vocab_parallel_cross_entropy(shard, labels, vocab_offset=rank * Vl,
                                        group=dist.group.WORLD)
    assert torch.allclose(loss, ref, rtol=1e-4), (float(loss), float(ref))
    loss.backward()

    l2 = logits.clone().requires_grad_(True)
    ref2 = torch.nn.functional.cross_entropy(l2.float(), labels,
                                             ignore_index=-100, reduction="sum")
    ref2.backward()
    ref_shard_grad = l2.grad[:, rank * Vl : (rank + 1) * Vl]
    assert torch.allclose(shard.grad.float(), ref_shard_grad, atol=1e-4), \
        (shard.grad.float() - ref_shard_grad).abs().max()
    return float(loss)


def test_vocab_parallel_ce_tp2():
    out = run_distributed(_vp_ce_fn, world=2)
    assert abs(out[0] - out[1]) < 1e-4


def _blockdiag_cp_worker(rank, world):
    """Packed docs under CP: sharded blockdiag attention == single-process
    varlen reference over the full packed sequence."""
    import torch

    from automodel_amd.ops.attention import flash_attention_varlen
    from automodel_amd.parallel.cp import (
        CPContext,
        cp_blockdiag_attention,
        disable_cp,
        enable_cp,
        local_global_positions,
    )

    torch.manual_seed(0)
    B, T, H, D = 1, 32, 2, 16
    cu = torch.tensor([0, 10, 17, 32], dtype=torch.int32)
    q = torch.randn(B, T, H, D, dtype=torch.float64)
    k = torch.randn(B, T, H, D, dtype=torch.float64)
    v = torch.randn(B, T, H, D, dtype=torch.float64)
    ref = flash_attention_varlen(q.float(), k.float(), v.float(), cu,
                                 backend="sdpa")

    import torch.distributed as dist

    cp = enable_cp(dist.group.WORLD)
    C = T // (2 * world)
    gpos = local_global_positions(rank, world, C)
    q_l = q[:, gpos].float().requires_grad_(True)
    k_l = k[:, gpos].float().requires_grad_(True)
    v_l = v[:, gpos].float()
    out = cp_blockdiag_attention(q_l, k_l, v_l, cu)
    torch.testing.assert_close(out, ref[:, gpos], atol=1e-5, rtol=1e-5)
    # backward runs (gather bwd reduce-scatters dK to owners)
    out.sum().backward()
    assert q_l.grad is not None and k_l.grad is not None
    disable_cp()


def test_cp_blockdiag_matches_varlen_reference():
    run_distributed(_blockdiag_cp_worker, world=2)


def _blockdiag_model_worker(rank, world):
    """Full model forward: CP + packed varlen context routes attention
    through cp_blockdiag_attention (wiring test)."""
    import torch

    from automodel_amd.models.llama.model import LlamaForCausalLM
    from automodel_amd.ops.attention import set_varlen_context
    from automodel_amd.parallel.cp import (
        disable_cp,
        enable_cp,
        local_global_positions,
    )

    torch.manual_seed(0)
    m = LlamaForCausalLM(dict(vocab_size=100, hidden_size=32, intermediate_size=64,
                              num_hidden_layers=2, num_attention_heads=2,
                              num_key_value_heads=1, max_position_embeddings=64))
    m.init_weights(device="cpu")
    m.eval()
    T = 32
    ids = torch.randint(0, 100, (1, T))
    cu = torch.tensor([0, 12, 32], dtype=torch.int32)
    with torch.no_grad():
        set_varlen_context(cu)
        ref = m(ids, return_hidden=True)
        set_varlen_context(None)

        import torch.distributed as dist

        enable_cp(dist.group.WORLD)
        C = T // (2 * world)
        gpos = local_global_positions(rank, world, C)
        set_varlen_context(cu)
        local = m(ids[:, gpos], position_ids=gpos.unsqueeze(0),
                  return_hidden=True)
        set_varlen_context(None)
        disable_cp()
    torch.testing.assert_close(local, ref[:, gpos], atol=1e-5, rtol=1e-5)


def test_cp_blockdiag_full_model():
    run_distributed(_blockdiag_model_worker, world=2)


def _cp_ring_fn(rank, world):
    import torch.distributed as dist

    from automodel_amd.ops.attention import attention_ref
    from automodel_amd.parallel.cp import cp_ring_attention, disable_cp, enable_cp

    torch.manual_seed(3)
    B, S, Hq, Hk, D = 2, 32, 4, 2, 16
    q = torch.randn(B, S, Hq, D)
    k = torch.randn(B, S, Hk, D)
    v = torch.randn(B, S, Hk, D)
    o_ref = attention_ref(q, k, v, causal=True)
    enable_cp(dist.group.WORLD)
    try:
        C = S // (2 * world)
        g0, g1 = zigzag_chunk_ids(rank, world)
        sel = torch.cat([torch.arange(g0 * C, (g0 + 1) * C),
                         torch.arange(g1 * C, (g1 + 1) * C)])
        ql = q[:, sel].contiguous().requires_grad_(True)
        kl = k[:, sel].contiguous().requires_grad_(True)
        vl = v[:, sel].contiguous().requires_grad_(True)
        o_local = cp_ring_attention(ql, kl, vl, causal=True)
        assert torch.allclose(o_local, o_ref[:, sel], atol=1e-4), \
            (o_local - o_ref[:, sel]).abs().max()
        q2 = q.clone().requires_grad_(True)
        k2 = k.clone().requires_grad_(True)
        v2 = v.clone().requires_grad_(True)
        o2 = attention_ref(q2, k2, v2, causal=True)
        do = torch.randn_like(o2)
        o2.backward(do)
        o_local.backward(do[:, sel])
        assert torch.allclose(ql.grad, q2.grad[:, sel], atol=1e-4), \
            (ql.grad - q2.grad[:, sel]).abs().max()
        assert torch.allclose(kl.grad, k2.grad[:, sel], atol=1e-4), \
            (kl.grad - k2.grad[:, sel]).abs().max()
        assert torch.allclose(vl.grad, v2.grad[:, sel], atol=1e-4)
        # non-causal path
        o_nc = cp_ring_attention(ql.detach(), kl.detach(), vl.detach(), causal=False)
        o_nc_ref = attention_ref(q, k, v, causal=False)
        assert torch.allclose(o_nc, o_nc_ref[:, sel], atol=1e-4)
    finally:
        disable_cp()
    return True


def test_cp2_ring_attention_parity():
    """Ring-P2P KV rotation (zigzag, batch_isend_irecv) matches dense
    attention forward AND backward — incl. GQA head folding."""
    run_distributed(_cp_ring_fn, world=2)


def _cp_mech_fn(rank, world):
    import torch.distributed as dist

    from automodel_amd.ops.attention import attention_ref
    from automodel_amd.parallel.cp import cp_flash_attention, disable_cp, enable_cp

    torch.manual_seed(4)
    B, S, H, D = 1, 16, 2, 8
    q = torch.randn(B, S, H, D)
    k = torch.randn(B, S, H, D)
    v = torch.randn(B, S, H, D)
    o_ref = attention_ref(q, k, v, causal=True)
    enable_cp(dist.group.WORLD, mechanism="ring")
    try:
        C = S // (2 * world)
        g0, g1 = zigzag_chunk_ids(rank, world)
        sel = torch.cat([torch.arange(g0 * C, (g0 + 1) * C),
                         torch.arange(g1 * C, (g1 + 1) * C)])
        o = cp_flash_attention(q[:, sel].contiguous(), k[:, sel].contiguous(),
                               v[:, sel].contiguous(), causal=True)
        assert torch.allclose(o, o_ref[:, sel], atol=1e-4)
    finally:
        disable_cp()
    return True


def test_cp2_mechanism_ring_dispatch():
    """enable_cp(mechanism='ring') routes cp_flash_attention to the ring."""
    run_distributed(_cp_mech_fn, world=2)
