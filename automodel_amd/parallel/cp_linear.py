"""Context parallelism for linear-attention mixers (GatedDeltaNet / KDA).

Reference behavior: nemo_automodel's per-model linear-attention CP files
(components/models/qwen3_5_moe/cp_linear_attn.py, models/kimi_linear/cp.py,
context_parallel/mamba.py — SURVEY.md §2.4 "CP (Mamba/linear-attn)").

MI355X-native design: linear attention is a left-to-right state
recurrence, so CP is a CHUNK RELAY over xGMI point-to-point — the
sequence is split into ``world`` contiguous chunks, rank r runs the
chunked kernel over its chunk seeded with the recurrent state received
from rank r-1 and hands its final state to rank r+1 (one [B,H,Dk,Dv]
tensor per hop, tiny next to the activations). The backward pass relays
the state GRADIENT right-to-left through the same chain. Exactness:
q/k normalization and decays are per-token, so the relayed state makes
the distributed result identical to the single-rank full-sequence run
(parity-tested on gloo world 2, forward and backward).

The relay is implemented as one autograd.Function per rank: forward
recomputes nothing (the local graph is built under enable_grad and
re-derived in backward via torch.autograd.grad), comms are plain
send/recv on the CP group. Ranks necessarily run the scan in sequence;
in a layered model the pipeline fills — rank r+1's layer L overlaps
rank r's layer L+1 — so steady-state utilization approaches 1 like a
1F1B pipeline.
"""

from __future__ import annotations

import torch
import torch.distributed as dist

__all__ = ["cp_linear_scan", "split_cp_chunk"]


def split_cp_chunk(x: torch.Tensor, rank: int, world: int, dim: int = 1):
    """Contiguous chunk split along the sequence dim (equal-sized)."""
    assert x.shape[dim] % world == 0, "sequence must divide the cp world"
    return x.chunk(world, dim=dim)[rank].contiguous()


class _CPLinearScan(torch.autograd.Function):
    """State relay around a chunked linear-attention kernel.

    kernel(q, k, v, g, beta, initial_state=..., return_final_state=True)
    -> (out [B,S,H,Dv], state [B,H,Dk,Dv])
    """

    @staticmethod
    def forward(ctx, kernel, group, state_shape, q, k, v, g, beta):
        rank = dist.get_rank(group)
        world = dist.get_world_size(group)
        ranks = (dist.get_process_group_ranks(group)
                 if group is not None else list(range(world)))
        state_in = torch.zeros(*state_shape, dtype=torch.float32,
                               device=q.device)
        if rank > 0:
            dist.recv(state_in, src=ranks[rank - 1], group=group)
        with torch.enable_grad():
            qd = q.detach().requires_grad_()
            kd = k.detach().requires_grad_()
            vd = v.detach().requires_grad_()
            gd = g.detach().requires_grad_()
            bd = beta.detach().requires_grad_()
            sd = state_in.detach().requires_grad_()
            out, state_out = kernel(qd, kd, vd, gd, bd,
                                    initial_state=sd,
                                    return_final_state=True)
        if rank < world - 1:
            dist.send(state_out.detach().float().contiguous(),
                      dst=ranks[rank + 1], group=group)
        ctx.saved = (qd, kd, vd, gd, bd, sd, out, state_out)
        ctx.meta = (group, ranks, rank, world)
        return out.detach()

    @staticmethod
    def backward(ctx, d_out):
        qd, kd, vd, gd, bd, sd, out, state_out = ctx.saved
        group, ranks, rank, world = ctx.meta
        d_state_out = torch.zeros_like(state_out)
        if rank < world - 1:
            dist.recv(d_state_out, src=ranks[rank + 1], group=group)
        grads = torch.autograd.grad(
            (out, state_out), (qd, kd, vd, gd, bd, sd),
            (d_out, d_state_out), allow_unused=True)
        dq, dk, dv, dg, dbeta, d_state_in = grads
        if rank > 0:
            dist.send((d_state_in if d_state_in is not None
                       else torch.zeros_like(sd)).contiguous(),
                      dst=ranks[rank - 1], group=group)
        return None, None, None, dq, dk, dv, dg, dbeta


def cp_linear_scan(kernel, q, k, v, g, beta, group=None):
    """Run a chunked linear-attention kernel context-parallel.

    Inputs are THIS RANK's contiguous sequence chunk ([B, S_local, H, D]).
    ``kernel`` is ``gated_delta_rule_chunked`` or ``kda_chunked`` (any
    callable with the initial/final-state contract). Returns this rank's
    output chunk; gradients flow across ranks through the state relay.
    """
    B, _, H, Dk = k.shape
    Dv = v.shape[-1]
    state_shape = (B, H, Dk, Dv)
    return _CPLinearScan.apply(kernel, group, state_shape, q, k, v, g, beta)
