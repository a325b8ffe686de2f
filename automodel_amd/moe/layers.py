"""MoE building blocks: Gate (+aux loss, aux-free bias), FakeBalancedGate, MoE.

Reference behavior: nemo_automodel/components/moe/layers.py — Gate :225
(softmax/sigmoid scoring, expert groups, top-k, aux loss :670, bias update
:586), FakeBalancedGate :126 (ideal routing for benchmarks), MoE :733
(gate + routed experts + shared experts).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from automodel_amd.moe.config import MoEConfig
from automodel_amd.moe.experts import GroupedExperts
from automodel_amd.ops.swiglu import swiglu


class Gate(nn.Module):
    def __init__(self, hidden_size: int, cfg: MoEConfig):
        super().__init__()
        self.cfg = cfg
        self.weight = nn.Parameter(torch.empty(cfg.n_routed_experts, hidden_size))
        if cfg.expert_bias:
            # routing-only bias, updated out-of-band (aux-free balancing)
            self.register_buffer("e_score_correction_bias", torch.zeros(cfg.n_routed_experts))
        self.last_aux_loss: torch.Tensor | None = None

    def forward(self, x: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
        """x [T, H] -> (probs [T, K], indices [T, K])."""
        cfg = self.cfg
        logits = torch.nn.functional.linear(x.float(), self.weight.float())
        if cfg.score_func == "sigmoid":
            scores = logits.sigmoid()
        else:
            scores = logits.softmax(dim=-1)

        select_scores = scores
        if cfg.expert_bias:
            select_scores = scores + self.e_score_correction_bias

        if cfg.n_expert_groups > 1:
            T, E = scores.shape
            gs = select_scores.view(T, cfg.n_expert_groups, -1)
            group_scores = gs.topk(2, dim=-1)[0].sum(-1)
            group_idx = group_scores.topk(cfg.n_limited_groups, dim=-1)[1]
            mask = torch.zeros_like(group_scores, dtype=torch.bool)
            mask.scatter_(1, group_idx, True)
            select_scores = select_scores.masked_fill(
                ~mask.unsqueeze(-1).expand_as(gs).reshape(T, E), float("-inf")
            )

        from automodel_amd.moe.router_replay import active_router_replay

        rr = active_router_replay()
        if rr is not None and rr.mode == "replay":
            indices = rr.next_replay().to(x.device)   # R3: replayed routing
        else:
            base = logits if cfg.topk_then_softmax else select_scores
            _, indices = base.topk(cfg.n_activated_experts, dim=-1)
            if rr is not None and rr.mode == "record":
                rr.record(indices)
        if cfg.topk_then_softmax:   # granite-moe: softmax over the k logits
            probs = logits.gather(1, indices).softmax(dim=-1)
        else:
            probs = scores.gather(1, indices)
        if cfg.norm_topk_prob and cfg.n_activated_experts > 1:
            probs = probs / probs.sum(dim=-1, keepdim=True).clamp_min(1e-20)
        probs = probs * cfg.route_scale

        if self.training and cfg.aux_loss_coeff > 0:
            self.last_aux_loss = self._aux_loss(scores, indices)
        return probs.to(x.dtype), indices

    def _aux_loss(self, scores: torch.Tensor, indices: torch.Tensor) -> torch.Tensor:
        """Switch-style load-balancing loss (reference layers.py:670)."""
        cfg = self.cfg
        T, E = scores.shape
        counts = torch.zeros(E, device=scores.device)
        counts.scatter_add_(0, indices.reshape(-1),
                            torch.ones(indices.numel(), device=scores.device))
        f = counts / (T * cfg.n_activated_experts)      # fraction routed
        p = scores.mean(dim=0)                          # mean router prob
        return cfg.aux_loss_coeff * E * (f * p).sum()

    @torch.no_grad()
    def update_bias(self, expert_load: torch.Tensor) -> None:
        """Aux-free balancing: push bias against overload (reference :586)."""
        if not self.cfg.expert_bias:
            return
        mean_load = expert_load.float().mean()
        err = expert_load.float() - mean_load
        self.e_score_correction_bias -= self.cfg.bias_update_speed * err.sign()


class FakeBalancedGate(nn.Module):
    """Deterministic perfectly-balanced routing for benchmarks
    (reference layers.py:126 — used for the published numbers)."""

    def __init__(self, hidden_size: int, cfg: MoEConfig):
        super().__init__()
        self.cfg = cfg
        self.last_aux_loss = None

    def forward(self, x: torch.Tensor):
        T = x.shape[0]
        K, E = self.cfg.n_activated_experts, self.cfg.n_routed_experts
        base = torch.arange(T, device=x.device, dtype=torch.long) * K
        indices = ((base[:, None] + torch.arange(K, device=x.device)) % E)
        probs = torch.full((T, K), 1.0 / K, device=x.device, dtype=x.dtype)
        return probs, indices

    @torch.no_grad()
    def update_bias(self, expert_load):
        pass


class SharedExpert(nn.Module):
    def __init__(self, hidden_size: int, intermediate: int):
        super().__init__()
        self.gate_proj = nn.Linear(hidden_size, intermediate, bias=False)
        self.up_proj = nn.Linear(hidden_size, intermediate, bias=False)
        self.down_proj = nn.Linear(intermediate, hidden_size, bias=False)

    def forward(self, x):
        return self.down_proj(swiglu(self.gate_proj(x), self.up_proj(x)))


_SHARED_STREAMS: dict = {}


def _shared_stream(device) -> "torch.cuda.Stream":
    """Per-device side stream for shared-expert overlap."""
    key = device.index if hasattr(device, "index") else int(device)
    if key not in _SHARED_STREAMS:
        _SHARED_STREAMS[key] = torch.cuda.Stream(device=device)
    return _SHARED_STREAMS[key]


class MoE(nn.Module):
    """Gate -> dispatch -> grouped experts -> combine (+ shared experts)."""

    def __init__(self, hidden_size: int, cfg: MoEConfig, dispatcher=None):
        super().__init__()
        self.cfg = cfg
        inter = cfg.moe_intermediate_size or hidden_size * 4
        self.gate = (FakeBalancedGate if cfg.fake_balanced_gate else Gate)(hidden_size, cfg)
        self.experts = GroupedExperts(cfg.n_routed_experts, hidden_size, inter)
        self.dispatcher = dispatcher
        self.shared_experts = None
        if cfg.n_shared_experts > 0:
            shared_inter = cfg.shared_expert_intermediate_size or inter * cfg.n_shared_experts
            self.shared_experts = SharedExpert(hidden_size, shared_inter)
            if cfg.shared_expert_gate:   # qwen2-moe sigmoid token gate
                self.shared_expert_gate = nn.Linear(hidden_size, 1, bias=False)
        # per-forward expert load (for metrics + bias update)
        self.last_expert_load: torch.Tensor | None = None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, S, H = x.shape
        xf = x.reshape(-1, H)
        probs, indices = self.gate(xf)
        with torch.no_grad():
            load = torch.zeros(self.cfg.n_routed_experts, device=x.device)
            load.scatter_add_(0, indices.reshape(-1),
                              torch.ones(indices.numel(), device=x.device))
            self.last_expert_load = load
        if self.dispatcher is not None:
            y = self.dispatcher(xf, probs, indices, self.experts)
        elif self.shared_experts is not None and xf.is_cuda:
            # shared-expert overlap (reference token_dispatcher.py shared-
            # expert stream): the dense shared MLP runs on a side stream
            # concurrently with the routed dispatch+grouped-GEMM path —
            # stream-aware autograd keeps backward ordering correct
            ss = _shared_stream(xf.device)
            ss.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(ss):
                shared = self.shared_experts(xf)
                if getattr(self, "shared_expert_gate", None) is not None:
                    shared = torch.sigmoid(self.shared_expert_gate(xf)) * shared
            y = self.experts(xf, probs, indices)
            torch.cuda.current_stream().wait_stream(ss)
            shared.record_stream(torch.cuda.current_stream())
            return (y + shared).view(B, S, H)
        else:
            y = self.experts(xf, probs, indices)
        if self.shared_experts is not None:
            shared = self.shared_experts(xf)
            if getattr(self, "shared_expert_gate", None) is not None:
                shared = torch.sigmoid(self.shared_expert_gate(xf)) * shared
            y = y + shared
        # gate.last_aux_loss stays attached to the graph; the model adds
        # sum-of-aux to the main loss before backward (reference
        # MoEAuxLossAutoScaler moe_utils.py:568 achieves the same coupling).
        return y.view(B, S, H)
