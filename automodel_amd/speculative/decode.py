"""Speculative decoding: draft-propose, target-verify, accept-longest-prefix.

Reference behavior: nemo_automodel/components/speculative/decode_eval.py and
eagle/msd_decode.py (draft proposes gamma tokens, the target scores the whole
proposal in ONE forward, the longest prefix agreeing with the target's greedy
choice is accepted plus one corrected token — output is IDENTICAL to plain
target greedy decoding, only faster). On MI355X the win is in arithmetic
intensity: one [B, gamma+1] verify forward keeps the MFMA pipes busy where
token-by-token decode is bandwidth-bound.

Proposers:
  * EagleProposer — wraps speculative.draft.EagleDraftModel, carrying the
    target's fused aux hidden over verified tokens and the draft's own
    hidden over speculative ones (EAGLE hidden recycling);
  * NgramProposer — training-free lookup of the longest recent n-gram match
    (reference's prompt-lookup style baseline).
"""

from __future__ import annotations

from dataclasses import dataclass, field

import torch


@dataclass
class SpecStats:
    proposed: int = 0
    accepted: int = 0
    target_calls: int = 0
    tokens_out: int = 0
    per_round_accepts: list = field(default_factory=list)

    @property
    def acceptance_rate(self) -> float:
        return self.accepted / self.proposed if self.proposed else 0.0

    @property
    def tokens_per_target_call(self) -> float:
        return self.tokens_out / self.target_calls if self.target_calls else 0.0


class NgramProposer:
    """Propose the continuation that followed the most recent occurrence of
    the current (n)-gram suffix; pad with the last token when no match."""

    def __init__(self, n: int = 3):
        self.n = n

    @torch.no_grad()
    def propose(self, ids: torch.Tensor, gamma: int) -> torch.Tensor:
        B, T = ids.shape
        out = ids.new_full((B, gamma), 0)
        for b in range(B):
            seq = ids[b].tolist()
            n = min(self.n, T - 1)
            tail = seq[-n:] if n > 0 else []
            found = None
            for s in range(T - n - 1, -1, -1):
                if seq[s : s + n] == tail:
                    found = seq[s + n : s + n + gamma]
                    break
            if not found:
                found = [seq[-1]]
            found = (found + [found[-1]] * gamma)[:gamma]
            out[b] = torch.tensor(found, device=ids.device)
        return out

    def observe(self, *_args) -> None:  # no state beyond the ids themselves
        pass


class EagleProposer:
    """EAGLE draft proposer. Keeps a per-sequence hidden carry aligned with
    the ids; the carry for verified tokens is the target's fused aux hidden,
    for speculative tokens the draft's own backbone output."""

    def __init__(self, draft, target, aux_layers=None):
        self.draft = draft
        self.target = target
        self.aux_layers = aux_layers
        self.carry: torch.Tensor | None = None

    @torch.no_grad()
    def observe(self, ids: torch.Tensor) -> None:
        """Invalidate the carry after a verify round; the refresh happens
        lazily in propose() (ONE aux forward per round, not two)."""
        self.carry = None

    @torch.no_grad()
    def _refresh(self, ids: torch.Tensor) -> None:
        _, aux = self.target.forward_with_aux(ids, self.aux_layers)
        self.carry = self.draft.fuse_aux(aux)

    @torch.no_grad()
    def propose(self, ids: torch.Tensor, gamma: int) -> torch.Tensor:
        if self.carry is None or self.carry.shape[1] != ids.shape[1]:
            self._refresh(ids)
        cur_ids, cur_carry = ids, self.carry
        out = []
        for _ in range(gamma):
            h = self.draft.backbone(cur_ids, cur_carry)
            logits = self.draft.lm_head(h[:, -1])
            nxt = logits.argmax(-1, keepdim=True)
            out.append(nxt)
            cur_ids = torch.cat([cur_ids, nxt], dim=1)
            # recycle the draft's own last hidden as the carry for the new token
            cur_carry = torch.cat([cur_carry, h[:, -1:]], dim=1)
        return torch.cat(out, dim=1)


@torch.no_grad()
def speculative_generate(
    target,
    proposer,
    input_ids: torch.Tensor,
    max_new_tokens: int = 64,
    gamma: int = 4,
    eos_token_id: int | None = None,
) -> tuple[torch.Tensor, SpecStats]:
    """Greedy speculative decode. Output equals plain greedy decoding of
    ``target`` regardless of proposer quality (the guarantee); the proposer
    only affects speed. Batch size 1 per call (ragged accepts)."""
    assert input_ids.shape[0] == 1, "speculative_generate is per-sequence"
    target.eval()
    ids = input_ids
    stats = SpecStats()
    produced = 0
    while produced < max_new_tokens:
        g = min(gamma, max_new_tokens - produced)
        prop = proposer.propose(ids, g)                       # [1, g]
        ext = torch.cat([ids, prop], dim=1)                   # [1, T+g]
        logits = target(ext)                                  # ONE verify call
        stats.target_calls += 1
        stats.proposed += g
        # target's greedy choice after each prefix position T-1 .. T+g-1
        greedy = logits[:, ids.shape[1] - 1 :, :].argmax(-1)  # [1, g+1]
        n_acc = 0
        while n_acc < g and prop[0, n_acc] == greedy[0, n_acc]:
            n_acc += 1
        accepted_tokens = prop[:, :n_acc]
        correction = greedy[:, n_acc : n_acc + 1]             # bonus/corrected token
        ids = torch.cat([ids, accepted_tokens, correction], dim=1)
        stats.accepted += n_acc
        stats.per_round_accepts.append(n_acc)
        produced += n_acc + 1
        proposer.observe(ids)
        if eos_token_id is not None and bool((ids[0, -n_acc - 1 :] == eos_token_id).any()):
            break
    if produced > max_new_tokens:  # the +1 correction can overshoot by one
        ids = ids[:, : ids.shape[1] - (produced - max_new_tokens)]
        produced = max_new_tokens
    stats.tokens_out = produced
    return ids, stats
