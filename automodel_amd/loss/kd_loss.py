"""Knowledge-distillation losses: forward-KL (chunked) + soft-target CE.

Reference behavior: nemo_automodel/components/loss/kd_loss.py (forward KL
with chunking) and loss/triton/soft_cross_entropy.py (soft-target CE).
"""

from __future__ import annotations

import torch

from automodel_amd.ops._backend import hip_ops


class _SoftCEHip(torch.autograd.Function):
    """HIP soft CE: -sum softmax(t) * log_softmax(s), token-sum over rows."""

    @staticmethod
    def forward(ctx, s, t):
        s2 = s.reshape(-1, s.shape[-1]).contiguous()
        t2 = t.reshape(-1, t.shape[-1]).contiguous()
        loss, lse_s, lse_t = hip_ops().soft_ce_fwd(s2, t2)
        ctx.save_for_backward(s2, t2, lse_s, lse_t)
        ctx.shape = s.shape
        return loss.sum()

    @staticmethod
    def backward(ctx, dloss):
        s2, t2, lse_s, lse_t = ctx.saved_tensors
        g = hip_ops().soft_ce_bwd(s2, t2, lse_s, lse_t,
                                  dloss.reshape(1).float().contiguous())
        return g.view(ctx.shape), None


def soft_cross_entropy(student_logits: torch.Tensor, teacher_probs: torch.Tensor,
                       mask: torch.Tensor | None = None) -> torch.Tensor:
    """-sum_t sum_v p_teacher * log_softmax(student). Returns SUM over tokens."""
    logp = torch.log_softmax(student_logits.float(), dim=-1)
    per_tok = -(teacher_probs.float() * logp).sum(-1)
    if mask is not None:
        per_tok = per_tok * mask.float()
    return per_tok.sum()


def soft_cross_entropy_from_logits(student_logits: torch.Tensor,
                                   teacher_logits: torch.Tensor) -> torch.Tensor:
    """HIP one-pass soft CE on GPU bf16 (csrc/soft_ce.hip); torch fallback."""
    if student_logits.is_cuda and student_logits.dtype == torch.bfloat16:
        return _SoftCEHip.apply(student_logits, teacher_logits)
    return soft_cross_entropy(student_logits,
                              torch.softmax(teacher_logits.float(), dim=-1))


def forward_kl(
    student_logits: torch.Tensor,
    teacher_logits: torch.Tensor,
    mask: torch.Tensor | None = None,
    temperature: float = 1.0,
    chunk_size: int = 2048,
) -> torch.Tensor:
    """KL(teacher || student), token-sum, chunked over tokens to bound the
    fp32 softmax footprint (reference kd_loss.py chunking)."""
    V = student_logits.shape[-1]
    s = student_logits.reshape(-1, V)
    t = teacher_logits.reshape(-1, V)
    m = mask.reshape(-1) if mask is not None else None
    total = s.new_zeros((), dtype=torch.float32)
    for i in range(0, s.shape[0], chunk_size):
        sc = s[i : i + chunk_size].float() / temperature
        tc = t[i : i + chunk_size].float() / temperature
        logp_s = torch.log_softmax(sc, dim=-1)
        logp_t = torch.log_softmax(tc, dim=-1)
        kl = (logp_t.exp() * (logp_t - logp_s)).sum(-1)
        if m is not None:
            kl = kl * m[i : i + chunk_size].float()
        total = total + kl.sum() * (temperature**2)
    return total


class KDLoss(torch.nn.Module):
    """alpha * CE(student, labels) + (1-alpha) * KL(teacher || student)."""

    def __init__(self, alpha: float = 0.5, temperature: float = 1.0,
                 ignore_index: int = -100, chunk_size: int = 2048):
        super().__init__()
        self.alpha = alpha
        self.temperature = temperature
        self.ignore_index = ignore_index
        self.chunk_size = chunk_size

    def forward(self, student_logits, teacher_logits, labels):
        mask = labels != self.ignore_index
        ce = torch.nn.functional.cross_entropy(
            student_logits.reshape(-1, student_logits.shape[-1]).float(),
            labels.reshape(-1), ignore_index=self.ignore_index, reduction="sum")
        kl = forward_kl(student_logits, teacher_logits, mask,
                        self.temperature, self.chunk_size)
        return self.alpha * ce + (1 - self.alpha) * kl
