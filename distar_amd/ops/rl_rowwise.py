"""Fused rowwise entropy / teacher-KL (K13, experimental).

Autograd wrappers over `ops/hip/rl_rowwise.hip` for the RL loss's entropy
and KL terms (reference `rl_training/as_rl_utils.py` entropy/kl; eager
materializes (N,C) softmax+log_softmax intermediates).  Per-row outputs;
callers mask/normalize/mean exactly as the eager path does.

Validated on MI355X (round-2 GPU numerics tests); the HIP path is the
default on CUDA fp32 — set ``DISTAR_AMD_FUSED_RL_ROWWISE=0`` to force eager.
"""
import os

import torch
import torch.nn.functional as F


class _FusedEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits):
        from . import hip_ext
        ops = hip_ext.require()
        ent, lse = ops.entropy_fwd(logits)
        ctx.save_for_backward(logits, lse, ent)
        return ent

    @staticmethod
    def backward(ctx, gout):
        from . import hip_ext
        ops = hip_ext.require()
        logits, lse, ent = ctx.saved_tensors
        return ops.entropy_bwd(logits, lse, ent, gout.contiguous())


class _FusedKL(torch.autograd.Function):
    @staticmethod
    def forward(ctx, teacher_logits, student_logits):
        from . import hip_ext
        ops = hip_ext.require()
        kl, t_lse, s_lse = ops.kl_fwd(teacher_logits, student_logits)
        ctx.save_for_backward(teacher_logits, student_logits, t_lse, s_lse)
        return kl

    @staticmethod
    def backward(ctx, gout):
        from . import hip_ext
        ops = hip_ext.require()
        t_logits, s_logits, t_lse, s_lse = ctx.saved_tensors
        ds = ops.kl_bwd(t_logits, s_logits, t_lse, s_lse, gout.contiguous())
        return None, ds


def _use_hip(t):
    return (t.is_cuda and t.dtype == torch.float32
            and os.environ.get('DISTAR_AMD_FUSED_RL_ROWWISE', '1') != '0')


def rowwise_entropy(logits):
    """Per-row Shannon entropy of softmax(logits); (..., C) -> (...)."""
    if logits.is_cuda:
        logits = logits.float()
    if _use_hip(logits):
        shape = logits.shape[:-1]
        flat = _FusedEntropy.apply(
            logits.contiguous().view(-1, logits.shape[-1]))
        return flat.view(shape)
    log_p = F.log_softmax(logits, dim=-1)
    return -(log_p.exp() * log_p).sum(dim=-1)


def rowwise_kl(teacher_logits, student_logits):
    """Per-row KL(softmax(teacher) || softmax(student)); (..., C) -> (...).
    Teacher gets no gradient (detached), matching the RL loss semantics."""
    if student_logits.is_cuda:
        student_logits = student_logits.float()
        teacher_logits = teacher_logits.float()
    if _use_hip(student_logits):
        shape = student_logits.shape[:-1]
        C = student_logits.shape[-1]
        flat = _FusedKL.apply(
            teacher_logits.detach().contiguous().view(-1, C),
            student_logits.contiguous().view(-1, C))
        return flat.view(shape)
    t_log_p = F.log_softmax(teacher_logits.detach(), dim=-1)
    s_log_p = F.log_softmax(student_logits, dim=-1)
    return (t_log_p.exp() * (t_log_p - s_log_p)).sum(dim=-1)
