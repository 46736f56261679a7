"""Fused masked cross-entropy (K13, experimental).

Autograd wrapper over `ops/hip/ce_loss.hip` for the SL location-head loss
(reference `sl_training/sl_loss.py` uses `F.cross_entropy` on an
(N, 24320) fp32 tensor).  Forward returns per-row losses like
``F.cross_entropy(..., reduction='none') * mask``; callers reduce.

Validated on MI355X (round-2 GPU numerics tests); the HIP path is the
default on CUDA fp32 — set ``DISTAR_AMD_FUSED_CE=0`` to force eager.
"""
import os

import torch


class _FusedMaskedCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels, mask):
        from . import hip_ext
        ops = hip_ext.require()
        loss, lse = ops.masked_ce_fwd(logits, labels, mask)
        ctx.save_for_backward(logits, labels, lse,
                              mask if mask is not None else torch.empty(0))
        return loss

    @staticmethod
    def backward(ctx, gout):
        from . import hip_ext
        ops = hip_ext.require()
        logits, labels, lse, mask = ctx.saved_tensors
        mask = mask if mask.numel() else None
        dlogits = ops.masked_ce_bwd(logits, labels, mask, lse,
                                    gout.contiguous())
        return dlogits, None, None


def masked_cross_entropy(logits, labels, mask=None):
    """Per-row CE ``(logsumexp(logits_i) - logits_i[label_i]) * mask_i``.

    logits: (..., C) float; labels: (...,) long; mask: (...,) float or None.
    """
    if logits.is_cuda:
        logits = logits.float()
    use_hip = (logits.is_cuda and logits.dtype == torch.float32
               and os.environ.get('DISTAR_AMD_FUSED_CE', '1') != '0')
    C = logits.shape[-1]
    shape = labels.shape
    if use_hip:
        flat = _FusedMaskedCE.apply(
            logits.contiguous().view(-1, C), labels.contiguous().view(-1),
            None if mask is None else
            mask.contiguous().view(-1).float())
        return flat.view(shape)
    loss = torch.nn.functional.cross_entropy(
        logits.reshape(-1, C), labels.reshape(-1),
        reduction='none').view(shape)
    return loss if mask is None else loss * mask
