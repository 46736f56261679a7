"""Fused residual-add + LayerNorm (bf16 in/out, fp32 stats).

Autograd wrapper over `ops/hip/residual_ln.hip`, used by the transformer's
post-LN sites (`models/nn/transformer.py`) where the eager path under
autocast spends four passes (add, bf16->fp32 cast, LN, fp32->bf16 at the
next matmul).  CPU / non-bf16 callers use the eager composition.
"""
import os

import torch

from . import hip_ext


class _ResidualLN(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, a, weight, bias, eps):
        ops = hip_ext.require()
        y, s, mean, rstd = ops.residual_ln_fwd(
            x.contiguous(), None if a is None else a.contiguous(),
            weight.float(), bias.float(), eps)
        ctx.save_for_backward(s, mean, rstd, weight)
        ctx.has_residual = a is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        ops = hip_ext.require()
        s, mean, rstd, weight = ctx.saved_tensors
        dsum, dw, db = ops.residual_ln_bwd(
            dy.contiguous().to(s.dtype), s, mean, rstd, weight.float())
        da = dsum if ctx.has_residual else None
        return dsum, da, dw.to(weight.dtype), db.to(weight.dtype), None


def fused_residual_ln(x, residual, ln):
    """LN(x + residual) (residual may be None) through the HIP kernel when
    on GPU in bf16 with a supported width; eager otherwise."""
    C = x.shape[-1]
    if (x.is_cuda and x.dtype == torch.bfloat16 and C == 256
            and (residual is None or residual.dtype == torch.bfloat16)
            and os.environ.get('DISTAR_AMD_DISABLE_HIP') != '1'
            and os.environ.get('DISTAR_AMD_FUSED_LN') != '0'):
        return _ResidualLN.apply(x, residual, ln.weight, ln.bias, ln.eps)
    s = x if residual is None else x + residual
    return ln(s)
