"""Bilinear 2x upsample through the HIP kernel (ops/hip/upsample.hip) with a
torch fallback on CPU.  Replaces `F.interpolate(scale_factor=2,
mode='bilinear')` in the LocationHead (29% of the SL step as PyTorch's fp32
NCHW kernel — profiles/r01_notes.md)."""
import torch
import torch.nn.functional as F

from . import hip_ext


class _Upsample2x(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = hip_ext.maybe_ext(x)
        return ext.upsample2x(x.contiguous())

    @staticmethod
    def backward(ctx, gout):
        ext = hip_ext.maybe_ext(gout)
        return ext.upsample2x_backward(gout.contiguous())


def upsample2x_bilinear(x):
    import os
    # kernel processes 8-output / 4-source pixel groups per thread
    if (x.is_cuda and x.shape[-1] % 4 == 0
            and os.environ.get('DISTAR_AMD_DISABLE_HIP') != '1'):
        return _Upsample2x.apply(x)
    return F.interpolate(x, scale_factor=2., mode='bilinear')
