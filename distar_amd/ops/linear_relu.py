"""Linear+ReLU fused through the hipBLASLt epilogue.

`torch._addmm_activation` runs the ReLU inside the GEMM epilogue (no
separate clamp pass over the activation tensor: the transformer MLP's
(B*T, 512, 1024) relu alone is ~0.5 ms/pass at the SL bench), but has no
derivative formula — this Function supplies the backward: dy masked via
`threshold_backward` on the saved output (the same subgradient the eager
relu uses), then plain GEMM grads.  The fp32 master weight is cast to
the compute dtype per call, exactly as autocast would.
"""
import os

import torch
import torch.nn as nn
import torch.nn.functional as F


class _LinearReLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        xs = x.reshape(-1, x.shape[-1])
        w = weight.to(x.dtype)
        b = bias.to(x.dtype)
        out = torch._addmm_activation(b, xs, w.t(), use_gelu=False)
        ctx.save_for_backward(xs, w, out)
        ctx.wdtype = weight.dtype
        ctx.bdtype = bias.dtype
        ctx.outer = x.shape[:-1]
        return out.reshape(*x.shape[:-1], w.shape[0])

    @staticmethod
    def backward(ctx, dy):
        xs, w, out = ctx.saved_tensors
        dy2 = dy.reshape(out.shape).to(out.dtype)
        dy2 = torch.ops.aten.threshold_backward(dy2, out, 0)
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            dx = (dy2 @ w).reshape(*ctx.outer, w.shape[1])
        if ctx.needs_input_grad[1]:
            dw = (dy2.t() @ xs).to(ctx.wdtype)
        if ctx.needs_input_grad[2]:
            db = dy2.sum(0).to(ctx.bdtype)
        return dx, dw, db


class FusedLinearReLU(nn.Linear):
    """Drop-in [Linear, ReLU] (state-dict compatible with the Linear at
    the same Sequential index)."""

    def forward(self, x):
        if (x.is_cuda and self.bias is not None
                and (x.dtype == torch.bfloat16
                     or torch.is_autocast_enabled())
                and os.environ.get('DISTAR_AMD_DISABLE_HIP') != '1'
                and os.environ.get('DISTAR_AMD_FUSED_FC_RELU', '1') == '1'):
            return _LinearReLUFn.apply(x.to(torch.bfloat16), self.weight,
                                       self.bias)
        return F.relu(super().forward(x))


class _EmbeddingGEMMFn(torch.autograd.Function):
    """Embedding with the weight gradient as a one-hot GEMM: ATen's
    embedding_dense_backward (sum_and_scatter over int64 indices) ran
    ~3.9 ms/step at the RL bench; a (E, N) x (N, D) hipBLASLt GEMM over a
    bf16 one-hot replaces it."""

    @staticmethod
    def forward(ctx, idx, weight):
        ctx.save_for_backward(idx)
        ctx.E = weight.shape[0]
        ctx.wdtype = weight.dtype
        return F.embedding(idx, weight)

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        dyf = dy.reshape(-1, dy.shape[-1]).to(torch.bfloat16)
        oh = F.one_hot(idx.reshape(-1), ctx.E).to(torch.bfloat16)
        dw = (oh.t() @ dyf).to(ctx.wdtype)
        return None, dw


class EmbeddingGEMM(nn.Embedding):
    """Drop-in nn.Embedding (same state-dict) with the GEMM backward on
    GPU; the native path everywhere else."""

    def forward(self, idx):
        if (idx.is_cuda and self.weight.requires_grad
                and torch.is_grad_enabled()
                and os.environ.get('DISTAR_AMD_EMB_GEMM', '1') == '1'):
            return _EmbeddingGEMMFn.apply(idx, self.weight)
        return super().forward(idx)
