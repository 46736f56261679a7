"""K4: NCHW bf16 conv + 2x2 maxpool through the hand-written MFMA kernels.

`Conv2dHIP` is a drop-in nn.Conv2d (same parameters/state_dict) whose
forward dispatches stride-1 1x1/3x3 convs on GPU bf16 inputs to
`ops/hip/conv2d.hip`: implicit-GEMM with in-kernel im2col staging (no NHWC
transposes, no MIOpen workspace), backward-data through the same kernel
with flipped/transposed weights, backward-weight through the wgrad kernel.
`MaxPool2x2HIP` replaces nn.MaxPool2d(2, 2) with an argmax-saving fwd and a
gather (atomics-free) backward.
"""
import os

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import hip_ext


def _kpad(k):
    return (k + 127) // 128 * 128     # KC=128 chunks in the kernels


def _pack_weight(w, KH, KW):
    """(Cout, Cin, KH, KW) -> (Cout, Kpad) bf16 contiguous."""
    Cout = w.shape[0]
    flat = w.detach().reshape(Cout, -1).to(torch.bfloat16)
    K = flat.shape[1]
    Kp = _kpad(K)
    if Kp != K:
        flat = F.pad(flat, (0, Kp - K))
    return flat.contiguous()


def _pack_weight_flipped(w, KH, KW):
    """Backward-data weights: W'[ci][co][KH-1-dy][KW-1-dx] packed."""
    wf = w.detach().flip(2, 3).permute(1, 0, 2, 3)
    return _pack_weight(wf, KH, KW)


class _Conv2dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, KH, KW, padH, padW, relu=False):
        ops = hip_ext.require()
        wp = _pack_weight(weight, KH, KW)
        b32 = bias.detach().float().contiguous() if bias is not None else None
        out = ops.conv2d_fwd(x, wp, b32, weight.shape[0], KH, KW, padH, padW,
                             relu)
        if relu:
            ctx.save_for_backward(x, weight, out)   # out>0 is the relu mask
        else:
            ctx.save_for_backward(x, weight)
        ctx.relu = relu
        ctx.dims = (KH, KW, padH, padW)
        ctx.has_bias = bias is not None
        return out

    @staticmethod
    def backward(ctx, dy):
        ops = hip_ext.require()
        if ctx.relu:
            x, weight, out = ctx.saved_tensors
            dy = torch.ops.aten.threshold_backward(dy, out, 0)
        else:
            x, weight = ctx.saved_tensors
        KH, KW, padH, padW = ctx.dims
        dy = dy.contiguous().to(torch.bfloat16)
        Cout, Cin = weight.shape[0], weight.shape[1]
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            if KH == 1 and Cin < 16:
                # tiny-Cin 1x1 backward-data is a plain batched GEMM; the
                # direct VALU kernel ran 4.7 ms/step on the value encoder's
                # 16->10 dx at (272, 152, 160) (profiles r2ww)
                w2 = weight.detach().to(torch.bfloat16).reshape(Cout, Cin)
                dx = torch.matmul(
                    w2.t(), dy.reshape(dy.shape[0], Cout, -1)) \
                    .reshape(x.shape).contiguous()
            else:
                wpf = _pack_weight_flipped(weight, KH, KW)
                dx = ops.conv2d_fwd(dy, wpf, None, Cin, KH, KW,
                                    KH - 1 - padH, KW - 1 - padW, False)
        want_bias = ctx.has_bias and ctx.needs_input_grad[2]
        if ctx.needs_input_grad[1]:
            K = Cin * KH * KW
            dwp, dbias = ops.conv2d_wgrad(x, dy, KH, KW, padH, padW,
                                          _kpad(K), want_bias)
            dw = dwp[:K].view(Cin, KH, KW, Cout).permute(3, 0, 1, 2) \
                .contiguous().to(weight.dtype)
            if want_bias and dbias is not None and dbias.numel():
                db = dbias.to(weight.dtype)    # fused into the wgrad kernel
        if want_bias and db is None:
            db = dy.sum(dim=(0, 2, 3)).to(weight.dtype)
        return dx, dw, db, None, None, None, None, None


class _ConvSmallHWFn(torch.autograd.Function):
    """Materialized im2col + batched hipBLASLt GEMM conv (data movement =
    two dedicated HIP kernels: im2col / gather-col2im, no atomics).

    MEASURED NULL RESULT, kept env-gated (DISTAR_AMD_CONV_SMALLHW /
    DISTAR_AMD_CONV_1X1GEMM): the same-box A/B in profiles/r02_notes.md
    (r2gg) has the implicit-GEMM kernels at 382.4 ms/step vs 410.0 with
    the 19x20 stack on this path — the col round-trip plus 2048-image
    tiny batched GEMMs lose to the in-kernel gather that stays in L2."""

    @staticmethod
    def forward(ctx, x, weight, bias, KH, KW, padH, padW):
        ops = hip_ext.require()
        B, Cin, H, W = x.shape
        Cout = weight.shape[0]
        w2 = weight.detach().to(torch.bfloat16).reshape(Cout, -1)
        col = x.reshape(B, Cin, H * W) if KH == 1 else ops.im2col3x3(x)
        # matmul into a preallocated base tensor: returning a view of the
        # matmul result breaks the in-place relu that follows in
        # conv2d_block (autograd forbids in-place on custom-Function views)
        out = torch.empty((B, Cout, H, W), device=x.device,
                          dtype=torch.bfloat16)
        torch.matmul(w2, col, out=out.view(B, Cout, H * W))
        if bias is not None:
            out += bias.detach().to(out.dtype).view(1, -1, 1, 1)
        # keep col for wgrad when it is small (~1.8 GB per 19x20 conv);
        # recompute it in backward for larger-HW shapes
        keep = KH == 1 or col.numel() * 2 <= _col_save_bytes()
        ctx.save_for_backward(x, weight,
                              col if keep else torch.empty(0, device=x.device))
        ctx.kept_col = keep
        ctx.dims = (KH, KW, padH, padW)
        ctx.has_bias = bias is not None
        return out

    @staticmethod
    def backward(ctx, dy):
        ops = hip_ext.require()
        x, weight, col = ctx.saved_tensors
        B, Cin, H, W = x.shape
        Cout = weight.shape[0]
        KH = ctx.dims[0]
        dyf = dy.contiguous().to(torch.bfloat16).reshape(B, Cout, H * W)
        w2 = weight.detach().to(torch.bfloat16).reshape(Cout, -1)
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            dcol = torch.matmul(w2.t(), dyf)     # (B, K, P)
            dx = dcol.view(B, Cin, H, W) if KH == 1 \
                else ops.col2im3x3(dcol.contiguous(), H, W)
        if ctx.needs_input_grad[1]:
            if not ctx.kept_col:
                col = ops.im2col3x3(x)
            # per-image partials in bf16, image sum in fp32: the 2048-image
            # sum averages the per-partial rounding down ~sqrt(B)
            dwb = torch.bmm(dyf, col.transpose(1, 2))
            dw = dwb.sum(0, dtype=torch.float32).view_as(weight) \
                .to(weight.dtype)
        if ctx.has_bias and ctx.needs_input_grad[2]:
            db = dyf.sum(dim=(0, 2)).to(weight.dtype)
        return dx, dw, db, None, None, None, None


def _small_hw_limit():
    # default 0 = always implicit-GEMM: the materialized-col path measured
    # SLOWER at every threshold (r2ff/r2gg same-box A/B: off 382.4 ms,
    # 19x20-only 410.0, +38x40 421.6, +76x80 454.3 — the 2048-image tiny
    # batched GEMMs and the col round-trip lose to the in-kernel gather)
    try:
        return int(os.environ.get('DISTAR_AMD_CONV_SMALLHW', '0'))
    except ValueError:
        return 0


def _col_save_bytes():
    try:
        return int(os.environ.get('DISTAR_AMD_CONV_COL_SAVE_MB', '2048')) << 20
    except ValueError:
        return 2048 << 20


class Conv2dHIP(nn.Conv2d):
    def _use_hip(self, x):
        kh, kw = self.kernel_size
        return (x.is_cuda and x.dtype == torch.bfloat16
                and self.stride == (1, 1) and self.dilation == (1, 1)
                and self.groups == 1 and (kh, kw) in ((1, 1), (3, 3))
                and self.padding == (kh // 2, kw // 2)
                and os.environ.get('DISTAR_AMD_DISABLE_HIP') != '1'
                and os.environ.get('DISTAR_AMD_CONV') != '0')

    fuse_relu = False             # set by conv2d_block when act folds in

    def forward(self, x):
        if self._use_hip(x):
            kh, kw = self.kernel_size
            # 1x1 convs are batched GEMMs outright (col == x, no im2col);
            # 3x3 below the HW threshold go through materialized im2col
            if (kh == 1 and os.environ.get('DISTAR_AMD_CONV_1X1GEMM',
                                           '0') == '1') \
                    or x.shape[2] * x.shape[3] <= _small_hw_limit():
                out = _ConvSmallHWFn.apply(x.contiguous(), self.weight,
                                           self.bias, kh, kw, kh // 2,
                                           kw // 2)
                return F.relu(out) if self.fuse_relu else out
            return _Conv2dFn.apply(x.contiguous(), self.weight, self.bias,
                                   kh, kw, kh // 2, kw // 2, self.fuse_relu)
        if x.is_cuda and os.environ.get('DISTAR_AMD_CONV_DEBUG') == '1':
            import sys
            print(f'[Conv2dHIP fallback] shape={tuple(x.shape)} dtype={x.dtype} '
                  f'k={self.kernel_size} s={self.stride} p={self.padding} '
                  f'd={self.dilation} g={self.groups}', file=sys.stderr,
                  flush=True)
        out = super().forward(x)
        return F.relu(out) if self.fuse_relu else out


class _MaxPool2x2Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ops = hip_ext.require()
        out, idx = ops.maxpool2x2_fwd(x)
        ctx.save_for_backward(idx)
        ctx.hw = (x.shape[2], x.shape[3])
        return out

    @staticmethod
    def backward(ctx, dy):
        ops = hip_ext.require()
        (idx,) = ctx.saved_tensors
        H, W = ctx.hw
        return ops.maxpool2x2_bwd(dy.contiguous().to(torch.bfloat16), idx,
                                  H, W)


def conv2d(x, weight, bias=None, padding=0):
    """Functional conv through the HIP kernels when the shape qualifies
    (1x1 p0 / 3x3 p1, stride 1, GPU bf16); F.conv2d otherwise.  Used by the
    spatial encoder's split projection (sliced-weight calls that bypass the
    Conv2dHIP module)."""
    kh, kw = weight.shape[2], weight.shape[3]
    pad = (padding, padding) if isinstance(padding, int) else tuple(padding)
    if (x.is_cuda and x.dtype == torch.bfloat16 and (kh, kw) in ((1, 1), (3, 3))
            and pad == (kh // 2, kw // 2)
            and os.environ.get('DISTAR_AMD_DISABLE_HIP') != '1'
            and os.environ.get('DISTAR_AMD_CONV') != '0'):
        if (kh == 1 and os.environ.get('DISTAR_AMD_CONV_1X1GEMM', '0') == '1') \
                or x.shape[2] * x.shape[3] <= _small_hw_limit():
            return _ConvSmallHWFn.apply(x.contiguous(), weight, bias, kh, kw,
                                        kh // 2, kw // 2)
        return _Conv2dFn.apply(x.contiguous(), weight, bias, kh, kw,
                               kh // 2, kw // 2)
    return F.conv2d(x, weight, bias, padding=padding)


def max_pool2x2(x):
    """Functional 2x2/2 maxpool through the HIP kernels when possible."""
    if (x.is_cuda and x.dtype == torch.bfloat16
            and x.shape[2] % 2 == 0 and x.shape[3] % 2 == 0
            and os.environ.get('DISTAR_AMD_DISABLE_HIP') != '1'
            and os.environ.get('DISTAR_AMD_CONV') != '0'):
        return _MaxPool2x2Fn.apply(x.contiguous())
    return F.max_pool2d(x, 2, 2)


class MaxPool2x2HIP(nn.MaxPool2d):
    def __init__(self):
        super().__init__(2, 2)

    def forward(self, x):
        if (x.is_cuda and x.dtype == torch.bfloat16
                and x.shape[2] % 2 == 0 and x.shape[3] % 2 == 0
                and os.environ.get('DISTAR_AMD_DISABLE_HIP') != '1'
                and os.environ.get('DISTAR_AMD_CONV') != '0'):
            return _MaxPool2x2Fn.apply(x.contiguous())
        return super().forward(x)
