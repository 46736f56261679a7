"""Autograd wrapper for the fused LN-LSTM layer kernel (K5).

The per-layer forward becomes:
  1. one big GEMM + LayerNorm for LN_i(x @ W_ih^T) over all T steps (torch,
     bf16 autocast -> MFMA),
  2. ONE kernel launch for the entire T-step recurrent unroll
     (`lnlstm_forward_kernel`),
and backward:
  3. ONE kernel for the reverse unroll emitting dgates / d(hgates_raw),
  4. dW_hh as a single (T*B) GEMM; dx / dW_ih flow through torch's autograd
     of step 1.

Numerics: fp32 state/LN, bf16 W_hh products (matching autocast GEMM
precision).  Golden-tested against the eager fp32 cell in
tests/test_gpu.py::test_fused_lnlstm_matches_eager.
"""
import torch

from . import hip_ext


class FusedLNLSTMLayer(torch.autograd.Function):
    @staticmethod
    def forward(ctx, igates, h0, c0, w_hh, lnh_w, lnh_b, lnc_w, lnc_b):
        ext = hip_ext.maybe_ext(igates)
        w_bf = w_hh.detach().bfloat16().contiguous()
        h_all, c_all, hgates_raw, cellraw = ext.lnlstm_forward(
            igates, h0.contiguous(), c0.contiguous(), w_bf,
            lnh_w.detach().float(), lnh_b.detach().float(),
            lnc_w.detach().float(), lnc_b.detach().float())
        ctx.save_for_backward(igates, h_all, c_all, hgates_raw, cellraw,
                              w_hh, lnh_w, lnh_b, lnc_w, lnc_b)
        return h_all[1:], h_all[-1], c_all[-1]

    @staticmethod
    def backward(ctx, dout, dhT, dcT):
        (igates, h_all, c_all, hgates_raw, cellraw, w_hh,
         lnh_w, lnh_b, lnc_w, lnc_b) = ctx.saved_tensors
        ext = hip_ext.maybe_ext(igates)
        w_t_bf = w_hh.detach().t().contiguous().bfloat16()
        # dout from the stack is dense; dhT/dcT may be None or zeros
        dhT_c = dhT.contiguous().float() if dhT is not None else torch.Tensor()
        dcT_c = dcT.contiguous().float() if dcT is not None else torch.Tensor()
        (digates, dhgates_raw, dh0, dc0, dlnh_w, dlnh_b, dlnc_w, dlnc_b) = \
            ext.lnlstm_backward(
                dout.contiguous().float(), dhT_c, dcT_c,
                igates, h_all, c_all, hgates_raw, cellraw, w_t_bf,
                lnh_w.detach().float(), lnh_b.detach().float(),
                lnc_w.detach().float(), lnc_b.detach().float())
        T, B, G = igates.shape
        H = G // 4
        # dW_hh = sum_t dhgates_raw[t]^T @ h_prev[t]  -> one flat GEMM
        dw_hh = dhgates_raw.reshape(T * B, G).t().mm(
            h_all[:-1].reshape(T * B, H))
        return (digates, dh0, dc0, dw_hh, dlnh_w, dlnh_b, dlnc_w, dlnc_b)


def fused_lnlstm_layer(input_seq, state, cell):
    """Run one LN-LSTM layer (T, B, in) -> (T, B, H) with the fused kernels.

    ``cell`` is a models.nn.lnlstm.LayerNormLSTMCell (weights + LN modules).
    """
    T, B = input_seq.shape[0], input_seq.shape[1]
    igates = cell.layernorm_i(
        input_seq.reshape(T * B, -1).mm(cell.weight_ih.t())
    ).view(T, B, -1).float()
    h, c = state
    out, hT, cT = FusedLNLSTMLayer.apply(
        igates, h.float(), c.float(), cell.weight_hh,
        cell.layernorm_h.weight, cell.layernorm_h.bias,
        cell.layernorm_c.weight, cell.layernorm_c.bias)
    return out, (hT, cT)
