"""K1: entity-transformer attention (hand-written CDNA4 MFMA kernels).

Autograd wrapper over `ops/hip/entity_attn.hip` — flash-style fwd/bwd over
the packed qkv projection, with the key-padding mask passed as an integer
entity count per batch row instead of a materialized (B,1,N,N) mask tensor
(reference `model/module_utils.py:71-151` builds additive -1e9 masks; the
kernel reproduces that convention exactly in the P->0 limit).

Input is the `attention_pre` fc output (B, N, 3*H*128) bf16; output is the
(B, N, H*128) attention result laid out exactly as the eager
`permute(0,2,1,3).reshape(B,N,-1)` would produce, so `project` consumes it
directly — the permute/reshape copies disappear along with the mask.
"""
import torch

from . import hip_ext


class _EntityAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, qkv, entity_num, head_num, scale):
        ops = hip_ext.require()
        out, lse = ops.entity_attn_fwd(qkv, entity_num, head_num, scale)
        ctx.save_for_backward(
            qkv, out, lse,
            entity_num if entity_num is not None else torch.empty(0))
        ctx.head_num = head_num
        ctx.scale = scale
        return out

    @staticmethod
    def backward(ctx, dout):
        ops = hip_ext.require()
        qkv, out, lse, entity_num = ctx.saved_tensors
        entity_num = entity_num if entity_num.numel() else None
        dqkv = ops.entity_attn_bwd(qkv, entity_num, out,
                                   dout.contiguous().to(qkv.dtype), lse,
                                   ctx.head_num, ctx.scale)
        return dqkv, None, None, None


def entity_attention(qkv, entity_num, head_num, scale):
    """qkv: (B, N, 3*H*128) bf16 contiguous; entity_num: (B,) int32 valid-key
    counts (prefix mask) or None; returns (B, N, H*128) bf16."""
    if entity_num is not None:
        entity_num = entity_num.to(torch.int32).contiguous()
    return _EntityAttention.apply(qkv.contiguous(), entity_num, head_num,
                                  scale)
