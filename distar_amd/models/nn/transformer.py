"""Post/pre-LN transformer over entity tokens + attention pooling.

Functional parity with the reference's `module_utils.py:71-199` (Attention /
TransformerLayer / Transformer) and `:37-68` (AttentionPool).  Module / key
layout matches the reference checkpoints (`attention_pre.0.*`, `project.0.*`,
`layernorm1/2.*`, `mlp.{i}.0.*`, `embedding.0.*`).

MI355X note: on GPU in bf16 with head_dim 128 (the entity transformer), the
attention core runs the hand-written CDNA4 MFMA flash kernel
(ops/hip/entity_attn.hip, K1): the key-padding mask travels as an integer
entity count per row — no (B,1,N,N) mask tensor, no qkv permute copies.
Other shapes (the 20-token build-order mini-transformer, CPU) use the eager
composition with the reference's additive -1e9 mask convention.
"""
import math
import os
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .blocks import fc_block, build_normalization


class Attention(nn.Module):
    def __init__(self, input_dim, head_dim, output_dim, head_num, dropout=None):
        super().__init__()
        self.head_num = head_num
        self.head_dim = head_dim
        self.dropout = dropout
        self.attention_pre = fc_block(input_dim, head_dim * head_num * 3)  # q,k,v
        self.project = fc_block(head_dim * head_num, output_dim)

    def forward(self, x, mask: Optional[torch.Tensor] = None):
        """`mask`: (B, N) boolean key-validity PREFIX mask (sequence_mask of
        the entity count) or None."""
        assert len(x.shape) == 3
        B, N = x.shape[:2]
        qkv = self.attention_pre(x)                      # (B, N, 3*H*D)
        use_hip = (x.is_cuda and self.dropout is None and self.head_dim == 128
                   and qkv.dtype == torch.bfloat16
                   and os.environ.get('DISTAR_AMD_DISABLE_HIP') != '1'
                   and os.environ.get('DISTAR_AMD_ATTN') != '0')
        if use_hip:
            from ...ops.entity_attn import entity_attention
            entity_num = mask.int().sum(dim=-1) if mask is not None else None
            attention = entity_attention(qkv, entity_num, self.head_num,
                                         1.0 / math.sqrt(self.head_dim))
            return self.project(attention)
        qkv = qkv.view(B, N, 3, self.head_num, self.head_dim)
        qkv = qkv.permute(2, 0, 3, 1, 4)  # 3, B, H, N, D
        query, key, value = qkv[0], qkv[1], qkv[2]
        mask4 = None
        if mask is not None:
            # (B, N) key mask -> (B, 1, N, N) broadcast over heads/queries
            mask4 = mask.unsqueeze(1).repeat(1, mask.shape[1], 1).unsqueeze(1)
        if x.is_cuda and self.dropout is None:
            # library fused path (A/B comparisons via DISTAR_AMD_DISABLE_HIP).
            # Additive -1e9 mask keeps the reference's finite-logit semantics
            # for fully-padded query rows (a bool mask would yield NaNs).
            attn_mask = None
            if mask4 is not None:
                attn_mask = torch.zeros(mask4.shape, dtype=query.dtype,
                                        device=query.device)
                attn_mask.masked_fill_(~mask4, -1e9)
            attention = F.scaled_dot_product_attention(query, key, value,
                                                       attn_mask=attn_mask)
        else:
            score = torch.matmul(query, key.transpose(-2, -1)) / math.sqrt(self.head_dim)
            if mask4 is not None:
                score = score.masked_fill(~mask4, -1e9)
            score = F.softmax(score, dim=-1)
            if self.dropout is not None:
                score = self.dropout(score)
            attention = torch.matmul(score, value)                   # B, H, N, D
        attention = attention.permute(0, 2, 1, 3).reshape(B, N, -1)  # B, N, H*D
        return self.project(attention)


class TransformerLayer(nn.Module):
    def __init__(self, input_dim, head_dim, hidden_dim, output_dim, head_num,
                 mlp_num, dropout, activation, ln_type):
        super().__init__()
        self.attention = Attention(input_dim, head_dim, output_dim, head_num, dropout)
        self.layernorm1 = build_normalization('LN')(output_dim)
        self.dropout = dropout
        layers = []
        dims = [output_dim] + [hidden_dim] * (mlp_num - 1) + [output_dim]
        for i in range(mlp_num):
            layers.append(fc_block(dims[i], dims[i + 1], activation=activation))
        if self.dropout is not None:
            layers.append(self.dropout)
        self.mlp = nn.Sequential(*layers)
        self.layernorm2 = build_normalization('LN')(output_dim)
        self.ln_type = ln_type

    def forward(self, x, mask: Optional[torch.Tensor] = None):
        if self.ln_type == 'post':
            a = self.attention(x, mask)
            if self.dropout is not None:
                a = self.dropout(a)
            from ...ops.residual_ln import fused_residual_ln
            x = fused_residual_ln(a, x, self.layernorm1)
            m = self.mlp(x)
            if self.dropout is not None:
                m = self.dropout(m)
            x = fused_residual_ln(m, x, self.layernorm2)
        elif self.ln_type == 'pre':
            a = self.attention(self.layernorm1(x), mask)
            if self.dropout is not None:
                a = self.dropout(a)
            x = x + a
            m = self.mlp(self.layernorm2(x))
            if self.dropout is not None:
                m = self.dropout(m)
            x = x + m
        else:
            raise NotImplementedError(self.ln_type)
        return x, mask


class Transformer(nn.Module):
    def __init__(self, input_dim, head_dim=128, hidden_dim=1024, output_dim=256,
                 head_num=2, mlp_num=2, layer_num=3, dropout_ratio=0.0,
                 activation=nn.ReLU(), ln_type='pre'):
        super().__init__()
        self.embedding = fc_block(input_dim, output_dim, activation=activation)
        self.act = activation
        self.dropout = nn.Dropout(dropout_ratio) if dropout_ratio > 0 else None
        self.layers = nn.ModuleList([
            TransformerLayer(output_dim, head_dim, hidden_dim, output_dim,
                             head_num, mlp_num, self.dropout, self.act, ln_type)
            for _ in range(layer_num)
        ])

    def forward(self, x, mask: Optional[torch.Tensor] = None):
        # mask stays (B, N): Attention expands it for eager paths and turns
        # it into per-row entity counts for the HIP kernel
        x = self.embedding(x)
        if self.dropout is not None:
            x = self.dropout(x)
        for layer in self.layers:
            x, mask = layer(x, mask)
        return x


class AttentionPool(nn.Module):
    """Learned-query attention pooling over tokens
    (reference `module_utils.py:37-68`)."""

    def __init__(self, key_dim, head_num, output_dim, max_num=None):
        super().__init__()
        self.queries = nn.Parameter(torch.zeros(1, 1, head_num, key_dim))
        nn.init.xavier_uniform_(self.queries)
        self.head_num = head_num
        self.add_num = max_num is not None
        if self.add_num:
            self.num_ebed = nn.Embedding(num_embeddings=max_num, embedding_dim=output_dim)
        self.embed_fc = fc_block(key_dim * head_num, output_dim)

    def forward(self, x, num=None, mask=None):
        assert len(x.shape) == 3  # B, N, C
        score = (x.unsqueeze(2) * self.queries).sum(dim=3)  # B, N, H
        if mask is not None:
            assert len(mask.shape) == 3 and mask.shape[-1] == 1
            score = score.masked_fill(~mask.repeat(1, 1, self.head_num).bool(), -1e9)
        score = F.softmax(score, dim=1).unsqueeze(2)          # B, N, 1, H
        pooled = (x.unsqueeze(3) * score).sum(dim=1)          # B, C, H
        pooled = self.embed_fc(pooled.reshape(pooled.shape[0], -1))
        if self.add_num:
            pooled = pooled + F.relu(self.num_ebed(num.long()))
        return F.relu(pooled)
