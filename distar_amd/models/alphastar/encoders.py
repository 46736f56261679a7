"""Observation encoders: scalar / spatial / entity / value-feature.

Functional parity with the reference's
`distar/agent/default/model/obs_encoder/{scalar,spatial,entity,value}_encoder.py`
and `model/encoder.py`; module names match the reference checkpoints.

MI355X notes:
 - entity field encoding goes through frozen one-hot/binary Embedding lookups
   feeding a single 997->256 projection — the gather+GEMM pair is the shape a
   fused HIP embed kernel replaces.
 - the entity transformer (3 post-LN layers, 2 heads x 128) runs in bf16
   autocast on MFMA via batched GEMMs (K1).
 - scatter_connection is our row-major index_add formulation (K3), see
   `distar_amd/ops/scatter.py`.
"""
from typing import Dict, Tuple, List

import torch
import torch.nn as nn
from torch import Tensor

from ..nn.blocks import (fc_block, conv2d_block, build_activation, ResBlock,
                         sequence_mask, one_hot_embedding, binary_embedding)
from ..nn.transformer import Transformer, AttentionPool
from ...ops.scatter import scatter_connection, spatial_effect_plane
from ...lib.consts import MAX_ENTITY_NUM


def compute_denominator(x: Tensor, dim: int) -> Tensor:
    """Sinusoidal time-encoding denominators (reference scalar_encoder.py:11-16)."""
    x = x // 2 * 2
    x = torch.div(x, dim)
    x = torch.pow(10000., x)
    return torch.div(1., x)


class BeginningBuildOrderEncoder(nn.Module):
    """20-token transformer over the build-order prefix
    (reference scalar_encoder.py:19-54)."""

    def __init__(self, whole_cfg, bo_cfg):
        super().__init__()
        self.whole_cfg = whole_cfg
        self.cfg = bo_cfg
        self.output_dim = self.cfg.output_dim
        self.input_dim = self.cfg.action_one_hot_dim + 20 + self.cfg.binary_dim * 2
        self.act = build_activation(self.cfg.activation)
        self.transformer = Transformer(
            input_dim=self.input_dim, head_dim=self.cfg.head_dim,
            hidden_dim=self.cfg.output_dim * 2, output_dim=self.cfg.output_dim)
        self.embedd_fc = fc_block(self.cfg.output_dim, self.output_dim, activation=self.act)
        self.action_one_hot = one_hot_embedding(self.cfg.action_one_hot_dim)
        self.order_one_hot = one_hot_embedding(20)
        self.location_binary = binary_embedding(self.cfg.binary_dim)

    def forward(self, x, bo_location):
        B, L = x.shape[:2]
        x = self.action_one_hot(x.long())
        seq = torch.eye(L, device=x.device).unsqueeze(0).expand(B, L, L)
        x = torch.cat([x, seq], dim=2)
        loc_x = (bo_location % self.whole_cfg.model.spatial_x).long()
        loc_y = (bo_location // self.whole_cfg.model.spatial_x).long()
        x = torch.cat([x, self.location_binary(loc_x), self.location_binary(loc_y)], dim=2)
        x = self.transformer(x)
        x = x.mean(dim=1)
        return self.embedd_fc(x)


class ScalarEncoder(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.whole_cfg = cfg
        self.cfg = cfg.model.encoder.obs_encoder.scalar_encoder
        self.act = build_activation(self.cfg.activation)
        self.keys = []
        self.scalar_context_keys = []
        self.baseline_feature_keys = []
        self.one_hot_keys = []
        self.encode_modules = nn.ModuleDict()
        for k, item in self.cfg.module.items():
            if k == 'time':
                continue
            if item['arc'] == 'one_hot':
                from ...ops.linear_relu import EmbeddingGEMM
                enc = EmbeddingGEMM(item['num_embeddings'], item['embedding_dim'])
                nn.init.xavier_uniform_(enc.weight)
                self.encode_modules[k] = enc
                self.one_hot_keys.append(k)
            elif item['arc'] == 'fc':
                self.encode_modules[k] = fc_block(item['input_dim'], item['output_dim'],
                                                  activation=self.act)
            if item.get('scalar_context'):
                self.scalar_context_keys.append(k)
            if item.get('baseline_feature'):
                self.baseline_feature_keys.append(k)
        self.position_array = nn.Parameter(
            compute_denominator(torch.arange(0, self.cfg.module.time.output_dim,
                                             dtype=torch.float),
                                self.cfg.module.time.output_dim),
            requires_grad=False)
        self.time_embedding_dim = self.cfg.module.time.output_dim
        bo_cfg = self.cfg.module.beginning_order
        self.encode_modules['beginning_order'] = BeginningBuildOrderEncoder(self.whole_cfg, bo_cfg)

    def time_encoder(self, x: Tensor):
        assert len(x.shape) == 1
        arg = x.unsqueeze(1).float() * self.position_array.unsqueeze(0)
        v = torch.zeros(x.shape[0], self.time_embedding_dim, dtype=torch.float, device=x.device)
        v[:, 0::2] = torch.sin(arg[:, 0::2])
        v[:, 1::2] = torch.cos(arg[:, 1::2])
        return v

    def forward(self, x: Dict[str, Tensor]) -> Tuple[Tensor, Tensor, Tensor]:
        embedded_scalar, scalar_context, baseline_feature = [], [], []
        for key, item in self.cfg.module.items():
            assert key in x, key
            if key == 'time':
                continue
            if item['arc'] == 'one_hot':
                data = x[key].long().clamp_(max=item['num_embeddings'] - 1)
                embedding = self.act(self.encode_modules[key](data))
            elif key == 'beginning_order':
                embedding = self.encode_modules[key](x[key].float(), x['bo_location'].long())
            else:
                embedding = self.encode_modules[key](x[key].float())
            embedded_scalar.append(embedding)
            if key in self.scalar_context_keys:
                scalar_context.append(embedding)
            if key in self.baseline_feature_keys:
                baseline_feature.append(embedding)
        embedded_scalar.append(self.time_encoder(x['time']))
        return (torch.cat(embedded_scalar, dim=1), torch.cat(scalar_context, dim=1),
                torch.cat(baseline_feature, dim=1))


class SpatialEncoder(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.whole_cfg = cfg
        self.cfg = cfg.model.encoder.obs_encoder.spatial_encoder
        self.act = build_activation(self.cfg.activation)
        self.norm = None if self.cfg.norm_type == 'none' else self.cfg.norm_type
        self.project = conv2d_block(self.cfg.input_dim, self.cfg.project_dim, 1, 1, 0,
                                    activation=self.act, norm_type=self.norm)
        dims = [self.cfg.project_dim] + list(self.cfg.down_channels)
        self.down_channels = self.cfg.down_channels
        self.encode_modules = nn.ModuleDict()
        for k, item in self.cfg.module.items():
            if item['arc'] == 'one_hot':
                self.encode_modules[k] = one_hot_embedding(item['num_embeddings'])
        self.downsample = nn.ModuleList()
        for i in range(len(self.down_channels)):
            if self.cfg.downsample_type == 'conv2d':
                self.downsample.append(conv2d_block(dims[i], dims[i + 1], 4, 2, 1,
                                                    activation=self.act, norm_type=self.norm))
            elif self.cfg.downsample_type in ('avgpool', 'maxpool'):
                self.downsample.append(conv2d_block(dims[i], dims[i + 1], 3, 1, 1,
                                                    activation=self.act, norm_type=self.norm))
            else:
                raise KeyError(self.cfg.downsample_type)
        self.res = nn.ModuleList([ResBlock(dims[-1], self.act, norm_type=self.norm)
                                  for _ in range(self.cfg.resblock_num)])
        self.head_type = self.cfg.get('head_type', 'pool')
        if self.head_type == 'fc':
            spatial_y = self.whole_cfg.model.spatial_y
            spatial_x = self.whole_cfg.model.spatial_x
            self.fc = fc_block(dims[-1] * (spatial_y // 8) * (spatial_x // 8),
                               self.cfg.fc_dim, activation=self.act)
        else:
            self.gap = nn.AdaptiveAvgPool2d((1, 1))
            self.fc = fc_block(dims[-1], self.cfg.fc_dim, activation=self.act)

    def _assemble_const_planes(self, x, spatial_y, spatial_x, dtype, device):
        """The 24 input channels that carry no gradient (height map, 6
        one-hot categorical maps, 6 effect planes) built WITHOUT per-field
        embedding gathers or a 56-channel concat: channel-scatter writes into
        one buffer (hot positions only, ~100 MB instead of ~20 GB of
        materialized one-hot fp32 at the SL bench shape)."""
        B = x['height_map'].shape[0]
        off = 0
        with torch.no_grad():
            const = torch.zeros(B, 24, spatial_y, spatial_x, dtype=dtype,
                                device=device)
            for k, item in self.cfg.module.items():
                if item['arc'] == 'other':          # height_map
                    const[:, off] = x[k].to(dtype) / 256
                    off += 1
                elif item['arc'] == 'one_hot':
                    idx = (x[k].long() + off).unsqueeze(1)
                    const.scatter_(1, idx.clamp_(max=off + item['num_embeddings'] - 1),
                                   torch.ones((), dtype=dtype, device=device)
                                   .expand(B, 1, spatial_y, spatial_x))
                    off += item['num_embeddings']
                elif item['arc'] == 'scatter':      # effect plane
                    flat = const[:, off].reshape(B, spatial_y * spatial_x)
                    pos = x[k].long().clamp(0, spatial_y * spatial_x - 1)
                    flat.scatter_(1, pos, torch.ones((), dtype=dtype, device=device)
                                  .expand(B, pos.shape[1]))
                    off += 1
        return const, off

    def forward(self, x: Dict[str, Tensor], scatter_map: Tensor) -> Tuple[Tensor, List[Tensor]]:
        spatial_y = self.whole_cfg.model.spatial_y
        spatial_x = self.whole_cfg.model.spatial_x
        proj_dtype = torch.bfloat16 if torch.is_autocast_enabled() \
            else self.project[0].weight.dtype
        const, n_const = self._assemble_const_planes(
            x, spatial_y, spatial_x, proj_dtype, scatter_map.device)
        # split the 1x1 projection: conv(cat([const, scatter])) ==
        # conv(const) + conv(scatter) with the weight sliced — skips the
        # 56-channel concat while keeping project.0.weight whole (checkpoint
        # layout) and differentiable wrt both weight slices and scatter_map
        conv = self.project[0]
        w = conv.weight
        from ...ops.conv2d import conv2d as hip_conv2d
        out = hip_conv2d(const, w[:, :n_const].contiguous()) + \
            hip_conv2d(scatter_map.to(const.dtype),
                       w[:, n_const:].contiguous(), bias=conv.bias)
        for layer in list(self.project)[1:]:        # norm/act of the block
            out = layer(out)
        if getattr(conv, 'fuse_relu', False):       # act folded into the
            out = torch.relu(out)                   # conv it bypassed here
        map_skip = []
        for i in range(len(self.downsample)):
            map_skip.append(out)
            if self.cfg.downsample_type == 'avgpool':
                out = torch.nn.functional.avg_pool2d(out, 2, 2)
            elif self.cfg.downsample_type == 'maxpool':
                from ...ops.conv2d import max_pool2x2
                out = max_pool2x2(out)
            out = self.downsample[i](out)
        for block in self.res:
            map_skip.append(out)
            out = block(out)
        if self.head_type != 'fc':
            out = self.gap(out)
        out = out.reshape(out.shape[0], -1)
        return self.fc(out), map_skip


class EntityEncoder(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.whole_cfg = cfg
        self.cfg = cfg.model.encoder.obs_encoder.entity_encoder
        self.encode_modules = nn.ModuleDict()
        for k, item in self.cfg.module.items():
            if item['arc'] == 'one_hot':
                self.encode_modules[k] = one_hot_embedding(item['num_embeddings'])
            elif item['arc'] == 'binary':
                self.encode_modules[k] = binary_embedding(item['num_embeddings'])
        self.act = build_activation(self.cfg.activation)
        self.transformer = Transformer(
            input_dim=self.cfg.input_dim, head_dim=self.cfg.head_dim,
            hidden_dim=self.cfg.hidden_dim, output_dim=self.cfg.output_dim,
            head_num=self.cfg.head_num, mlp_num=self.cfg.mlp_num,
            layer_num=self.cfg.layer_num, dropout_ratio=self.cfg.dropout_ratio,
            activation=self.act, ln_type=self.cfg.ln_type)
        self.entity_fc = fc_block(self.cfg.output_dim, self.cfg.output_dim, activation=self.act)
        self.embed_fc = fc_block(self.cfg.output_dim, self.cfg.output_dim, activation=self.act)
        reduce_type = self.whole_cfg.model.entity_reduce_type
        if reduce_type == 'attention_pool':
            self.attention_pool = AttentionPool(key_dim=self.cfg.output_dim, head_num=2,
                                                output_dim=self.cfg.output_dim)
        elif reduce_type == 'attention_pool_add_num':
            self.attention_pool = AttentionPool(key_dim=self.cfg.output_dim, head_num=2,
                                                output_dim=self.cfg.output_dim,
                                                max_num=MAX_ENTITY_NUM + 1)
        # field metadata for the fused HIP embed kernel (K2): kind/offset/size
        # per field in module order, plus each field's column in its stack
        kinds, offsets, sizes, src_idx = [], [], [], []
        off = n_int = n_float = 0
        for k, item in self.cfg.module.items():
            if item['arc'] == 'one_hot':
                kinds.append(0)
                sizes.append(item['num_embeddings'])
                src_idx.append(n_int)
                n_int += 1
                width = item['num_embeddings']
            elif item['arc'] == 'binary':
                kinds.append(1)
                sizes.append(item['num_embeddings'])
                src_idx.append(n_int)
                n_int += 1
                width = item['num_embeddings']
            else:  # unsqueeze
                kinds.append(2)
                sizes.append(1)
                src_idx.append(n_float)
                n_float += 1
                width = 1
            offsets.append(off)
            off += width
        assert off == self.cfg.input_dim, (off, self.cfg.input_dim)
        self._embed_meta_cpu = (torch.tensor(kinds, dtype=torch.int32),
                                torch.tensor(offsets, dtype=torch.int32),
                                torch.tensor(sizes, dtype=torch.int32),
                                torch.tensor(src_idx, dtype=torch.int32))
        self._embed_meta_dev = None

    def embed_fields(self, x: Dict[str, Tensor]) -> Tensor:
        """36 field encoders -> (B, N, 997) concat.

        On device this is ONE HIP kernel (`entity_embed_kernel`, K2) writing
        the mostly-zero bf16 feature rows directly from the raw int fields —
        the eager path is 36 embedding gathers + a concat (~8 GB/step of
        fp32 traffic at the SL bench shape)."""
        import os
        first = next(iter(x.values()))
        if first.is_cuda and os.environ.get('DISTAR_AMD_DISABLE_HIP') != '1':
            from ...ops import hip_ext
            ext = hip_ext.maybe_ext(first)
            B, N = first.shape[:2]
            ints, floats = [], []
            for k, item in self.cfg.module.items():
                if item['arc'] in ('one_hot', 'binary'):
                    ints.append(x[k].reshape(B * N))
                else:
                    floats.append(x[k].reshape(B * N).float())
            int_stack = torch.stack(ints, dim=-1).int().contiguous()
            float_stack = torch.stack(floats, dim=-1).contiguous() if floats \
                else torch.empty(B * N, 0, device=first.device)
            if self._embed_meta_dev is None or \
                    self._embed_meta_dev[0].device != first.device:
                self._embed_meta_dev = tuple(t.to(first.device)
                                             for t in self._embed_meta_cpu)
            kinds, offsets, sizes, src_idx = self._embed_meta_dev
            out = ext.entity_embed(int_stack, float_stack, kinds, offsets,
                                   sizes, src_idx, self.cfg.input_dim)
            out = out.view(B, N, self.cfg.input_dim)
            return out if torch.is_autocast_enabled() else out.float()
        parts = []
        for k, item in self.cfg.module.items():
            assert k in x, k
            if item['arc'] == 'one_hot':
                data = x[k].long().clamp(min=0, max=item['num_embeddings'] - 1)
                parts.append(self.encode_modules[k](data))
            elif item['arc'] == 'binary':
                parts.append(self.encode_modules[k](x[k].long()))
            elif item['arc'] == 'unsqueeze':
                parts.append(x[k].float().unsqueeze(-1))
        return torch.cat(parts, dim=-1)

    def forward(self, x: Dict[str, Tensor], entity_num: Tensor):
        x = self.embed_fields(x)
        mask = sequence_mask(entity_num, max_len=x.shape[1])
        x = self.transformer(x, mask=mask)
        entity_embeddings = self.entity_fc(self.act(x))
        reduce_type = self.whole_cfg.model.entity_reduce_type
        if reduce_type in ('entity_num', 'selected_units_num'):
            x_mask = x * mask.unsqueeze(2)
            embedded_entity = x_mask.sum(dim=1) / entity_num.unsqueeze(-1)
        elif reduce_type == 'constant':
            x_mask = x * mask.unsqueeze(2)
            embedded_entity = x_mask.sum(dim=1) / 512
        elif reduce_type == 'attention_pool':
            embedded_entity = self.attention_pool(x, mask=mask.unsqueeze(2))
        elif reduce_type == 'attention_pool_add_num':
            embedded_entity = self.attention_pool(x, num=entity_num, mask=mask.unsqueeze(2))
        else:
            raise NotImplementedError(reduce_type)
        embedded_entity = self.embed_fc(embedded_entity)
        return entity_embeddings, embedded_entity, mask


class Encoder(nn.Module):
    """Scalar + entity + spatial encode, with the entity->map scatter
    (reference model/encoder.py)."""

    def __init__(self, cfg):
        super().__init__()
        self.whole_cfg = cfg
        self.cfg = cfg.model.encoder
        self.encoder = nn.ModuleDict()
        self.scalar_encoder = ScalarEncoder(cfg)
        self.spatial_encoder = SpatialEncoder(cfg)
        self.entity_encoder = EntityEncoder(cfg)
        self.scatter_project = fc_block(self.cfg.scatter.input_dim,
                                        self.cfg.scatter.output_dim, activation='relu')
        self.scatter_dim = self.cfg.scatter.output_dim
        self.scatter_type = self.cfg.scatter.get('scatter_type', 'cover')

    def forward(self, spatial_info: Dict[str, Tensor], entity_info: Dict[str, Tensor],
                scalar_info: Dict[str, Tensor], entity_num: Tensor):
        embedded_scalar, scalar_context, baseline_feature = self.scalar_encoder(scalar_info)
        entity_embeddings, embedded_entity, entity_mask = self.entity_encoder(entity_info, entity_num)
        entity_location = torch.cat([entity_info['x'].unsqueeze(-1),
                                     entity_info['y'].unsqueeze(-1)], dim=-1)
        shape = spatial_info['height_map'].shape
        project_embeddings = self.scatter_project(entity_embeddings)
        project_embeddings = project_embeddings * entity_mask.unsqueeze(2)
        scatter_map = scatter_connection(shape, project_embeddings, entity_location,
                                         self.scatter_dim, self.scatter_type)
        embedded_spatial, map_skip = self.spatial_encoder(spatial_info, scatter_map)
        lstm_input = torch.cat([embedded_scalar, embedded_entity, embedded_spatial], dim=-1)
        return lstm_input, scalar_context, baseline_feature, entity_embeddings, map_skip


class ValueEncoder(nn.Module):
    """Critic-only features from both players' observations
    (reference obs_encoder/value_encoder.py; key layout matched: learnable
    Embedding per one-hot module, downsample as a Sequential with interleaved
    MaxPool, output order [fc..., spatial, beginning_order])."""

    def __init__(self, cfg):
        super().__init__()
        self.whole_cfg = cfg
        self.cfg = cfg.model.value.encoder
        self.act = build_activation('relu')
        self.encode_modules = nn.ModuleDict()
        for k, item in self.cfg.modules.items():
            if item['arc'] == 'fc':
                self.encode_modules[k] = fc_block(item['input_dim'], item['output_dim'],
                                                  activation=self.act)
            elif item['arc'] == 'one_hot':
                from ...ops.linear_relu import EmbeddingGEMM
                self.encode_modules[k] = EmbeddingGEMM(item['num_embeddings'],
                                                       item['embedding_dim'])
        bo_cfg = self.cfg.modules.beginning_order
        self.encode_modules['beginning_order'] = BeginningBuildOrderEncoder(self.whole_cfg, bo_cfg)
        self.scatter_project = fc_block(self.cfg.scatter.scatter_input_dim,
                                        self.cfg.scatter.scatter_dim, activation=self.act)
        self.scatter_type = self.cfg.scatter.scatter_type
        self.scatter_dim = self.cfg.scatter.scatter_dim
        sp = self.cfg.spatial
        self.project = conv2d_block(sp.input_dim, sp.project_dim, 1, 1, 0, activation=self.act)
        dims = [sp.project_dim] + list(sp.down_channels)
        from ...ops.conv2d import MaxPool2x2HIP
        down_layers = []
        for i in range(len(sp.down_channels)):
            down_layers.append(MaxPool2x2HIP())
            down_layers.append(conv2d_block(dims[i], dims[i + 1], 3, 1, 1, activation=self.act))
        self.downsample = nn.Sequential(*down_layers)
        self.resblock_num = sp.resblock_num
        self.res = nn.ModuleList([ResBlock(dims[-1], self.act, norm_type=None)
                                  for _ in range(sp.resblock_num)])
        spatial_y = self.whole_cfg.model.spatial_y
        spatial_x = self.whole_cfg.model.spatial_x
        self.spatial_fc = fc_block(dims[-1] * (spatial_y // 8) * (spatial_x // 8),
                                   sp.spatial_fc_dim, activation=self.act)

    def forward(self, x):
        fc_embedding, spatial_embedding = [], []
        for k, item in self.cfg.modules.items():
            if item['arc'] == 'fc':
                fc_embedding.append(self.encode_modules[k](x[k].float()))
            elif item['arc'] == 'one_hot':
                spatial_embedding.append(self.encode_modules[k](x[k].long()))
        bo_embedding = self.encode_modules['beginning_order'](
            x['beginning_order'].float(), x['bo_location'].long())
        fc_embedding = torch.cat(fc_embedding, dim=-1)
        spatial_embedding = torch.cat(spatial_embedding, dim=-1)
        project_embedding = self.scatter_project(spatial_embedding)
        unit_mask = sequence_mask(x['total_unit_count'], max_len=project_embedding.shape[1])
        project_embedding = project_embedding * unit_mask.unsqueeze(2)
        entity_location = torch.cat([x['unit_x'].unsqueeze(-1), x['unit_y'].unsqueeze(-1)], dim=-1)
        b, c, h, w = x['own_units_spatial'].shape
        scatter_map = scatter_connection((b, h, w), project_embedding, entity_location,
                                         self.scatter_dim, self.scatter_type)
        spatial_x = torch.cat([scatter_map,
                               x['own_units_spatial'].to(scatter_map.dtype),
                               x['enemy_units_spatial'].to(scatter_map.dtype)],
                              dim=1)
        spatial_x = self.project(spatial_x)
        spatial_x = self.downsample(spatial_x)
        for i in range(self.resblock_num):
            spatial_x = self.res[i](spatial_x)
        spatial_x = self.spatial_fc(spatial_x.reshape(spatial_x.shape[0], -1))
        return torch.cat([fc_embedding, spatial_x, bo_embedding], dim=-1)
