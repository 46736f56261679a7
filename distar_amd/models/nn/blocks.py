"""NN building blocks.

Functional parity with the reference's `distar/ctools/torch_utils/network/*`
(nn_module.py, res_block.py, activation.py, normalization.py, rnn.py), written
fresh for this framework.  Blocks are ``nn.Sequential`` with the Linear/Conv at
index 0 so that state-dict keys (`<name>.0.weight`) match the reference
checkpoint layout exactly (drop-in checkpoint compatibility).
"""
import math

import torch
import torch.nn as nn
import torch.nn.functional as F


# ---------------------------------------------------------------- activation

class GLU(nn.Module):
    """Gated linear unit conditioned on a context vector
    (reference `module_utils.py:508-524`)."""

    def __init__(self, input_dim, output_dim, context_dim, input_type='fc'):
        super().__init__()
        assert input_type in ('fc', 'conv2d')
        if input_type == 'fc':
            self.layer1 = fc_block(context_dim, input_dim)
            self.layer2 = fc_block(input_dim, output_dim)
        else:
            self.layer1 = conv2d_block(context_dim, input_dim, 1, 1, 0)
            self.layer2 = conv2d_block(input_dim, output_dim, 1, 1, 0)

    def forward(self, x, context):
        gate = torch.sigmoid(self.layer1(context))
        return self.layer2(gate * x)


def build_activation(activation, inplace=True):
    """`inplace=False` for activations that follow a 3-D nn.Linear: its
    output is a VIEW of the flattened addmm result, and an in-place op on
    that view makes autograd rebase the graph with THREE same-shape
    copies in backward (~15 ms/step across the fc sites at the SL bench;
    out-of-place relu has none and identical memory traffic)."""
    if isinstance(activation, nn.Module):
        return activation
    if activation == 'relu':
        return nn.ReLU(inplace=inplace)
    if activation == 'glu':
        return GLU
    if activation == 'prelu':
        return nn.PReLU(init=0.0)
    if activation in (None, 'none'):
        return None
    raise KeyError(f'invalid activation: {activation}')


def build_normalization(norm_type, dim=None):
    if norm_type is None or norm_type == 'none':
        return None
    key = norm_type if dim is None else f'{norm_type}{dim}D'
    norms = {
        'BN': nn.BatchNorm1d, 'BN1D': nn.BatchNorm1d, 'BN2D': nn.BatchNorm2d,
        'LN': nn.LayerNorm, 'LN1D': nn.LayerNorm, 'LN2D': nn.LayerNorm,
        'IN': nn.InstanceNorm1d, 'IN1D': nn.InstanceNorm1d, 'IN2D': nn.InstanceNorm2d,
        'SyncBN': nn.SyncBatchNorm, 'SyncBN1D': nn.SyncBatchNorm, 'SyncBN2D': nn.SyncBatchNorm,
    }
    if key in norms:
        return norms[key]
    raise KeyError(f'invalid norm type: {norm_type}')


def _weight_init(weight, init_type='xavier', activation=None):
    if init_type is None:
        return
    if init_type == 'xavier':
        nn.init.xavier_uniform_(weight)
    elif init_type == 'kaiming':
        nn.init.kaiming_uniform_(weight, a=math.sqrt(5))
    elif init_type == 'orthogonal':
        nn.init.orthogonal_(weight)
    else:
        raise KeyError(f'invalid init type: {init_type}')


def sequential_pack(layers):
    assert isinstance(layers, list)
    return nn.Sequential(*layers)


# ------------------------------------------------------------------- blocks

def fc_block(in_channels, out_channels, init_type='xavier', activation=None,
             norm_type=None, use_dropout=False, dropout_probability=0.5):
    """Linear [+norm] [+act] [+dropout]; Linear lives at index 0."""
    act = build_activation(activation, inplace=False) \
        if isinstance(activation, str) else activation
    if isinstance(act, nn.ReLU) and act.inplace:
        act = nn.ReLU(inplace=False)   # see build_activation docstring
    has_norm = norm_type is not None and norm_type != 'none'
    if isinstance(act, nn.ReLU) and not has_norm:
        # relu runs inside the hipBLASLt GEMM epilogue (ops/linear_relu.py);
        # same state-dict keys (the Linear stays at index 0)
        from ...ops.linear_relu import FusedLinearReLU
        block = [FusedLinearReLU(in_channels, out_channels)]
        _weight_init(block[0].weight, init_type, activation)
    else:
        block = [nn.Linear(in_channels, out_channels)]
        _weight_init(block[0].weight, init_type, activation)
        if has_norm:
            block.append(build_normalization(norm_type, dim=1)(out_channels))
        if act is not None:
            block.append(act)
    if use_dropout:
        block.append(nn.Dropout(dropout_probability))
    return sequential_pack(block)


def fc_block2(in_channels, out_channels, activation=None, norm_type=None, gain=1.0):
    """fc_block variant with explicit xavier gain and zero bias
    (reference nn_module.fc_block2; used by value heads)."""
    block = [nn.Linear(in_channels, out_channels)]
    nn.init.xavier_uniform_(block[0].weight, gain)
    nn.init.constant_(block[0].bias, 0.0)
    act = build_activation(activation, inplace=False) \
        if isinstance(activation, str) else activation
    if isinstance(act, nn.ReLU) and act.inplace:
        act = nn.ReLU(inplace=False)   # see build_activation docstring
    if act is not None:
        block.append(act)
    if norm_type is not None and norm_type != 'none':
        block.append(build_normalization(norm_type, dim=1)(out_channels))
    return sequential_pack(block)


def conv2d_block(in_channels, out_channels, kernel_size, stride=1, padding=0,
                 dilation=1, groups=1, init_type='xavier', pad_type='zero',
                 activation=None, norm_type=None):
    block = []
    assert pad_type in ('zero', 'reflect', 'replication')
    if pad_type == 'reflect':
        block.append(nn.ReflectionPad2d(padding))
        padding = 0
    elif pad_type == 'replication':
        block.append(nn.ReplicationPad2d(padding))
        padding = 0
    from ...ops.conv2d import Conv2dHIP
    conv = Conv2dHIP(in_channels, out_channels, kernel_size, stride,
                     padding=padding, dilation=dilation, groups=groups)
    _weight_init(conv.weight, init_type, activation)
    block.append(conv)
    has_norm = norm_type is not None and norm_type != 'none'
    if has_norm:
        block.append(build_normalization(norm_type, dim=2)(out_channels))
    act = build_activation(activation) if isinstance(activation, str) else activation
    if act is not None:
        if isinstance(act, nn.ReLU) and not has_norm:
            # fold the relu into the conv kernel epilogue (saves a full
            # read+write pass; the backward mask reuses the saved output)
            conv.fuse_relu = True
        else:
            block.append(act)
    return sequential_pack(block)


def deconv2d_block(in_channels, out_channels, kernel_size, stride=1, padding=0,
                   output_padding=0, groups=1, init_type='xavier',
                   activation=None, norm_type=None):
    deconv = nn.ConvTranspose2d(in_channels, out_channels, kernel_size, stride,
                                padding=padding, output_padding=output_padding,
                                groups=groups)
    _weight_init(deconv.weight, init_type, activation)
    block = [deconv]
    if norm_type is not None and norm_type != 'none':
        block.append(build_normalization(norm_type, dim=2)(out_channels))
    act = build_activation(activation) if isinstance(activation, str) else activation
    if act is not None:
        block.append(act)
    return sequential_pack(block)


class ResBlock(nn.Module):
    """Two 3x3 conv blocks with residual add (reference res_block.ResBlock)."""

    def __init__(self, in_channels, activation=nn.ReLU(), norm_type='BN'):
        super().__init__()
        self.act = build_activation(activation)
        self.conv1 = conv2d_block(in_channels, in_channels, 3, 1, 1,
                                  activation=self.act, norm_type=norm_type)
        self.conv2 = conv2d_block(in_channels, in_channels, 3, 1, 1,
                                  activation=None, norm_type=norm_type)

    def forward(self, x):
        residual = x
        x = self.conv1(x)
        x = self.conv2(x)
        return self.act(x + residual)


class ResFCBlock(nn.Module):
    """fc1(norm,act) -> fc2(norm) -> +residual -> act."""

    def __init__(self, in_channels, activation=nn.ReLU(), norm_type='BN'):
        super().__init__()
        self.act = build_activation(activation)
        self.fc1 = fc_block(in_channels, in_channels, activation=self.act, norm_type=norm_type)
        self.fc2 = fc_block(in_channels, in_channels, activation=None, norm_type=norm_type)

    def forward(self, x):
        residual = x
        x = self.fc1(x)
        x = self.fc2(x)
        return self.act(x + residual)


class ResFCBlock2(nn.Module):
    """Norm-after-residual variant used by the value nets
    (reference res_block.ResFCBlock2)."""

    def __init__(self, in_channels, activation=nn.ReLU(), norm_type='LN'):
        super().__init__()
        self.act = build_activation(activation)
        self.fc1 = fc_block(in_channels, in_channels, activation=self.act, norm_type=None)
        self.fc2 = fc_block(in_channels, in_channels, activation=None, norm_type=None)
        self.norm = build_normalization(norm_type)(in_channels)

    def forward(self, x):
        residual = x
        x = self.fc1(x)
        x = self.fc2(x)
        return self.norm(x + residual)


class GatedResBlock(nn.Module):
    """Gated residual conv block (reference `module_utils.py:204-231`)."""

    def __init__(self, in_channels, out_channels, kernel_size, stride, padding,
                 activation=nn.ReLU(), norm_type='BN'):
        super().__init__()
        assert stride == 1 and in_channels == out_channels
        self.act = build_activation(activation)
        self.conv1 = conv2d_block(in_channels, out_channels, 3, 1, 1,
                                  activation=self.act, norm_type=norm_type)
        self.conv2 = conv2d_block(out_channels, out_channels, 3, 1, 1,
                                  activation=None, norm_type=norm_type)
        self.GateWeightG = nn.Sequential(
            conv2d_block(out_channels, out_channels, 1, 1, 0, activation=self.act, norm_type=None),
            conv2d_block(out_channels, out_channels, 1, 1, 0, activation=self.act, norm_type=None),
            conv2d_block(out_channels, out_channels, 1, 1, 0, activation=self.act, norm_type=None),
            conv2d_block(out_channels, out_channels, 1, 1, 0, activation=None, norm_type=None),
        )
        self.UpdateSP = nn.Parameter(torch.full((1,), 0.1))

    def forward(self, x, noise_map):
        residual = x
        x = self.conv1(x)
        x = self.conv2(x)
        # keep the compute dtype: the fp32 UpdateSP parameter would promote
        # the whole chain (and every downstream conv) to fp32 under autocast
        x = torch.tanh(x * torch.sigmoid(self.GateWeightG(noise_map))) \
            * self.UpdateSP.to(x.dtype)
        return self.act(x + residual)


class FiLM(nn.Module):
    """Feature-wise linear modulation (reference `module_utils.py:234-241`):
    per-channel affine conditioning ``gamma * x + beta`` with (N, C) gammas
    and betas broadcast over the spatial axes."""

    def forward(self, x, gammas, betas):
        return torch.addcmul(betas[:, :, None, None], gammas[:, :, None, None], x)


class FiLMedResBlock(nn.Module):
    """Conditioned residual conv block (reference `module_utils.py:244-353`).

    Only the configuration the reference actually instantiates is kept
    (`action_arg_head.py:400`: ``FiLMedResBlock(dim, with_cond=[True])``,
    i.e. 3x3 input projection + 3x3 conv + conv-film + residual relu);
    parameter names match for checkpoint-key parity.
    """

    def __init__(self, in_dim, out_dim=None, with_residual=True,
                 with_cond=(True,), kernel_size=3, with_input_proj=3):
        super().__init__()
        out_dim = out_dim or in_dim
        assert kernel_size % 2 == 1 and with_input_proj % 2 == 1
        self.with_residual = with_residual
        self.with_cond = list(with_cond)
        self.input_proj = nn.Conv2d(in_dim, in_dim, kernel_size=with_input_proj,
                                    padding=with_input_proj // 2)
        self.conv1 = nn.Conv2d(in_dim, out_dim, kernel_size=kernel_size,
                               padding=kernel_size // 2)
        self.film = FiLM() if self.with_cond[0] else None
        for m in (self.input_proj, self.conv1):
            nn.init.kaiming_normal_(m.weight)

    def forward(self, x, gammas=None, betas=None):
        x = F.relu(self.input_proj(x))
        out = self.conv1(x)
        if self.film is not None:
            out = self.film(out, gammas, betas)
        if self.with_residual:
            out = F.relu(x + out)
        return out


# ------------------------------------------------------------------ helpers

def sequence_mask(lengths: torch.Tensor, max_len=None) -> torch.Tensor:
    """(B,) lengths -> (B, max_len) bool mask (reference rnn.sequence_mask)."""
    if max_len is None:
        max_len = int(lengths.max())
    return torch.arange(max_len, device=lengths.device).unsqueeze(0) < lengths.unsqueeze(1)


def get_binary_embed_mat(bit_num):
    """(2**bit_num, bit_num) matrix of binary digits, MSB first
    (reference entity_encoder.get_binary_embed_mat)."""
    n = torch.arange(2 ** bit_num, dtype=torch.long)
    bits = (n.unsqueeze(1) >> torch.arange(bit_num - 1, -1, -1)) & 1
    return bits.float()


def one_hot_embedding(num_embeddings):
    """Frozen identity embedding (a one-hot lookup that shows up in the
    reference state_dict as `<name>.weight`)."""
    return nn.Embedding.from_pretrained(torch.eye(num_embeddings), freeze=True, padding_idx=None)


def binary_embedding(bit_num):
    return nn.Embedding.from_pretrained(get_binary_embed_mat(bit_num), freeze=True, padding_idx=None)
