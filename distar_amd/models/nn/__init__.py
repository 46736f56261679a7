from .blocks import (fc_block, fc_block2, conv2d_block, deconv2d_block, GLU,
                     build_activation, build_normalization, ResBlock, ResFCBlock,
                     ResFCBlock2, GatedResBlock, sequence_mask,
                     one_hot_embedding, binary_embedding, get_binary_embed_mat)
from .transformer import Transformer, TransformerLayer, Attention, AttentionPool
from .lnlstm import script_lnlstm, LayerNormLSTMCell, StackedLSTM
