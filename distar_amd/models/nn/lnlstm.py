"""Layer-norm LSTM (the AlphaStar core memory).

Semantics match the reference's `distar/agent/default/model/lstm.py:120-234`
(LayerNormLSTMCell / LSTMLayer / StackedLSTM, factory `script_lnlstm`) and the
checkpoint key layout is identical (`layers.{l}.cell.weight_ih`, `.weight_hh`,
`.layernorm_{i,h,c}.*`).

MI355X-first restructure: the reference multiplies x·W_ih one timestep at a
time inside the unroll.  Here each layer precomputes LN_i(x·W_ih) for ALL T
steps in one (T·B, 4H) GEMM + one LayerNorm launch (LN is row-local, so this
is numerically identical), leaving only the h-recurrent half in the per-step
loop.  When the fused HIP extension is available on device, the whole
per-step tail (h·W_hh MFMA + LN + gate math + cell LN) runs as one persistent
kernel per layer — see `distar_amd/ops/lnlstm.py`.
"""
from typing import List, Optional, Tuple

import torch
import torch.nn as nn
from torch import Tensor
from torch.nn import Parameter


class LayerNormLSTMCell(nn.Module):
    def __init__(self, input_size, hidden_size):
        super().__init__()
        self.input_size = input_size
        self.hidden_size = hidden_size
        self.weight_ih = Parameter(torch.randn(4 * hidden_size, input_size))
        self.weight_hh = Parameter(torch.randn(4 * hidden_size, hidden_size))
        # layernorms provide the learnable biases
        self.layernorm_i = nn.LayerNorm(4 * hidden_size)
        self.layernorm_h = nn.LayerNorm(4 * hidden_size)
        self.layernorm_c = nn.LayerNorm(hidden_size)

    def forward(self, input: Tensor, state: Tuple[Tensor, Tensor]) -> Tuple[Tensor, Tuple[Tensor, Tensor]]:
        hx, cx = state
        igates = self.layernorm_i(torch.mm(input, self.weight_ih.t()))
        hy, cy = self._recurrent_step(igates, hx, cx)
        return hy, (hy, cy)

    def _recurrent_step(self, igates: Tensor, hx: Tensor, cx: Tensor) -> Tuple[Tensor, Tensor]:
        """Everything after LN_i(x·W_ih): the per-step recurrent tail."""
        hgates = self.layernorm_h(torch.mm(hx, self.weight_hh.t()))
        gates = igates + hgates
        i, f, g, o = gates.chunk(4, 1)
        i = torch.sigmoid(i)
        f = torch.sigmoid(f)
        g = torch.tanh(g)
        o = torch.sigmoid(o)
        cy = self.layernorm_c(f * cx + i * g)
        hy = o * torch.tanh(cy)
        return hy, cy


class LSTMLayer(nn.Module):
    def __init__(self, cell, *cell_args):
        super().__init__()
        self.cell = cell(*cell_args)

    def forward(self, input: Tensor, state: Tuple[Tensor, Tensor]) -> Tuple[Tensor, Tuple[Tensor, Tensor]]:
        import os
        use_fused = input.is_cuda and os.environ.get('DISTAR_AMD_DISABLE_HIP') != '1'
        if use_fused and self.cell.hidden_size >= 128 \
                and os.environ.get('DISTAR_AMD_NO_FUSED_CORE') == '1':
            use_fused = False
        if use_fused and self.cell.hidden_size < 128 \
                and os.environ.get('DISTAR_AMD_NO_FUSED_SU') == '1':
            use_fused = False
        if use_fused:
            # fused HIP path: ONE kernel for the whole T-step unroll
            from ...ops.lnlstm import fused_lnlstm_layer
            return fused_lnlstm_layer(input, state, self.cell)
        T, B = input.shape[0], input.shape[1]
        cell = self.cell
        # One big GEMM + LN for the input half of all T steps.
        igates_all = cell.layernorm_i(
            torch.mm(input.reshape(T * B, -1), cell.weight_ih.t())
        ).view(T, B, -1)
        h, c = state
        outputs = []
        for t in range(T):
            h, c = cell._recurrent_step(igates_all[t], h, c)
            outputs.append(h)
        return torch.stack(outputs), (h, c)


class StackedLSTM(nn.Module):
    def __init__(self, num_layers, layer, first_layer_args, other_layer_args):
        super().__init__()
        self.layers = nn.ModuleList(
            [layer(*first_layer_args)] + [layer(*other_layer_args) for _ in range(num_layers - 1)]
        )

    def forward(self, input: Tensor, states: Optional[List[Tuple[Tensor, Tensor]]]
                ) -> Tuple[Tensor, List[Tuple[Tensor, Tensor]]]:
        if states is None:
            B = input.shape[1]
            H = self.layers[0].cell.hidden_size
            zeros = torch.zeros(B, H, dtype=input.dtype, device=input.device)
            states = [(zeros, zeros) for _ in self.layers]
        output_states: List[Tuple[Tensor, Tensor]] = []
        output = input
        for i, layer in enumerate(self.layers):
            output, out_state = layer(output, states[i])
            output_states.append(out_state)
        return output, output_states


def script_lnlstm(input_size, hidden_size, num_layers, bias=True,
                  batch_first=False, dropout=False, bidirectional=False,
                  decompose_layernorm=False):
    """Factory mirroring the reference API (`lstm.py:36-59`)."""
    assert bias and not batch_first and not dropout and not bidirectional
    assert not decompose_layernorm
    return StackedLSTM(num_layers, LSTMLayer,
                       first_layer_args=[LayerNormLSTMCell, input_size, hidden_size],
                       other_layer_args=[LayerNormLSTMCell, hidden_size, hidden_size])
