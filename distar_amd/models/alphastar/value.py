"""Per-baseline value networks (reference `model/value.py`)."""
import math

import torch
import torch.nn as nn

from ..nn.blocks import fc_block, fc_block2, build_activation, ResFCBlock2


class ValueBaseline(nn.Module):
    """fc -> res_num x ResFCBlock2 -> fc(1) [-> atan squash]."""

    def __init__(self, cfg, use_value_feature=False):
        super().__init__()
        self.act = build_activation(cfg.activation)
        input_dim = cfg.input_dim + 1056 if use_value_feature else cfg.input_dim
        self.project = fc_block(input_dim, cfg.res_dim, activation=self.act, norm_type=None)
        self.res = nn.Sequential(*[ResFCBlock2(cfg.res_dim, self.act, cfg.norm_type)
                                   for _ in range(cfg.res_num)])
        self.value_fc = fc_block2(cfg.res_dim, 1, activation=None, norm_type=None, gain=0.1)
        self.atan = cfg.atan
        self.PI = math.pi

    def forward(self, x):
        x = self.project(x)
        x = self.res(x)
        x = self.value_fc(x).squeeze(1)
        if self.atan:
            x = (2.0 / self.PI) * torch.atan((self.PI / 2.0) * x)
        return x
