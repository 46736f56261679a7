"""Auxiliary network modules for component parity (reference
`ctools/torch_utils/network/{soft_argmax,upsample}.py`)."""
import torch
import torch.nn as nn
import torch.nn.functional as F


class SoftArgmax(nn.Module):
    """Differentiable 2D argmax over a (B, 1, H, W) heatmap -> (B, 2) coords
    (reference soft_argmax.py:12-60)."""

    def forward(self, x):
        assert x.dim() == 4 and x.shape[1] == 1
        B, _, H, W = x.shape
        device = x.device
        h_kernel = torch.arange(0, H, device=device, dtype=x.dtype).view(1, 1, H, 1)
        w_kernel = torch.arange(0, W, device=device, dtype=x.dtype).view(1, 1, 1, W)
        probs = F.softmax(x.view(B, -1), dim=1).view(B, 1, H, W)
        h = (probs * h_kernel).sum(dim=(1, 2, 3))
        w = (probs * w_kernel).sum(dim=(1, 2, 3))
        return torch.stack([h, w], dim=1)


class NearestUpsample(nn.Module):
    def __init__(self, scale_factor=2.):
        super().__init__()
        self.scale_factor = scale_factor

    def forward(self, x):
        return F.interpolate(x, scale_factor=self.scale_factor, mode='nearest')


class BilinearUpsample(nn.Module):
    def __init__(self, scale_factor=2.):
        super().__init__()
        self.scale_factor = scale_factor

    def forward(self, x):
        from ...ops.upsample import upsample2x_bilinear
        if self.scale_factor == 2.:
            return upsample2x_bilinear(x)
        return F.interpolate(x, scale_factor=self.scale_factor, mode='bilinear')
