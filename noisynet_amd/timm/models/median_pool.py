"""Unfold-based median pooling (reference timm/models/median_pool.py:48)."""

import torch.nn as nn
import torch.nn.functional as F


class MedianPool2d(nn.Module):
    """Median pool (usable as median filter when stride=1).

    Args:
        kernel_size: size of pooling kernel, int or 2-tuple
        stride: pool stride, int or 2-tuple
        padding: pool padding, int or 4-tuple (l, r, t, b) as in F.pad
        same: override padding and enforce same padding
    """

    def __init__(self, kernel_size=3, stride=1, padding=0, same=False):
        super().__init__()
        self.k = (kernel_size, kernel_size) if isinstance(kernel_size, int) \
            else tuple(kernel_size)
        self.stride = (stride, stride) if isinstance(stride, int) \
            else tuple(stride)
        self.padding = (padding,) * 4 if isinstance(padding, int) \
            else tuple(padding)
        self.same = same

    def _padding(self, x):
        if self.same:
            ih, iw = x.size()[2:]
            if ih % self.stride[0] == 0:
                ph = max(self.k[0] - self.stride[0], 0)
            else:
                ph = max(self.k[0] - (ih % self.stride[0]), 0)
            if iw % self.stride[1] == 0:
                pw = max(self.k[1] - self.stride[1], 0)
            else:
                pw = max(self.k[1] - (iw % self.stride[1]), 0)
            return (pw // 2, pw - pw // 2, ph // 2, ph - ph // 2)
        return self.padding

    def forward(self, x):
        x = F.pad(x, self._padding(x), mode='reflect')
        x = x.unfold(2, self.k[0], self.stride[0]) \
             .unfold(3, self.k[1], self.stride[1])
        x = x.contiguous().view(x.size()[:4] + (-1,)).median(dim=-1)[0]
        return x
