"""Selectable global pooling (reference models/adaptive_avgmax_pool.py)."""

import torch
import torch.nn as nn
import torch.nn.functional as F


def adaptive_avgmax_pool2d(x, output_size=1):
    x_avg = F.adaptive_avg_pool2d(x, output_size)
    x_max = F.adaptive_max_pool2d(x, output_size)
    return 0.5 * (x_avg + x_max)


def adaptive_catavgmax_pool2d(x, output_size=1):
    x_avg = F.adaptive_avg_pool2d(x, output_size)
    x_max = F.adaptive_max_pool2d(x, output_size)
    return torch.cat((x_avg, x_max), 1)


def select_adaptive_pool2d(x, pool_type='avg', output_size=1):
    if pool_type == 'avg':
        return F.adaptive_avg_pool2d(x, output_size)
    if pool_type == 'avgmax':
        return adaptive_avgmax_pool2d(x, output_size)
    if pool_type == 'catavgmax':
        return adaptive_catavgmax_pool2d(x, output_size)
    if pool_type == 'max':
        return F.adaptive_max_pool2d(x, output_size)
    raise ValueError('Invalid pool type: %s' % pool_type)


def adaptive_pool_feat_mult(pool_type='avg'):
    return 2 if pool_type == 'catavgmax' else 1


class SelectAdaptivePool2d(nn.Module):
    def __init__(self, output_size=1, pool_type='avg'):
        super().__init__()
        self.output_size = output_size
        self.pool_type = pool_type

    def feat_mult(self):
        return adaptive_pool_feat_mult(self.pool_type)

    def forward(self, x):
        return select_adaptive_pool2d(x, self.pool_type, self.output_size)
