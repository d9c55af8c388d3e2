"""EfficientNet-family generator: block-string decoder, channel/depth
scaling, block modules and the EfficientNet trunk.

Capability parity with reference models/efficientnet.py:53-352 / the timm
copy (timm/models/efficientnet.py): the Wightman block-definition strings
('ir_r2_k3_s2_e6_c24_se0.25', 'ds_...', 'er_...', 'cn_...') decode into
stages; depth_multiplier scales repeats (ceil), channel_multiplier rounds
channels to multiples of 8; blocks are ConvBnAct, DepthwiseSeparableConv,
InvertedResidual (with optional CondConv routing), EdgeResidual, each with
optional SqueezeExcite and drop_connect.
"""

import math
import re
from copy import deepcopy

import torch
import torch.nn as nn

from .. import ops
from .activations import HardSigmoid, HardSwish, Sigmoid, Swish
from .adaptive_avgmax_pool import SelectAdaptivePool2d
from .conv2d_layers import (CondConv2d, NativeBatchNorm2d, NativeConv2d,
                            select_conv2d)


def make_divisible(v, divisor=8, min_value=None):
    min_value = min_value or divisor
    new_v = max(min_value, int(v + divisor / 2) // divisor * divisor)
    if new_v < 0.9 * v:
        new_v += divisor
    return new_v


def round_channels(channels, multiplier=1.0, divisor=8, channel_min=None):
    if not multiplier:
        return channels
    return make_divisible(channels * multiplier, divisor, channel_min)


def drop_connect(inputs, training=False, drop_connect_rate=0.):
    """Stochastic depth (reference models/efficientnet.py:436-446)."""
    if not training:
        return inputs
    keep_prob = 1 - drop_connect_rate
    random_tensor = keep_prob + torch.rand(
        (inputs.size()[0], 1, 1, 1), dtype=inputs.dtype, device=inputs.device)
    random_tensor.floor_()
    return inputs.div(keep_prob) * random_tensor


_ACTS = {
    're': nn.ReLU, 'r6': nn.ReLU6, 'hs': HardSwish, 'sw': Swish,
}


def _decode_block_str(block_str):
    """One block string -> (block_args, num_repeat)."""
    assert isinstance(block_str, str)
    parts = block_str.split('_')
    block_type = parts[0]
    options = {}
    noskip = False
    for op in parts[1:]:
        if op == 'noskip':
            noskip = True
        elif op.startswith('n'):
            # activation: nre, nr6, nhs, nsw
            options['act'] = op[1:]
        else:
            splits = re.split(r'(\d.*)', op)
            if len(splits) >= 2:
                options[splits[0]] = splits[1]

    act_layer = _ACTS.get(options.get('act'), None)
    num_repeat = int(options.get('r', 1))
    kernel = options.get('k', '3')
    if '.' in kernel:
        kernel_size = [int(k) for k in kernel.split('.')]
    else:
        kernel_size = int(kernel)

    ba = dict(
        block_type=block_type,
        dw_kernel_size=kernel_size,
        stride=int(options.get('s', 1)),
        out_chs=int(options.get('c', 16)),
        act_layer=act_layer,
        noskip=noskip,
    )
    if 'e' in options:
        ba['exp_ratio'] = float(options['e'])
    if 'se' in options:
        ba['se_ratio'] = float(options['se'])
    if 'cc' in options:
        ba['num_experts'] = int(options['cc'])
    if 'd' in options:
        ba['dilation'] = int(options['d'])
    return ba, num_repeat


def _scale_stage_depth(stack_args, repeats, depth_multiplier=1.0,
                       depth_trunc='ceil'):
    num_repeat = sum(repeats)
    if depth_trunc == 'round':
        num_repeat_scaled = max(1, round(num_repeat * depth_multiplier))
    else:
        num_repeat_scaled = int(math.ceil(num_repeat * depth_multiplier))
    repeats_scaled = []
    for r in repeats[::-1]:
        rs = max(1, round((r / num_repeat * num_repeat_scaled)))
        repeats_scaled.append(rs)
        num_repeat -= r
        num_repeat_scaled -= rs
    repeats_scaled = repeats_scaled[::-1]
    sa_scaled = []
    for ba, rep in zip(stack_args, repeats_scaled):
        sa_scaled.extend([deepcopy(ba) for _ in range(rep)])
    return sa_scaled


def decode_arch_def(arch_def, depth_multiplier=1.0, depth_trunc='ceil'):
    arch_args = []
    for stack_strings in arch_def:
        assert isinstance(stack_strings, list)
        stack_args = []
        repeats = []
        for block_str in stack_strings:
            ba, rep = _decode_block_str(block_str)
            stack_args.append(ba)
            repeats.append(rep)
        arch_args.append(_scale_stage_depth(stack_args, repeats,
                                            depth_multiplier, depth_trunc))
    return arch_args


class SqueezeExcite(nn.Module):
    def __init__(self, in_chs, se_ratio=0.25, reduced_base_chs=None,
                 act_layer=nn.ReLU, gate_fn=None, divisor=1):
        super().__init__()
        self.gate_fn = gate_fn or Sigmoid()
        reduced_chs = make_divisible((reduced_base_chs or in_chs) * se_ratio,
                                     divisor)
        self.avg_pool = nn.AdaptiveAvgPool2d(1)
        self.conv_reduce = NativeConv2d(in_chs, reduced_chs, 1, bias=True)
        self.act1 = act_layer(inplace=True)
        self.conv_expand = NativeConv2d(reduced_chs, in_chs, 1, bias=True)

    def forward(self, x):
        x_se = self.avg_pool(x)
        x_se = self.conv_reduce(x_se)
        x_se = self.act1(x_se)
        x_se = self.conv_expand(x_se)
        if x.is_cuda:
            # fused gate*broadcast-scale (csrc/activations.hip se_scale)
            gate = ('hard_sigmoid'
                    if isinstance(self.gate_fn, HardSigmoid) else 'sigmoid')
            return ops.se_scale(x, x_se, gate)
        return x * self.gate_fn(x_se)


class ConvBnAct(nn.Module):
    def __init__(self, in_chs, out_chs, kernel_size, stride=1, dilation=1,
                 pad_type='', act_layer=nn.ReLU, norm_kwargs=None):
        super().__init__()
        norm_kwargs = norm_kwargs or {}
        self.conv = select_conv2d(in_chs, out_chs, kernel_size, stride=stride,
                                  dilation=dilation, padding=pad_type)
        self.bn1 = NativeBatchNorm2d(out_chs, **norm_kwargs)
        self.act1 = act_layer(inplace=True)

    def forward(self, x):
        return self.act1(self.bn1(self.conv(x)))


class DepthwiseSeparableConv(nn.Module):
    def __init__(self, in_chs, out_chs, dw_kernel_size=3, stride=1, dilation=1,
                 pad_type='', act_layer=nn.ReLU, noskip=False,
                 pw_kernel_size=1, pw_act=False, se_ratio=0.,
                 norm_kwargs=None, drop_connect_rate=0.):
        super().__init__()
        norm_kwargs = norm_kwargs or {}
        self.has_se = se_ratio is not None and se_ratio > 0.
        self.has_residual = (stride == 1 and in_chs == out_chs) and not noskip
        self.has_pw_act = pw_act
        self.drop_connect_rate = drop_connect_rate

        self.conv_dw = select_conv2d(in_chs, in_chs, dw_kernel_size,
                                     stride=stride, dilation=dilation,
                                     padding=pad_type, depthwise=True)
        self.bn1 = NativeBatchNorm2d(in_chs, **norm_kwargs)
        self.act1 = act_layer(inplace=True)
        if self.has_se:
            self.se = SqueezeExcite(in_chs, se_ratio=se_ratio,
                                    act_layer=act_layer)
        self.conv_pw = select_conv2d(in_chs, out_chs, pw_kernel_size,
                                     padding=pad_type)
        self.bn2 = NativeBatchNorm2d(out_chs, **norm_kwargs)
        self.act2 = act_layer(inplace=True) if self.has_pw_act else nn.Identity()

    def forward(self, x):
        residual = x
        x = self.act1(self.bn1(self.conv_dw(x)))
        if self.has_se:
            x = self.se(x)
        x = self.act2(self.bn2(self.conv_pw(x)))
        if self.has_residual:
            if self.drop_connect_rate > 0.:
                x = drop_connect(x, self.training, self.drop_connect_rate)
            x = x + residual
        return x


class InvertedResidual(nn.Module):
    """IR block with optional SE and CondConv routing
    (reference models/efficientnet.py:535-598)."""

    def __init__(self, in_chs, out_chs, dw_kernel_size=3, stride=1, dilation=1,
                 pad_type='', act_layer=nn.ReLU, noskip=False, exp_ratio=1.0,
                 exp_kernel_size=1, pw_kernel_size=1, se_ratio=0.,
                 norm_kwargs=None, conv_kwargs=None, drop_connect_rate=0.,
                 num_experts=0):
        super().__init__()
        norm_kwargs = norm_kwargs or {}
        conv_kwargs = conv_kwargs or {}
        mid_chs = make_divisible(in_chs * exp_ratio)
        self.has_se = se_ratio is not None and se_ratio > 0.
        self.has_residual = (in_chs == out_chs and stride == 1) and not noskip
        self.drop_connect_rate = drop_connect_rate
        self.num_experts = num_experts
        if num_experts > 0:
            conv_kwargs = dict(conv_kwargs, num_experts=num_experts)
            self.routing_fn = nn.Linear(in_chs, num_experts)

        self.conv_pw = select_conv2d(in_chs, mid_chs, exp_kernel_size,
                                     padding=pad_type, **conv_kwargs)
        self.bn1 = NativeBatchNorm2d(mid_chs, **norm_kwargs)
        self.act1 = act_layer(inplace=True)
        self.conv_dw = select_conv2d(mid_chs, mid_chs, dw_kernel_size,
                                     stride=stride, dilation=dilation,
                                     padding=pad_type, depthwise=True,
                                     **conv_kwargs)
        self.bn2 = NativeBatchNorm2d(mid_chs, **norm_kwargs)
        self.act2 = act_layer(inplace=True)
        if self.has_se:
            self.se = SqueezeExcite(mid_chs, se_ratio=se_ratio,
                                    reduced_base_chs=in_chs,
                                    act_layer=act_layer)
        self.conv_pwl = select_conv2d(mid_chs, out_chs, pw_kernel_size,
                                      padding=pad_type, **conv_kwargs)
        self.bn3 = NativeBatchNorm2d(out_chs, **norm_kwargs)

    def forward(self, x):
        residual = x
        if self.num_experts > 0:
            pooled = x.mean(dim=(2, 3))
            routing = torch.sigmoid(self.routing_fn(pooled))
            x = self.conv_pw(x, routing)
            x = self.act1(self.bn1(x))
            x = self.conv_dw(x, routing) if isinstance(self.conv_dw, CondConv2d) \
                else self.conv_dw(x)
            x = self.act2(self.bn2(x))
            if self.has_se:
                x = self.se(x)
            x = self.conv_pwl(x, routing) if isinstance(self.conv_pwl, CondConv2d) \
                else self.conv_pwl(x)
            x = self.bn3(x)
        else:
            x = self.act1(self.bn1(self.conv_pw(x)))
            x = self.act2(self.bn2(self.conv_dw(x)))
            if self.has_se:
                x = self.se(x)
            x = self.bn3(self.conv_pwl(x))
        if self.has_residual:
            if self.drop_connect_rate > 0.:
                x = drop_connect(x, self.training, self.drop_connect_rate)
            x = x + residual
        return x


class EdgeResidual(nn.Module):
    """EdgeTPU residual: expansion conv is a full (not dw) conv
    (reference models/efficientnet.py:601-653)."""

    def __init__(self, in_chs, out_chs, exp_kernel_size=3, exp_ratio=1.0,
                 fake_in_chs=0, stride=1, dilation=1, pad_type='',
                 act_layer=nn.ReLU, noskip=False, pw_kernel_size=1,
                 se_ratio=0., norm_kwargs=None, drop_connect_rate=0.,
                 dw_kernel_size=None):
        super().__init__()
        norm_kwargs = norm_kwargs or {}
        if dw_kernel_size is not None:  # decoder passes k as dw_kernel_size
            exp_kernel_size = dw_kernel_size
        mid_chs = make_divisible((fake_in_chs or in_chs) * exp_ratio)
        self.has_se = se_ratio is not None and se_ratio > 0.
        self.has_residual = (in_chs == out_chs and stride == 1) and not noskip
        self.drop_connect_rate = drop_connect_rate

        self.conv_exp = select_conv2d(in_chs, mid_chs, exp_kernel_size,
                                      padding=pad_type)
        self.bn1 = NativeBatchNorm2d(mid_chs, **norm_kwargs)
        self.act1 = act_layer(inplace=True)
        if self.has_se:
            self.se = SqueezeExcite(mid_chs, se_ratio=se_ratio,
                                    reduced_base_chs=in_chs,
                                    act_layer=act_layer)
        self.conv_pwl = select_conv2d(mid_chs, out_chs, pw_kernel_size,
                                      stride=stride, padding=pad_type)
        self.bn2 = NativeBatchNorm2d(out_chs, **norm_kwargs)

    def forward(self, x):
        residual = x
        x = self.act1(self.bn1(self.conv_exp(x)))
        if self.has_se:
            x = self.se(x)
        x = self.bn2(self.conv_pwl(x))
        if self.has_residual:
            if self.drop_connect_rate > 0.:
                x = drop_connect(x, self.training, self.drop_connect_rate)
            x = x + residual
        return x


class EfficientNetBuilder:
    """Turns decoded arch args into nn.Sequential stages
    (reference models/efficientnet.py:223-352)."""

    def __init__(self, channel_multiplier=1.0, channel_divisor=8,
                 channel_min=None, pad_type='', act_layer=nn.ReLU,
                 norm_kwargs=None, drop_connect_rate=0.):
        self.channel_multiplier = channel_multiplier
        self.channel_divisor = channel_divisor
        self.channel_min = channel_min
        self.pad_type = pad_type
        self.act_layer = act_layer
        self.norm_kwargs = norm_kwargs
        self.drop_connect_rate = drop_connect_rate
        self.in_chs = None

    def _round_channels(self, chs):
        return round_channels(chs, self.channel_multiplier,
                              self.channel_divisor, self.channel_min)

    def _make_block(self, ba, block_idx, block_count):
        bt = ba.pop('block_type')
        ba['out_chs'] = out_chs = self._round_channels(ba['out_chs'])
        act = ba.pop('act_layer', None) or self.act_layer
        drop_rate = self.drop_connect_rate * block_idx / block_count
        common = dict(pad_type=self.pad_type, act_layer=act,
                      norm_kwargs=self.norm_kwargs)
        in_chs = self.in_chs
        if bt == 'ir':
            block = InvertedResidual(in_chs, drop_connect_rate=drop_rate,
                                     **common, **ba)
        elif bt == 'ds' or bt == 'dsa':
            ba.pop('exp_ratio', None)
            block = DepthwiseSeparableConv(in_chs, pw_act=(bt == 'dsa'),
                                           drop_connect_rate=drop_rate,
                                           **common, **ba)
        elif bt == 'er':
            block = EdgeResidual(in_chs, drop_connect_rate=drop_rate,
                                 **common, **ba)
        elif bt == 'cn':
            ba.pop('noskip', None)
            block = ConvBnAct(in_chs, ba.pop('out_chs'),
                              kernel_size=ba.pop('dw_kernel_size'),
                              stride=ba.pop('stride'),
                              pad_type=self.pad_type, act_layer=act,
                              norm_kwargs=self.norm_kwargs)
        else:
            raise ValueError('unknown block type %s' % bt)
        self.in_chs = out_chs
        return block

    def __call__(self, in_chs, block_args):
        self.in_chs = in_chs
        total = sum(len(stage) for stage in block_args)
        stages = []
        idx = 0
        for stage_args in block_args:
            blocks = []
            for ba in stage_args:
                blocks.append(self._make_block(dict(ba), idx, total))
                idx += 1
            stages.append(nn.Sequential(*blocks))
        return stages


class EfficientNet(nn.Module):
    """Generic EfficientNet trunk: stem conv -> stages -> head conv -> pool
    -> classifier, with the reference's trailing bn_out BatchNorm1d option
    (models/efficientnet.py:691,708)."""

    def __init__(self, block_args, num_classes=1000, in_chans=3,
                 stem_size=32, num_features=1280, channel_multiplier=1.0,
                 channel_divisor=8, channel_min=None, pad_type='',
                 act_layer=Swish, drop_rate=0., drop_connect_rate=0.,
                 norm_kwargs=None, global_pool='avg', bn_out=False):
        super().__init__()
        norm_kwargs = norm_kwargs or {}
        self.num_classes = num_classes
        self.num_features = num_features
        self.drop_rate = drop_rate

        stem_size = round_channels(stem_size, channel_multiplier,
                                   channel_divisor, channel_min)
        self.conv_stem = select_conv2d(in_chans, stem_size, 3, stride=2,
                                       padding=pad_type)
        self.bn1 = NativeBatchNorm2d(stem_size, **norm_kwargs)
        self.act1 = act_layer(inplace=True)

        builder = EfficientNetBuilder(channel_multiplier, channel_divisor,
                                      channel_min, pad_type, act_layer,
                                      norm_kwargs, drop_connect_rate)
        self.blocks = nn.Sequential(*builder(stem_size, block_args))
        head_chs = builder.in_chs

        self.conv_head = select_conv2d(head_chs, num_features, 1,
                                       padding=pad_type)
        self.bn2 = NativeBatchNorm2d(num_features, **norm_kwargs)
        self.act2 = act_layer(inplace=True)
        self.global_pool = SelectAdaptivePool2d(pool_type=global_pool)
        self.classifier = nn.Linear(num_features * self.global_pool.feat_mult(),
                                    num_classes)
        self.bn_out = nn.BatchNorm1d(num_classes) if bn_out else None

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                # "goog" init: depthwise convs divide fan_out by groups
                fan_out = m.kernel_size[0] * m.kernel_size[1] * m.out_channels
                fan_out //= m.groups
                m.weight.data.normal_(0, math.sqrt(2.0 / fan_out))
                if m.bias is not None:
                    m.bias.data.zero_()
            elif isinstance(m, nn.BatchNorm2d):
                m.weight.data.fill_(1.0)
                m.bias.data.zero_()
            elif isinstance(m, nn.Linear):
                fan_out = m.weight.size(0)
                init_range = 1.0 / math.sqrt(fan_out)
                m.weight.data.uniform_(-init_range, init_range)
                if m.bias is not None:
                    m.bias.data.zero_()

    def features(self, x):
        x = self.act1(self.bn1(self.conv_stem(x)))
        x = self.blocks(x)
        x = self.act2(self.bn2(self.conv_head(x)))
        return x

    def forward(self, x, epoch=0, i=0, acc=0.0):
        x = self.features(x)
        x = self.global_pool(x)
        x = x.flatten(1)
        if self.drop_rate > 0.:
            x = ops.dropout(x, self.drop_rate, self.training)
        x = ops.linear(x, self.classifier.weight, self.classifier.bias)
        if self.bn_out is not None:
            x = self.bn_out(x)
        return x

    def as_sequential(self):
        layers = [self.conv_stem, self.bn1, self.act1]
        layers.extend(self.blocks)
        layers.extend([self.conv_head, self.bn2, self.act2,
                       self.global_pool, nn.Flatten(), self.classifier])
        return nn.Sequential(*layers)
