"""Conv padding/variant layers (reference models/conv2d_layers.py, byte-
identical to the timm copy there): TF-"SAME" padding, MixedConv2d (per-group
kernel sizes), CondConv2d (per-sample expert-mixed weights via grouped conv)
and the select_conv2d dispatch."""

import math
from functools import partial

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops


def _is_static_pad(kernel_size, stride=1, dilation=1, **_):
    return stride == 1 and (dilation * (kernel_size - 1)) % 2 == 0


def _get_padding(kernel_size, stride=1, dilation=1, **_):
    return ((stride - 1) + dilation * (kernel_size - 1)) // 2


def _calc_same_pad(i, k, s, d):
    return max((math.ceil(i / s) - 1) * s + (k - 1) * d + 1 - i, 0)


def conv2d_same(x, weight, bias=None, stride=(1, 1), padding=(0, 0),
                dilation=(1, 1), groups=1):
    ih, iw = x.size()[-2:]
    kh, kw = weight.size()[-2:]
    pad_h = _calc_same_pad(ih, kh, stride[0], dilation[0])
    pad_w = _calc_same_pad(iw, kw, stride[1], dilation[1])
    if pad_h > 0 or pad_w > 0:
        x = F.pad(x, [pad_w // 2, pad_w - pad_w // 2,
                      pad_h // 2, pad_h - pad_h // 2])
    if groups == 1 and dilation == (1, 1) and pad_h % 2 == 0 and pad_w % 2 == 0:
        return ops.conv2d(x, weight, bias, stride[0], 0)
    # documented exception: dilated / grouped-non-depthwise SAME convs are
    # outside the model zoo's hot paths (no registered model hits this)
    return F.conv2d(x, weight, bias, stride, (0, 0), dilation, groups)


class Conv2dSame(nn.Conv2d):
    """Tensorflow-like 'SAME' convolution wrapper."""

    def __init__(self, in_channels, out_channels, kernel_size, stride=1,
                 padding=0, dilation=1, groups=1, bias=True):
        super().__init__(in_channels, out_channels, kernel_size, stride, 0,
                         dilation, groups, bias)

    def forward(self, x):
        return conv2d_same(x, self.weight, self.bias, self.stride,
                           self.padding, self.dilation, self.groups)


def get_padding_value(padding, kernel_size, **kwargs):
    dynamic = False
    if isinstance(padding, str):
        padding = padding.lower()
        if padding == 'same':
            if _is_static_pad(kernel_size, **kwargs):
                padding = _get_padding(kernel_size, **kwargs)
            else:
                padding = 0
                dynamic = True
        elif padding == 'valid':
            padding = 0
        else:
            padding = _get_padding(kernel_size, **kwargs)
    return padding, dynamic


class NativeBatchNorm2d(nn.BatchNorm2d):
    """nn.BatchNorm2d routed through the fused BN HIP kernel
    (csrc/bn_act.hip: one stats pass + one finalize kernel instead of
    MIOpen's 5-kernel chain). State-dict identical to nn.BatchNorm2d."""

    def forward(self, x):
        if self.weight is None:  # affine=False: rare, keep torch
            return super().forward(x)
        if self.training and self.track_running_stats \
                and self.num_batches_tracked is not None:
            self.num_batches_tracked.add_(1)
        mom = self.momentum if self.momentum is not None else 0.1
        return ops.bn_act(x, self.weight, self.bias, self.running_mean,
                          self.running_var,
                          self.training or not self.track_running_stats,
                          mom, self.eps, relu=False, act_max=0.0)


class NativeConv2d(nn.Conv2d):
    """nn.Conv2d whose forward dispatches to the MFMA / depthwise HIP
    kernels for the standard cases (state_dict-identical to nn.Conv2d)."""

    def forward(self, x):
        if self.groups == 1 and self.dilation == (1, 1):
            return ops.conv2d(x, self.weight, self.bias, self.stride,
                              self.padding)
        if (self.groups == self.in_channels == self.out_channels
                and self.dilation == (1, 1)):
            return ops.depthwise_conv2d(x, self.weight, self.bias,
                                        self.stride, self.padding)
        # documented exception: dilation / grouped-non-depthwise -- not on
        # any registered model's hot path
        return F.conv2d(x, self.weight, self.bias, self.stride, self.padding,
                        self.dilation, self.groups)


def create_conv2d_pad(in_chs, out_chs, kernel_size, **kwargs):
    padding = kwargs.pop('padding', '')
    kwargs.setdefault('bias', False)
    padding, is_dynamic = get_padding_value(padding, kernel_size, **kwargs)
    if is_dynamic:
        return Conv2dSame(in_chs, out_chs, kernel_size, **kwargs)
    return NativeConv2d(in_chs, out_chs, kernel_size, padding=padding, **kwargs)


def _split_channels(num_chan, num_groups):
    split = [num_chan // num_groups for _ in range(num_groups)]
    split[0] += num_chan - sum(split)
    return split


class MixedConv2d(nn.ModuleDict):
    """Mixed grouped convolution: per-group kernel sizes (MixNet)."""

    def __init__(self, in_channels, out_channels, kernel_size=3, stride=1,
                 padding='', dilation=1, depthwise=False, **kwargs):
        super().__init__()
        kernel_size = kernel_size if isinstance(kernel_size, list) else [kernel_size]
        num_groups = len(kernel_size)
        in_splits = _split_channels(in_channels, num_groups)
        out_splits = _split_channels(out_channels, num_groups)
        self.in_channels = sum(in_splits)
        self.out_channels = sum(out_splits)
        for idx, (k, in_ch, out_ch) in enumerate(zip(kernel_size, in_splits, out_splits)):
            conv_groups = out_ch if depthwise else 1
            self.add_module(str(idx), create_conv2d_pad(
                in_ch, out_ch, k, stride=stride, padding=padding,
                dilation=dilation, groups=conv_groups, **kwargs))
        self.splits = in_splits

    def forward(self, x):
        x_split = torch.split(x, self.splits, 1)
        return torch.cat([c(x_split[i]) for i, c in enumerate(self.values())], 1)


def get_condconv_initializer(initializer, num_experts, expert_shape):
    def condconv_initializer(weight):
        num_params = 1
        for d in expert_shape:
            num_params *= d
        if (len(weight.shape) != 2 or weight.shape[0] != num_experts
                or weight.shape[1] != num_params):
            raise ValueError('CondConv variables must have shape [num_experts, num_params]')
        for i in range(num_experts):
            initializer(weight[i].view(expert_shape))
    return condconv_initializer


class CondConv2d(nn.Module):
    """Conditionally-parameterized convolution: per-sample expert mixing
    followed by a grouped conv with batch*groups groups (reference
    conv2d_layers.py:152-240)."""

    __constants__ = ['bias', 'in_channels', 'out_channels', 'dynamic_padding']

    def __init__(self, in_channels, out_channels, kernel_size=3, stride=1,
                 padding='', dilation=1, groups=1, bias=False, num_experts=4):
        super().__init__()
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.kernel_size = (kernel_size, kernel_size)
        self.stride = (stride, stride)
        padding_val, is_padding_dynamic = get_padding_value(
            padding, kernel_size, stride=stride, dilation=dilation)
        self.dynamic_padding = is_padding_dynamic
        self.padding = (padding_val, padding_val) if not isinstance(padding_val, tuple) else padding_val
        self.dilation = (dilation, dilation)
        self.groups = groups
        self.num_experts = num_experts

        self.weight_shape = (self.out_channels, self.in_channels // self.groups) + self.kernel_size
        weight_num_param = 1
        for wd in self.weight_shape:
            weight_num_param *= wd
        self.weight = nn.Parameter(torch.Tensor(self.num_experts, weight_num_param))
        if bias:
            self.bias_shape = (self.out_channels,)
            self.bias = nn.Parameter(torch.Tensor(self.num_experts, self.out_channels))
        else:
            self.register_parameter('bias', None)
        self.reset_parameters()

    def reset_parameters(self):
        init_weight = get_condconv_initializer(
            partial(nn.init.kaiming_uniform_, a=math.sqrt(5)),
            self.num_experts, self.weight_shape)
        init_weight(self.weight)
        if self.bias is not None:
            fan_in = self.weight_shape[1] * self.weight_shape[2] * self.weight_shape[3]
            bound = 1 / math.sqrt(fan_in)
            init_bias = get_condconv_initializer(
                partial(nn.init.uniform_, a=-bound, b=bound),
                self.num_experts, self.bias_shape)
            init_bias(self.bias)

    def forward(self, x, routing_weights):
        B, C, H, W = x.shape
        # expert mixing GEMM on the MFMA kernel
        weight = ops.linear(routing_weights, self.weight.t().contiguous()) \
            if x.is_cuda else torch.matmul(routing_weights, self.weight)
        new_weight_shape = (B * self.out_channels,
                            self.in_channels // self.groups) + self.kernel_size
        weight = weight.view(new_weight_shape)
        bias = None
        if self.bias is not None:
            bias = torch.matmul(routing_weights, self.bias).view(B * self.out_channels)
        # Per-sample conv via unfold + one batched GEMM (K13): the
        # reference's batch*groups grouped conv (conv2d_layers.py:204-221)
        # is replaced by ops.per_sample_conv2d -- no F.conv2d on this path.
        if self.dilation == (1, 1):
            xs = x
            padding = self.padding
            if self.dynamic_padding:
                ih, iw = x.size()[-2:]
                pad_h = _calc_same_pad(ih, self.kernel_size[0], self.stride[0], 1)
                pad_w = _calc_same_pad(iw, self.kernel_size[1], self.stride[1], 1)
                xs = F.pad(x, (pad_w // 2, pad_w - pad_w // 2,
                               pad_h // 2, pad_h - pad_h // 2))
                padding = (0, 0)
            per_bias = None
            if bias is not None:
                per_bias = bias.view(B, self.out_channels)
            return ops.per_sample_conv2d(xs, weight, per_bias,
                                         stride=self.stride, padding=padding,
                                         groups=self.groups)
        # dilated CondConv (unused by the model zoo): grouped-conv route
        x = x.view(1, B * C, H, W)
        if self.dynamic_padding:
            out = conv2d_same(x, weight, bias, stride=self.stride,
                              padding=self.padding, dilation=self.dilation,
                              groups=self.groups * B)
        else:
            out = F.conv2d(x, weight, bias, stride=self.stride,
                           padding=self.padding, dilation=self.dilation,
                           groups=self.groups * B)
        return out.permute([1, 0, 2, 3]).view(
            B, self.out_channels, out.shape[-2], out.shape[-1])


def select_conv2d(in_chs, out_chs, kernel_size, **kwargs):
    assert 'groups' not in kwargs
    if isinstance(kernel_size, list):
        assert 'num_experts' not in kwargs
        m = MixedConv2d(in_chs, out_chs, kernel_size, **kwargs)
    else:
        depthwise = kwargs.pop('depthwise', False)
        groups = out_chs if depthwise else 1
        if 'num_experts' in kwargs and kwargs['num_experts'] > 0:
            m = CondConv2d(in_chs, out_chs, kernel_size, groups=groups, **kwargs)
        else:
            m = create_conv2d_pad(in_chs, out_chs, kernel_size, groups=groups,
                                  **kwargs)
    return m
