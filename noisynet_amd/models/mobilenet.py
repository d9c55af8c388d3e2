"""Quantization-aware MobileNetV2, rebuilt from reference models/mobilenet.py.

Same module/state_dict layout (features.N.{conv,bn} / ConvBNReLU owns a
QuantMeasure, InvertedResidual has conv1/conv2/conv3+bn with quantize1-3,
final fc1 + optional bn_out) so reference checkpoints and the
merge_batchnorm key-walk work. Depthwise 3x3 layers run on the dedicated
NHWC depthwise kernel; 1x1 expand/project convs on the MFMA implicit-GEMM
kernel; ReLU6 is the fused relu_clip(6).
"""

import torch
import torch.nn as nn

from .. import ops
from ..quant import QuantMeasure


def _make_divisible(v, divisor, min_value=None):
    if min_value is None:
        min_value = divisor
    new_v = max(min_value, int(v + divisor / 2) // divisor * divisor)
    if new_v < 0.9 * v:
        new_v += divisor
    return new_v


def _conv(x, conv):
    if conv.groups == 1:
        return ops.conv2d(x, conv.weight, conv.bias, conv.stride, conv.padding)
    if conv.groups == conv.in_channels == conv.out_channels:
        return ops.depthwise_conv2d(x, conv.weight, conv.bias, conv.stride,
                                    conv.padding)
    return nn.functional.conv2d(x, conv.weight, conv.bias, conv.stride,
                                conv.padding, conv.dilation, conv.groups)


class ConvBNReLU(nn.Module):
    def __init__(self, args, in_planes, out_planes, kernel_size=3, stride=1,
                 groups=1):
        super().__init__()
        self.args = args
        self.padding = (kernel_size - 1) // 2
        self.conv = nn.Conv2d(in_planes, out_planes, kernel_size, stride,
                              self.padding, groups=groups, bias=False)
        self.bn = nn.BatchNorm2d(out_planes)
        self.kernel_size = kernel_size
        self.stride = stride
        self.groups = groups
        if args.q_a > 0:
            self.quantize = QuantMeasure(args.q_a, stochastic=args.stochastic,
                                         scale=args.q_scale,
                                         calculate_running=args.calculate_running,
                                         pctl=args.pctl,
                                         debug=args.debug_quant)

    def forward(self, x):
        args = self.args
        if args.q_a > 0:
            x = self.quantize(x)
        x = _conv(x, self.conv)
        if args.merge_bn:
            bias = (self.bn.bias.view(1, -1, 1, 1)
                    - self.bn.running_mean.data.view(1, -1, 1, 1)
                    * self.bn.weight.data.view(1, -1, 1, 1)
                    / torch.sqrt(self.bn.running_var.data.view(1, -1, 1, 1) + args.eps))
            x = x + bias
            x = ops.relu_clip(x, 6.0)  # ReLU6
        else:
            x = ops.bn_act(x, self.bn.weight, self.bn.bias,
                           self.bn.running_mean, self.bn.running_var,
                           self.training, self.bn.momentum, self.bn.eps,
                           relu=True, act_max=6.0,
                           sync=getattr(args, 'sync_bn', False))
        return x


class InvertedResidual(nn.Module):
    def __init__(self, args, inp, oup, stride, expand_ratio):
        super().__init__()
        self.args = args
        self.stride = stride
        self.expand_ratio = expand_ratio
        assert stride in (1, 2)
        hidden_dim = int(round(inp * expand_ratio))
        self.use_res_connect = stride == 1 and inp == oup

        self.conv1 = ConvBNReLU(args, inp, hidden_dim, kernel_size=1)
        self.conv2 = ConvBNReLU(args, hidden_dim, hidden_dim, stride=stride,
                                groups=hidden_dim)
        self.conv3 = nn.Conv2d(hidden_dim, oup, 1, 1, 0, bias=False)
        self.bn = nn.BatchNorm2d(oup)
        if args.q_a > 0:
            qm = lambda: QuantMeasure(args.q_a, stochastic=args.stochastic,
                                      scale=args.q_scale,
                                      calculate_running=args.calculate_running,
                                      pctl=args.pctl, debug=args.debug_quant)
            self.quantize1 = qm()
            self.quantize2 = qm()
            self.quantize3 = qm()

    def forward(self, x):
        args = self.args
        input = x
        if self.expand_ratio != 1:
            x = self.conv1(x)
        x = self.conv2(x)
        if args.q_a > 0:
            x = self.quantize3(x)
        x = _conv(x, self.conv3)
        if args.merge_bn:
            bias = (self.bn.bias.view(1, -1, 1, 1)
                    - self.bn.running_mean.data.view(1, -1, 1, 1)
                    * self.bn.weight.data.view(1, -1, 1, 1)
                    / torch.sqrt(self.bn.running_var.data.view(1, -1, 1, 1) + args.eps))
            x = x + bias
        else:
            x = ops.bn_act(x, self.bn.weight, self.bn.bias,
                           self.bn.running_mean, self.bn.running_var,
                           self.training, self.bn.momentum, self.bn.eps,
                           relu=False, act_max=0.0,
                           sync=getattr(args, 'sync_bn', False))
        if self.use_res_connect:
            return x + input
        return x


class MobileNetV2(nn.Module):
    def __init__(self, args, num_classes=1000, width_mult=1.0,
                 inverted_residual_setting=None, round_nearest=8):
        super().__init__()
        self.args = args
        input_channel = 32
        last_channel = 1280
        if inverted_residual_setting is None:
            inverted_residual_setting = [
                [1, 16, 1, 1], [6, 24, 2, 2], [6, 32, 3, 2], [6, 64, 4, 2],
                [6, 96, 3, 1], [6, 160, 3, 2], [6, 320, 1, 1]]

        input_channel = _make_divisible(input_channel * width_mult, round_nearest)
        self.last_channel = _make_divisible(last_channel * max(1.0, width_mult),
                                            round_nearest)
        features = [ConvBNReLU(args, 3, input_channel, stride=2)]
        for t, c, n, s in inverted_residual_setting:
            output_channel = _make_divisible(c * width_mult, round_nearest)
            for i in range(n):
                stride = s if i == 0 else 1
                features.append(InvertedResidual(args, input_channel,
                                                 output_channel, stride,
                                                 expand_ratio=t))
                input_channel = output_channel
        features.append(ConvBNReLU(args, input_channel, self.last_channel,
                                   kernel_size=1))
        self.features = nn.Sequential(*features)
        self.drop1 = nn.Dropout(0.2)
        self.fc1 = nn.Linear(self.last_channel, num_classes)
        if args.bn_out:
            self.bn_out = nn.BatchNorm1d(num_classes,
                                         track_running_stats=args.track_running_stats)
        if args.q_a > 0:
            self.quantize = QuantMeasure(args.q_a, stochastic=args.stochastic,
                                         scale=args.q_scale,
                                         calculate_running=args.calculate_running,
                                         pctl=args.pctl, debug=args.debug_quant)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode='fan_out')
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, 0, 0.01)
                nn.init.zeros_(m.bias)

    def forward(self, x, epoch=0, i=0, acc=0.0):
        args = self.args
        x = self.features(x)
        x = x.mean([2, 3])
        x = ops.dropout(x, 0.2, self.training)
        if args.q_a > 0:
            x = self.quantize(x)
        x = ops.linear(x, self.fc1.weight, self.fc1.bias)
        if args.bn_out:
            x = ops.bn_act(x, self.bn_out.weight, self.bn_out.bias,
                           self.bn_out.running_mean, self.bn_out.running_var,
                           self.training or not args.track_running_stats,
                           self.bn_out.momentum, self.bn_out.eps, relu=False,
                           act_max=0.0)
        return x


def mobilenet_v2(parameters):
    """MobileNetV2: Inverted Residuals and Linear Bottlenecks
    (https://arxiv.org/abs/1801.04381), noise/quant-aware rebuild."""
    return MobileNetV2(parameters)
