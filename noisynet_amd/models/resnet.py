"""Noise-aware ResNet-18 (ImageNet), rebuilt from reference models/resnet.py.

Same layer graph and state_dict keys as the reference (conv1/bn1,
layer1-4 of BasicBlock{conv1,bn1,conv2,bn2[,conv3,bn3]}, fc) so reference
checkpoints load; NoisyConv2d everywhere, per-block QuantMeasure pairs,
Hardtanh clipping via the fused relu_clip kernel when act_max>0, the
forward-time BN-merge bias path (models/resnet.py:89-106) and the op-amp
offset distortion hooks (distort_tensor, models/resnet.py:70-77).
"""

import math

import torch
import torch.nn as nn

from .. import ops
from ..hardware_model import NoisyConv2d, NoisyLinear, distort_tensor
from ..quant import QuantMeasure


def _bn_bias(bn, eps):
    return (bn.bias.data.view(1, -1, 1, 1)
            - bn.running_mean.data.view(1, -1, 1, 1)
            * bn.weight.data.view(1, -1, 1, 1)
            / torch.sqrt(bn.running_var.data.view(1, -1, 1, 1) + eps))


class BasicBlock(nn.Module):
    def __init__(self, args, inplanes, planes, stride=1, downsample=None):
        super().__init__()
        self.args = args
        self.downsample = downsample
        self.stride = stride
        self.offset = args.offset

        if self.offset > 0:
            self.generate_offsets = True
            self.register_buffer('act1_offsets', torch.zeros(1))
            self.register_buffer('act2_offsets', torch.zeros(1))

        self.conv1 = NoisyConv2d(inplanes, planes, kernel_size=3, stride=stride,
                                 padding=1, bias=False, num_bits=0,
                                 num_bits_weight=args.q_w, noise=args.n_w,
                                 test_noise=args.n_w_test,
                                 stochastic=args.stochastic,
                                 debug=args.debug_noise)
        self.bn1 = nn.BatchNorm2d(planes, track_running_stats=args.track_running_stats)
        self.conv2 = NoisyConv2d(planes, planes, kernel_size=3, stride=1,
                                 padding=1, bias=False, num_bits=0,
                                 num_bits_weight=args.q_w, noise=args.n_w,
                                 test_noise=args.n_w_test,
                                 stochastic=args.stochastic,
                                 debug=args.debug_noise)
        self.bn2 = nn.BatchNorm2d(planes, track_running_stats=args.track_running_stats)

        if downsample is not None:
            ds_in, ds_out, ds_strides = downsample
            self.ds_strides = ds_strides
            self.conv3 = NoisyConv2d(ds_in, ds_out, kernel_size=1,
                                     stride=ds_strides, bias=False, num_bits=0,
                                     num_bits_weight=args.q_w, noise=args.n_w,
                                     test_noise=args.n_w_test,
                                     stochastic=args.stochastic,
                                     debug=args.debug_noise)
            self.bn3 = nn.BatchNorm2d(ds_out, track_running_stats=args.track_running_stats)

        if args.q_a > 0:
            self.quantize1 = QuantMeasure(args.q_a, stochastic=args.stochastic,
                                          scale=args.q_scale,
                                          calculate_running=args.calculate_running,
                                          pctl=args.pctl, debug=args.debug_quant,
                                          inplace=args.q_inplace)
            self.quantize2 = QuantMeasure(args.q_a, stochastic=args.stochastic,
                                          scale=args.q_scale,
                                          calculate_running=args.calculate_running,
                                          pctl=args.pctl, debug=args.debug_quant,
                                          inplace=args.q_inplace)

    def _bn_or_bias(self, x, bn, relu=False, act_max=0.0):
        args = self.args
        if args.merge_bn:
            x = x + _bn_bias(bn, args.eps)
            if relu:
                x = ops.relu_clip(x, act_max if act_max > 0 else 0.0)
            return x
        return ops.bn_act(x, bn.weight, bn.bias, bn.running_mean,
                          bn.running_var,
                          self.training or not args.track_running_stats,
                          bn.momentum, bn.eps, relu=relu, act_max=act_max,
                          sync=getattr(args, 'sync_bn', False))

    def forward(self, x):
        args = self.args
        act_max = args.act_max

        if args.distort_pre_act and self.offset:
            x = distort_tensor(self, args, x,
                               scale=args.offset * self.quantize1.running_max,
                               stop=False)
        if args.q_a > 0:
            x = self.quantize1(x)
        if args.distort_act and self.offset:
            x = distort_tensor(self, args, x, scale=args.offset * x.max(),
                               stop=False)

        residual = x
        out = self.conv1(x)
        out = self._bn_or_bias(out, self.bn1, relu=True, act_max=act_max)

        if args.distort_pre_act and self.offset:
            out = distort_tensor(self, args, out,
                                 scale=args.offset * self.quantize2.running_max,
                                 stop=True)
        if args.q_a > 0:
            out = self.quantize2(out)
        if args.distort_act and self.offset:
            out = distort_tensor(self, args, out, scale=args.offset * out.max(),
                                 stop=True)

        out = self.conv2(out)
        out = self._bn_or_bias(out, self.bn2, relu=False)

        if self.downsample is not None:
            residual = self.conv3(x)
            residual = self._bn_or_bias(residual, self.bn3, relu=False)

        out = out + residual
        out = ops.relu_clip(out, act_max if act_max > 0 else 0.0)
        return out


class ResNet(nn.Module):
    def __init__(self, args, num_classes=1000):
        super().__init__()
        self.args = args
        self.inplanes = 64

        self.offset = args.offset
        self.offset_input = args.offset_input
        if self.offset > 0:
            self.generate_offsets = True
            self.register_buffer('act2_offsets', torch.zeros(1))
        if self.offset_input > 0:
            self.generate_offsets = True
            self.register_buffer('input_offsets', torch.zeros(1))

        self.conv1 = NoisyConv2d(3, 64, kernel_size=7, stride=2, padding=3,
                                 bias=False, num_bits=0, num_bits_weight=args.q_w,
                                 noise=args.n_w, test_noise=args.n_w_test,
                                 stochastic=args.stochastic,
                                 debug=args.debug_noise)
        self.bn1 = nn.BatchNorm2d(64, track_running_stats=args.track_running_stats)

        # q_a_first selection (models/resnet.py:215-222)
        if args.q_a_first > 0:
            self.q_a_first = args.q_a_first
        elif args.q_a > 0:
            self.q_a_first = 6
        else:
            self.q_a_first = 0
        if self.q_a_first > 0:
            self.quantize1 = QuantMeasure(self.q_a_first, stochastic=args.stochastic,
                                          scale=args.q_scale,
                                          calculate_running=args.calculate_running,
                                          pctl=args.pctl, debug=args.debug_quant,
                                          inplace=args.q_inplace)
        if args.q_a > 0:
            self.quantize2 = QuantMeasure(args.q_a, stochastic=args.stochastic,
                                          scale=args.q_scale,
                                          calculate_running=args.calculate_running,
                                          pctl=args.pctl, debug=args.debug_quant,
                                          inplace=args.q_inplace)

        self.layer1 = self._make_layer(args, 64)
        self.layer2 = self._make_layer(args, 128, stride=2)
        self.layer3 = self._make_layer(args, 256, stride=2)
        self.layer4 = self._make_layer(args, 512, stride=2)

        self.fc = NoisyLinear(512, num_classes, bias=True, num_bits=0,
                              num_bits_weight=args.q_w, noise=args.n_w,
                              test_noise=args.n_w_test,
                              stochastic=args.stochastic,
                              debug=args.debug_noise)
        if args.bn_out:
            self.bn_out = nn.BatchNorm1d(num_classes,
                                         track_running_stats=args.track_running_stats)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                n = m.kernel_size[0] * m.kernel_size[1] * m.out_channels
                m.weight.data.normal_(0, math.sqrt(2. / n))
            elif isinstance(m, nn.BatchNorm2d):
                m.weight.data.fill_(1)
                m.bias.data.zero_()

    def _make_layer(self, args, planes, stride=1):
        downsample = None
        if stride != 1 or self.inplanes != planes:
            downsample = (self.inplanes, planes, stride)
        blocks = [BasicBlock(args, self.inplanes, planes, stride, downsample),
                  BasicBlock(args, planes, planes)]
        self.inplanes = planes
        return nn.Sequential(*blocks)

    def forward(self, x, epoch=0, i=0, acc=0.0):
        args = self.args
        act_max = args.act_max

        if args.distort_pre_act and self.offset_input:
            x = distort_tensor(self, args, x,
                               scale=args.offset_input * self.quantize1.running_max,
                               stop=self.offset == 0)
        if self.q_a_first > 0:
            x = self.quantize1(x)
        if args.distort_act and self.offset_input > 0:
            x = distort_tensor(self, args, x, scale=args.offset_input * x.max(),
                               stop=self.offset == 0)

        x = self.conv1(x)
        if args.merge_bn:
            x = x + _bn_bias(self.bn1, args.eps)
        else:
            x = ops.bn_act(x, self.bn1.weight, self.bn1.bias,
                           self.bn1.running_mean, self.bn1.running_var,
                           self.training or not args.track_running_stats,
                           self.bn1.momentum, self.bn1.eps, relu=False,
                           act_max=0.0, sync=getattr(args, 'sync_bn', False))
        x = ops.relu_clip(x, act_max if act_max > 0 else 0.0)
        x = ops.maxpool_nhwc(x, 3, 2, 1)

        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)

        k = min(7, x.size(2))  # CIFAR-sized inputs reach layer4 at < 7x7
        x = ops.avgpool_nhwc(x, k, 1, 0)
        x = x.reshape(x.size(0), -1)

        if args.distort_pre_act and self.offset > 0:
            x = distort_tensor(self, args, x,
                               scale=args.offset * self.quantize2.running_max,
                               stop=True)
        if args.q_a > 0:
            x = self.quantize2(x)
        if args.distort_act and self.offset:
            x = distort_tensor(self, args, x, scale=args.offset * x.max(),
                               stop=True)

        x = self.fc(x)
        if args.bn_out:
            x = ops.bn_act(x, self.bn_out.weight, self.bn_out.bias,
                           self.bn_out.running_mean, self.bn_out.running_var,
                           self.training or not args.track_running_stats,
                           self.bn_out.momentum, self.bn_out.eps, relu=False,
                           act_max=0.0)
        return x


def ResNet18(parameters):
    return ResNet(parameters)
