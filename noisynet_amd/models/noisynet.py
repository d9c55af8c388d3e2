"""The 4-layer CIFAR-10 ConvNet ("NoisyNet" proper).

Architecture and forward semantics reproduce reference noisynet.py:326-695:
conv1(3->fm1*w, fs x fs) -> [noise] -> maxpool2x2 -> bn1 -> relu -> clip ->
[dropout_conv] -> [quant] -> conv2 -> [noise] -> pool -> bn2 -> relu -> clip
-> [dropout] -> flatten -> [quant] -> linear1 -> [noise] -> bn3 -> relu ->
clip -> [dropout] -> [quant] -> linear2 -> [noise] -> bn4.

MI355X-first execution: when no introspection flag is set (the hot path),
each noisy layer is ONE fused kernel call (ops.fused_noisy_conv2d /
fused_noisy_linear: MFMA implicit-GEMM with dual accumulators computing the
clean output and the sigma-conv from the same input tiles, Gaussian noise
sampled in-kernel) and pool/BN/ReLU/clip run as fused epilogue kernels.
The introspective path (plot/merge_bn/L2_act*/print_stats/train_act_max)
falls back to the compositional ops to expose every intermediate the
reference exposes (model.conv1_, model.relu1_, ...).
"""

import torch
from torch import nn

from .. import ops
from ..hardware_model import NoisyConv2d, NoisyLinear, add_noise_calculate_power
from ..quant import QuantMeasure


class Net(nn.Module):
    def __init__(self, args=None):
        super().__init__()
        self.args = args
        self.create_dir = True

        if args.train_act_max:
            self.act_max1 = nn.Parameter(torch.Tensor([0]), requires_grad=True)
            self.act_max2 = nn.Parameter(torch.Tensor([0]), requires_grad=True)
            self.act_max3 = nn.Parameter(torch.Tensor([0]), requires_grad=True)
        if args.train_w_max:
            self.w_max1 = nn.Parameter(torch.Tensor([0]), requires_grad=True)
            self.w_min1 = nn.Parameter(torch.Tensor([0]), requires_grad=True)

        self.pool = nn.MaxPool2d(2, 2)
        self.relu = nn.ReLU()

        self.quantize1 = QuantMeasure(args.q_a1, stochastic=args.stochastic,
                                      pctl=args.pctl, max_value=1.0,
                                      debug=args.debug_quant)
        self.quantize2 = QuantMeasure(args.q_a2, stochastic=args.stochastic,
                                      pctl=args.pctl, debug=args.debug_quant)
        self.quantize3 = QuantMeasure(args.q_a3, stochastic=args.stochastic,
                                      pctl=args.pctl,
                                      max_value=args.act_max / (1. - args.dropout),
                                      debug=args.debug_quant)
        self.quantize4 = QuantMeasure(args.q_a4, stochastic=args.stochastic,
                                      pctl=args.pctl, debug=args.debug_quant)

        fm1, fm2 = args.fm1 * args.width, args.fm2 * args.width
        fc = args.fc * args.width
        self.conv1 = NoisyConv2d(3, fm1, kernel_size=args.fs, bias=args.use_bias,
                                 num_bits=0, num_bits_weight=args.q_w1,
                                 noise=args.n_w1, test_noise=args.n_w_test,
                                 stochastic=args.stochastic, debug=args.debug_noise)
        self.conv2 = NoisyConv2d(fm1, fm2, kernel_size=args.fs, bias=args.use_bias,
                                 num_bits=0, num_bits_weight=args.q_w2,
                                 noise=args.n_w2, test_noise=args.n_w_test,
                                 stochastic=args.stochastic, debug=args.debug_noise)
        self.linear1 = NoisyLinear(fm2 * args.fs * args.fs, fc, bias=args.use_bias,
                                   num_bits=0, num_bits_weight=args.q_w3,
                                   noise=args.n_w3, test_noise=args.n_w_test,
                                   stochastic=args.stochastic, debug=args.debug_noise)
        self.linear2 = NoisyLinear(fc, 10, bias=args.use_bias,
                                   num_bits=0, num_bits_weight=args.q_w4,
                                   noise=args.n_w4, test_noise=args.n_w_test,
                                   stochastic=args.stochastic, debug=args.debug_noise)

        if args.batchnorm:
            self.bn1 = nn.BatchNorm2d(fm1, track_running_stats=args.track_running_stats)
            self.bn2 = nn.BatchNorm2d(fm2, track_running_stats=args.track_running_stats)
            if args.bn3:
                self.bn3 = nn.BatchNorm1d(fc, track_running_stats=args.track_running_stats)
            if args.bn4:
                self.bn4 = nn.BatchNorm1d(10, track_running_stats=args.track_running_stats)

        if args.dropout > 0:
            self.dropout = nn.Dropout(p=args.dropout)

        # telemetry lists (reset per epoch by the driver)
        self.power = [[] for _ in range(args.num_layers)]
        self.nsr = [[] for _ in range(args.num_layers)]
        self.input_sparsity = [[] for _ in range(args.num_layers)]

    # ------------------------------------------------------------------
    def _introspective(self):
        a = self.args
        return (a.plot or a.write or a.merge_bn or a.distort_act
                or a.L2_act1 > 0 or a.L2_act2 > 0 or a.L2_act3 > 0
                or a.L2_act4 > 0 or a.print_stats or a.L3_act > 0
                or a.train_act_max or a.uniform_ind > 0 or a.uniform_dep > 0
                or a.normal_ind > 0 or a.normal_dep > 0)

    def _noisy_layer_fused(self, x, layer, current, merged_dac, i, layer_num,
                           is_conv):
        """One fused pass: quantized-weight conv/GEMM + sigma + noise."""
        a = self.args
        w_raw = layer.weight
        wq, bias = layer.effective_weight()
        want_telemetry = i < 20
        with torch.no_grad():
            # input_max is a full-tensor reduce over the activation; for
            # merged-DAC layers it only feeds the power telemetry
            # (first-20-batch), so skip it in steady state
            if merged_dac:
                w_max = w_raw.detach().abs().max()
                factor = 0.1 * w_max / current
                sigma_mode = 'abs'
                power_denom = (x.detach().max() * w_max if want_telemetry
                               else w_max)
            else:
                input_max = x.detach().max()
                factor = 0.1 * input_max / current
                sigma_mode = 'abs2'
                power_denom = input_max
        telem = ops.NoiseTelemetry() if want_telemetry else None
        if is_conv:
            out = ops.fused_noisy_conv2d(
                x, wq, w_raw.detach(), bias, 1, 0, sigma_mode, factor,
                current=current, power_denom=power_denom,
                want_telemetry=want_telemetry, telemetry_out=telem)
        else:
            out = ops.fused_noisy_linear(
                x, wq, w_raw.detach(), bias, sigma_mode, factor,
                current=current, power_denom=power_denom,
                want_telemetry=want_telemetry, telemetry_out=telem)
        if want_telemetry and telem.power is not None:
            self.power[layer_num].append(float(telem.power))
            self.nsr[layer_num].append(float(telem.nsr))
            self.input_sparsity[layer_num].append(float(telem.input_sparsity))
        return out

    def _bn_act(self, x, bn, act_max):
        """Fused BN + ReLU + clip."""
        return ops.bn_act(x, bn.weight, bn.bias, bn.running_mean,
                          bn.running_var,
                          self.training or not self.args.track_running_stats,
                          bn.momentum, bn.eps, relu=True, act_max=act_max,
                          sync=getattr(self.args, 'sync_bn', False))

    # ------------------------------------------------------------------
    def forward(self, input, epoch=0, i=0, s=0, acc=0.0):
        args = self.args
        if not self._introspective():
            return self._forward_fused(input, i)
        return self._forward_reference(input, epoch, i, s, acc)

    # -- the hot path ---------------------------------------------------
    def _forward_fused(self, input, i=0):
        args = self.args

        x = self.quantize1(input) if args.q_a1 > 0 else input
        self.input = x

        if args.current1 > 0:
            x = self._noisy_layer_fused(x, self.conv1, args.current1,
                                        args.merged_dac, i, 0, True)
        else:
            x = self.conv1(x)
        self.conv1_ = x
        x = ops.maxpool2x2(x)
        if args.batchnorm:
            x = self._bn_act(x, self.bn1, args.act_max1)
        else:
            x = ops.relu_clip(x, args.act_max1)
        if args.dropout_conv > 0:
            x = ops.dropout(x, args.dropout_conv, self.training)
        if args.q_a2 > 0:
            x = self.quantize2(x)

        if args.current2 > 0:
            x = self._noisy_layer_fused(x, self.conv2, args.current2,
                                        False, i, 1, True)
        else:
            x = self.conv2(x)
        self.conv2_ = x
        x = ops.maxpool2x2(x)
        if args.batchnorm:
            x = self._bn_act(x, self.bn2, args.act_max2)
        else:
            x = ops.relu_clip(x, args.act_max2)
        if args.dropout > 0:
            x = ops.dropout(x, args.dropout, self.training)

        x = x.reshape(x.size(0), -1)
        if args.q_a3 > 0:
            x = self.quantize3(x)

        if args.current3 > 0:
            x = self._noisy_layer_fused(x, self.linear1, args.current3,
                                        args.merged_dac, i, 2, False)
        else:
            x = self.linear1(x)
        self.linear1_ = x
        if args.batchnorm and args.bn3:
            x = self._bn_act(x, self.bn3, args.act_max3)
        else:
            x = ops.relu_clip(x, args.act_max3)
        if args.dropout > 0:
            x = ops.dropout(x, args.dropout, self.training)
        if args.q_a4 > 0:
            x = self.quantize4(x)

        if args.current4 > 0:
            x = self._noisy_layer_fused(x, self.linear2, args.current4,
                                        False, i, 3, False)
        else:
            x = self.linear2(x)
        self.linear2_ = x
        if args.batchnorm and args.bn4:
            x = ops.bn_act(x, self.bn4.weight, self.bn4.bias,
                           self.bn4.running_mean, self.bn4.running_var,
                           self.training or not args.track_running_stats,
                           self.bn4.momentum, self.bn4.eps,
                           relu=False, act_max=0.0,
                           sync=getattr(args, 'sync_bn', False))
        self.linear2_out = x
        return x

    # -- the introspective path (exact reference op order + attributes) --
    def _forward_reference(self, input, epoch=0, i=0, s=0, acc=0.0):
        args = self.args
        arrays = []

        if args.q_a1 > 0:
            self.input = self.quantize1(input)
        else:
            self.input = input

        self.conv1_no_bias = self.conv1(self.input)

        if args.merge_bn:
            self.bias1 = (self.bn1.bias.view(1, -1, 1, 1)
                          - self.bn1.running_mean.data.view(1, -1, 1, 1)
                          * self.bn1.weight.data.view(1, -1, 1, 1)
                          / torch.sqrt(self.bn1.running_var.data.view(1, -1, 1, 1) + 1e-7))
            self.conv1_ = self.conv1_no_bias + self.bias1
        else:
            self.conv1_ = self.conv1_no_bias

        if args.current1 > 0 or args.distort_act:
            conv1_out = add_noise_calculate_power(
                self, args, arrays, self.input, self.conv1.weight, self.conv1_,
                layer_type='conv', i=i, layer_num=0, merged_dac=args.merged_dac)
        else:
            conv1_out = self.conv1_

        pool1 = self.pool(conv1_out)
        if args.batchnorm and not args.merge_bn:
            self.pool1_out = self.bn1(pool1)
        else:
            self.pool1_out = pool1

        self.relu1_ = self.relu(self.pool1_out)
        if args.act_max1 > 0:
            if args.train_act_max:
                self.relu1_clipped = torch.where(self.relu1_ > self.act_max1,
                                                 self.act_max1, self.relu1_)
            else:
                self.relu1_clipped = torch.clamp(self.relu1_, max=args.act_max1)
            self.relu1 = self.relu1_clipped
        else:
            self.relu1 = self.relu1_

        if args.L3_act > 0:
            self.relu1_.retain_grad()
            self.relu1.retain_grad()
        if args.train_act_max:
            self.relu1_clipped.retain_grad()
        if args.train_w_max:
            self.w_max1.retain_grad()
            self.w_min1.retain_grad()

        if args.dropout_conv > 0:
            self.relu1 = self.dropout(self.relu1)
        if args.q_a2 > 0:
            self.relu1 = self.quantize2(self.relu1)

        self.conv2_no_bias = self.conv2(self.relu1)
        if args.merge_bn:
            self.bias2 = (self.bn2.bias.view(1, -1, 1, 1)
                          - self.bn2.running_mean.data.view(1, -1, 1, 1)
                          * self.bn2.weight.data.view(1, -1, 1, 1)
                          / torch.sqrt(self.bn2.running_var.data.view(1, -1, 1, 1) + 1e-7))
            self.conv2_ = self.conv2_no_bias + self.bias2
        else:
            self.conv2_ = self.conv2_no_bias

        if args.current2 > 0 or args.distort_act:
            conv2_out = add_noise_calculate_power(
                self, args, arrays, self.relu1, self.conv2.weight, self.conv2_,
                layer_type='conv', i=i, layer_num=1, merged_dac=False)
        else:
            conv2_out = self.conv2_

        pool2 = self.pool(conv2_out)
        if args.batchnorm and not args.merge_bn:
            self.pool2_out = self.bn2(pool2)
        else:
            self.pool2_out = pool2

        self.relu2_ = self.relu(self.pool2_out)
        if args.act_max2 > 0:
            if args.train_act_max:
                self.relu2_clipped = torch.where(self.relu2_ > self.act_max2,
                                                 self.act_max2, self.relu2_)
            else:
                self.relu2_clipped = torch.clamp(self.relu2_, max=args.act_max2)
            self.relu2 = self.relu2_clipped
        else:
            self.relu2 = self.relu2_

        if args.L3_act > 0:
            self.relu2.retain_grad()
        if args.dropout > 0:
            self.relu2 = self.dropout(self.relu2)

        self.relu2 = self.relu2.reshape(self.relu2.size(0), -1)
        if args.q_a3 > 0:
            self.relu2 = self.quantize3(self.relu2)

        self.linear1_no_bias = self.linear1(self.relu2)
        if args.merge_bn:
            self.bias3 = (self.bn3.bias.view(1, -1)
                          - self.bn3.running_mean.data.view(1, -1)
                          * self.bn3.weight.data.view(1, -1)
                          / torch.sqrt(self.bn3.running_var.data.view(1, -1) + 1e-7))
            self.linear1_ = self.linear1_no_bias + self.bias3
        else:
            self.linear1_ = self.linear1_no_bias

        if args.current3 > 0 or args.distort_act:
            linear1_out = add_noise_calculate_power(
                self, args, arrays, self.relu2, self.linear1.weight,
                self.linear1_, layer_type='linear', i=i, layer_num=2,
                merged_dac=args.merged_dac)
        else:
            linear1_out = self.linear1_

        if args.batchnorm and args.bn3 and not args.merge_bn:
            self.linear1_out = self.bn3(linear1_out)
        else:
            self.linear1_out = linear1_out

        self.relu3_ = self.relu(self.linear1_out)
        if args.act_max3 > 0:
            if args.train_act_max:
                self.relu3_clipped = torch.where(self.relu3_ > self.act_max3,
                                                 self.act_max3, self.relu3_)
            else:
                self.relu3_clipped = torch.clamp(self.relu3_, max=args.act_max3)
            self.relu3 = self.relu3_clipped
        else:
            self.relu3 = self.relu3_

        if args.L3_act > 0:
            self.relu3.retain_grad()
        if args.dropout > 0:
            self.relu3 = self.dropout(self.relu3)
        if args.q_a4 > 0:
            self.relu3 = self.quantize4(self.relu3)

        self.linear2_no_bias = self.linear2(self.relu3)
        if args.bn4 and args.merge_bn:
            if self.training:
                raise RuntimeError('Merging BatchNorm during training!')
            self.bias4 = (self.bn4.bias.view(1, -1)
                          - self.bn4.running_mean.data.view(1, -1)
                          * self.bn4.weight.data.view(1, -1)
                          / torch.sqrt(self.bn4.running_var.data.view(1, -1) + 1e-7))
            self.linear2_ = self.linear2_no_bias + self.bias4
        else:
            self.linear2_ = self.linear2_no_bias
            self.bias4 = torch.Tensor([0])

        if args.current4 > 0 or args.distort_act:
            linear2_out = add_noise_calculate_power(
                self, args, arrays, self.relu3, self.linear2.weight,
                self.linear2_, layer_type='linear', i=i, layer_num=3,
                merged_dac=False)
        else:
            linear2_out = self.linear2_

        if args.batchnorm and args.bn4 and not args.merge_bn:
            self.linear2_out = self.bn4(linear2_out)
        else:
            self.linear2_out = linear2_out

        if args.plot or args.write:
            from .. import plot_histograms  # lazy; matplotlib is heavy
            plot_histograms.capture_and_emit(self, args, arrays, epoch, i, s, acc)

        return self.linear2_out


def noisynet(args):
    return Net(args)
