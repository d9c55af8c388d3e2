"""Analog hardware (current-noise + power) model and noisy layers.

Rebuilds reference hardware_model.py on the fused native op layer:

* ``add_noise_calculate_power`` keeps the reference's exact math
  (hardware_model.py:16-127, restated in SURVEY.md §2.3) but on GPU the
  sigma-conv + Gaussian sample + add are ONE fused kernel pass via
  ``ops.fused_noisy_conv2d`` / ``ops.fused_noisy_linear`` -- the reference
  pays a full second cuDNN conv over |W| plus a separate curand sample.
* ``NoisyConv2d``/``NoisyLinear`` reproduce hardware_model.py:310-423.
* ``AddNoise`` / ``distort_tensor`` reproduce :291-307 / :426-458.
"""

import torch
from torch import nn

from . import ops
from .quant import QuantMeasure


def add_noise_calculate_power(self, args, arrays, input, weights, output,
                              layer_type='conv', i=0, layer_num=0,
                              merged_dac=True, stride=1, padding=0):
    """Inject analog VMM noise into ``output`` and log power/NSR/sparsity.

    Exact semantics of hardware_model.py:16-127. ``self`` is the model (for
    .training and the power/nsr/input_sparsity telemetry lists); noise math
    runs under no_grad so gradients flow only through the clean output.

    NOTE on the fused path: the clean ``output`` tensor passed in was already
    produced by the layer's conv/linear, so this call runs the conv kernel in
    sigma-only mode (y accumulator disabled; sigma + in-kernel Gaussian).
    Model forward paths that know they will add noise should instead call
    ``ops.fused_noisy_conv2d`` directly (one pass computes y AND noise); the
    models in ``noisynet_amd.models`` do exactly that.
    """
    if args.distort_act:
        with torch.no_grad():
            noise = output * torch.empty_like(output).uniform_(-args.noise, args.noise)
        return output + noise

    training = self.training
    simple = None
    for mode in ('uniform_ind', 'uniform_dep', 'normal_ind', 'normal_dep'):
        a = getattr(args, mode, 0.0)
        if a > 0 and (training or args.noise_test):
            simple = (mode, a)
            break

    if simple is not None:
        mode, a = simple
        noise = ops.simple_noise(output, mode, a)
        if mode == 'uniform_dep':
            return output * noise
        return output + noise

    current = args.layer_currents[layer_num]
    want_telemetry = i < 20

    with torch.no_grad():
        x = input.detach()
        w = weights.detach()
        input_max = x.max()
        if merged_dac:
            w_max = w.abs().max()
            factor = 0.1 * w_max / current
            sigma_mode = 'abs'
            power_denom = input_max * w_max
        else:
            factor = 0.1 * input_max / current
            sigma_mode = 'abs2'
            power_denom = input_max

        if layer_type == 'conv':
            noise, sig_mean = ops.sigma_noise_conv2d(
                x, w, sigma_mode, factor, stride, padding,
                want_sigma_abs=want_telemetry)
        else:
            noise, sig_mean = ops.sigma_noise_linear(
                x, w, sigma_mode, factor, want_sigma_abs=want_telemetry)

        if want_telemetry:
            p = 1.0e-6 * 1.2 * current * sig_mean / power_denom
            self.power[layer_num].append(float(p))
            # nsr divides by max of the CLEAN output (hardware_model.py:87)
            self.nsr[layer_num].append(
                float((noise.abs().mean() / output.detach().max()).item()))
            self.input_sparsity[layer_num].append(
                float((x > 0).sum().item() / x.numel()))

    return output + noise


class AddNoise(torch.autograd.Function):
    """out = w + w*U(-noise,+noise); identity STE (hardware_model.py:291-307)."""

    @staticmethod
    def forward(ctx, input, noise=0, debug=False):
        with torch.no_grad():
            if input.is_cuda and ops.has_ext():
                return ops.ext().mult_uniform_noise(
                    input, float(noise),
                    int(torch.randint(0, 2 ** 62, (1,)).item()))
            return ops.reference.mult_uniform_noise(input, noise)

    @staticmethod
    def backward(ctx, grad_output):
        return grad_output, None, None


class NoisyConv2d(nn.Conv2d):
    """Conv2d with optional input/weight fake-quant or mult. weight noise.

    hardware_model.py:310-366. On GPU the conv runs through the MFMA
    implicit-GEMM kernel (ops.conv2d); weight quantization is an elementwise
    HIP kernel over the (small) weight tensor feeding the conv.
    """

    def __init__(self, in_channels, out_channels, kernel_size, stride=1,
                 padding=0, dilation=1, groups=1, bias=False, num_bits=0,
                 num_bits_weight=0, noise=0.5, test_noise=0, stochastic=True,
                 debug=False):
        super().__init__(in_channels, out_channels, kernel_size, stride,
                         padding, dilation, groups, bias)
        self.num_bits = num_bits
        self.fms = out_channels
        self.fs = kernel_size
        self.noise = noise
        self.num_bits_weight = num_bits_weight
        if num_bits > 0:
            self.quantize_input = QuantMeasure(num_bits, stochastic=stochastic,
                                               debug=debug)
        if num_bits_weight > 0:
            self.quantize_weights = QuantMeasure(num_bits_weight,
                                                 min_value=-1.0, max_value=1.0,
                                                 stochastic=stochastic,
                                                 debug=debug)
        self.stochastic = stochastic
        self.debug = debug
        self.test_noise = test_noise

    def effective_weight(self):
        weight = self.weight
        bias = self.bias
        if self.num_bits_weight > 0:
            weight = self.quantize_weights(self.weight)
        elif self.test_noise > 0 and not self.training:
            weight = AddNoise.apply(self.weight, self.test_noise, self.debug)
            if bias is not None:
                bias = AddNoise.apply(self.bias, self.test_noise, self.debug)
        elif self.noise > 0 and self.training:
            weight = AddNoise.apply(self.weight, self.noise, self.debug)
            if bias is not None:
                bias = AddNoise.apply(self.bias, self.noise, self.debug)
        return weight, bias

    def forward(self, input):
        if 0 < self.num_bits < 8:
            qinput = self.quantize_input(input)
        else:
            qinput = input
        weight, bias = self.effective_weight()
        if self.groups == 1 and self.dilation == (1, 1):
            return ops.conv2d(qinput, weight, bias, self.stride, self.padding)
        if (self.groups == self.in_channels == self.out_channels
                and self.dilation == (1, 1)):
            return ops.depthwise_conv2d(qinput, weight, bias, self.stride,
                                        self.padding)
        # exotic group counts (e.g. CondConv's per-sample grouped conv) go
        # through the library path
        return nn.functional.conv2d(qinput, weight, bias, self.stride,
                                    self.padding, self.dilation, self.groups)


class NoisyLinear(nn.Linear):
    """hardware_model.py:369-423 on the MFMA GEMM kernel."""

    def __init__(self, in_features, out_features, bias=False, num_bits=0,
                 num_bits_weight=0, noise=0, test_noise=0, stochastic=True,
                 debug=False):
        super().__init__(in_features, out_features, bias)
        self.fc_in = in_features
        self.fc_out = out_features
        self.num_bits = num_bits
        self.num_bits_weight = num_bits_weight
        self.noise = noise
        if num_bits > 0:
            self.quantize_input = QuantMeasure(num_bits, stochastic=stochastic,
                                               debug=debug)
        if num_bits_weight > 0:
            self.quantize_weights = QuantMeasure(num_bits_weight,
                                                 min_value=-1.0, max_value=1.0,
                                                 stochastic=stochastic,
                                                 debug=debug)
        self.stochastic = stochastic
        self.debug = debug
        self.test_noise = test_noise

    def effective_weight(self):
        weight = self.weight
        bias = self.bias
        if 0 < self.num_bits_weight < 8:
            weight = self.quantize_weights(self.weight)
        elif self.test_noise > 0 and not self.training:
            weight = AddNoise.apply(self.weight, self.test_noise, self.debug)
            if bias is not None:
                bias = AddNoise.apply(self.bias, self.test_noise, self.debug)
        elif self.noise > 0 and self.training:
            weight = AddNoise.apply(self.weight, self.noise, self.debug)
            if bias is not None:
                bias = AddNoise.apply(self.bias, self.noise, self.debug)
        return weight, bias

    def forward(self, input):
        if 0 < self.num_bits < 8:
            qinput = self.quantize_input(input)
        else:
            qinput = input
        weight, bias = self.effective_weight()
        return ops.linear(qinput, weight, bias)


def distort_tensor(self, args, input, scale=0, stop=False):
    """Op-amp offset simulation: persistent per-activation Normal offsets
    generated once and reused across batches (hardware_model.py:426-458)."""
    with torch.no_grad():
        if args.offset or args.offset_input:
            if self.generate_offsets:
                distr = torch.randn_like(input) * scale
                if 224 in list(input.shape):
                    self.input_offsets = distr
                elif stop:
                    self.act2_offsets = distr
                else:
                    self.act1_offsets = distr
                if stop:
                    self.generate_offsets = False
            if 224 in list(input.shape):
                out = input + self.input_offsets
            elif stop:
                out = input + self.act2_offsets
            else:
                out = input + self.act1_offsets
        else:
            noise = input * torch.empty_like(input).uniform_(-args.noise, args.noise)
            out = input + noise
    return out
