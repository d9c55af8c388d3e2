"""Quantization core: UniformQuantize / QuantMeasure (API-compatible).

Rebuilds the behaviour of reference quant.py:10-137 and the evolved copies in
hardware_model.py:130-288 on top of the native op layer
(``noisynet_amd.ops.fake_quant`` -> csrc/quantize.hip on GPU).
"""

import torch
from torch import nn

from . import ops


class UniformQuantize(torch.autograd.Function):
    """API shim matching reference hardware_model.py:130-183.

    Prefer ``ops.fake_quant``; this class exists so user code written against
    the reference (`UniformQuantize().apply(x, bits, mn, mx, stoch)`) works.
    """

    @staticmethod
    def forward(ctx, input, num_bits=8, min_value=None, max_value=None,
                stochastic=0.5, inplace=False, debug=False):
        ctx.min_value = min_value
        ctx.max_value = max_value
        ctx.save_for_backward(input)
        with torch.no_grad():
            return ops.reference.fake_quant_forward(
                input, num_bits, min_value, max_value, stochastic
            ) if not input.is_cuda else ops.ext().fake_quant_fwd(
                input, int(num_bits), float(min_value), float(max_value),
                float(stochastic),
                int(torch.randint(0, 2 ** 62, (1,)).item()))

    @staticmethod
    def backward(ctx, grad_output):
        (input,) = ctx.saved_tensors
        if input.is_cuda:
            grad = ops.ext().ste_mask(grad_output, input,
                                      float(ctx.min_value),
                                      float(ctx.max_value))
        else:
            grad = ops.reference.fake_quant_backward(
                grad_output, input, ctx.min_value, ctx.max_value)
        return grad, None, None, None, None, None, None


class QuantMeasure(nn.Module):
    """Activation/weight range observer + fake-quantizer.

    Reproduces hardware_model.py:186-288:
      * calculate_running: for the first few batches compute the pctl-th
        percentile (kthvalue) of the tensor and append to ``running_list``;
        the driver averages the list into ``running_max`` after 5 batches.
      * negative-range mode (min_value < 0, for weights): separate pos/neg
        percentiles set running_min/running_max immediately.
      * eval: stochastic rounding off.
      * ImageNet first layer (an input dim == 224): pctl 0.92 iff 4-bit,
        else 1.0 (hardware_model.py:241-245).
    """

    def __init__(self, num_bits=8, momentum=0.0, stochastic=0.5, min_value=0.,
                 max_value=0., scale=1, calculate_running=False, pctl=90.,
                 debug=False, inplace=False):
        super().__init__()
        self.register_buffer('running_min', torch.zeros(1))
        self.register_buffer('running_max', torch.zeros([]))
        self.momentum = momentum
        self.num_bits = num_bits
        self.stochastic = stochastic
        self.inplace = inplace
        self.debug = debug
        self.max_value = max_value
        self.min_value = min_value
        self.scale = scale
        self.calculate_running = calculate_running
        self.running_list = []
        self.pctl = pctl
        # calibrated range cached as Python floats: the hot path must not
        # read device scalars (host sync; also illegal under hipGraph
        # capture). Invalidated when buffers change (finish_calibration,
        # checkpoint load).
        self._range_cache = None
        if pctl < 1:
            raise ValueError('pctl is {} please check'.format(pctl))

    def _load_from_state_dict(self, *args, **kwargs):
        self._range_cache = None
        return super()._load_from_state_dict(*args, **kwargs)

    def forward(self, input):
        with torch.no_grad():
            min_value = self.min_value
            max_value = self.max_value
            if self.calculate_running:
                if self.min_value < 0:  # weights: separate pos/neg percentiles
                    pos = input[input > 0]
                    neg = input[input < 0].abs()
                    pctl_pos = ops.kth_percentile(pos, self.pctl)
                    pctl_neg = ops.kth_percentile(neg, self.pctl)
                    self.running_min = -pctl_neg.reshape(1)
                    self.running_max = pctl_pos.reshape([])
                    self.calculate_running = False
                    min_value = float(self.running_min.item())
                    max_value = float(self.running_max.item())
                else:
                    if 224 in list(input.shape):
                        if self.num_bits == 4:
                            pctl_val = torch.tensor(0.92)
                        else:
                            pctl_val = torch.tensor(1.0)
                    else:
                        pctl_val = ops.kth_percentile(input, self.pctl)
                    max_value = input.max().item()
                    self.running_list.append(pctl_val)
            elif self._range_cache is not None:
                min_value, max_value = self._range_cache
            else:
                cacheable = True
                if self.min_value < 0 and float(self.running_min.min()) < 0:
                    min_value = float(self.running_min.item())
                    max_value = float(self.running_max.item())
                elif self.max_value > 0:
                    max_value = self.max_value
                elif float(self.running_max) > 0:
                    max_value = float(self.running_max)
                else:
                    max_value = input.max().item()  # data-dependent
                    cacheable = False
                if cacheable:
                    self._range_cache = (min_value, max_value)

            stoch = self.stochastic if self.training else 0

        return ops.fake_quant(input, self.num_bits, min_value, max_value, stoch)


def finish_calibration(model, device=None):
    """Average each QuantMeasure's running_list into running_max.

    The driver-side calibration stop of noisynet.py:1251-1259 /
    main.py:944-951, factored into a reusable helper.
    """
    with torch.no_grad():
        for m in model.modules():
            if isinstance(m, QuantMeasure):
                m._range_cache = None
                if m.calculate_running and m.running_list:
                    m.calculate_running = False
                    vals = torch.stack([torch.as_tensor(v, dtype=torch.float32)
                                        for v in m.running_list])
                    m.running_max = vals.mean().to(
                        device if device is not None else m.running_max.device)


def start_calibration(model):
    """Arm every QuantMeasure for range calibration (noisynet.py:1226-1231)."""
    for m in model.modules():
        if isinstance(m, QuantMeasure):
            m.calculate_running = True
            m.running_list = []
