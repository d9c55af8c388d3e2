"""Autograd-facing op layer with native (HIP/gfx950) vs reference dispatch.

Every hot op has two execution paths:
  * CUDA(ROCm) tensors -> the hand-written CDNA4 kernels in csrc/ (mandatory
    on GPU; a missing extension raises, never silently falls back), laid out
    NHWC (``torch.channels_last``) for convs.
  * CPU tensors -> ``noisynet_amd.ops.reference`` (pure PyTorch), which is
    also the oracle the GPU kernels are unit-tested against.

Autograd structure mirrors the reference semantics:
  * fake-quant uses a saturated STE (zero outside [min,max]) --
    hardware_model.py:175-183;
  * VMM noise is sampled under no_grad and added to the clean pre-activation,
    so gradients flow only through the clean conv/GEMM --
    hardware_model.py:23,125;
  * conv forward/dgrad/wgrad are mutually composable autograd Functions so
    the L3/L3_act/L4 gradient penalties (double/triple backward,
    noisynet.py:1348-1476) work through the custom kernels.
"""

import os as _os

import torch
import torch.nn.functional as F

from . import reference as ref
from ._ext import ext, use_native

# ---------------------------------------------------------------------------
# seeds for in-kernel Philox RNG: deterministic under torch.manual_seed
# ---------------------------------------------------------------------------

def _next_seed(device):
    # Draw from torch's generator so --seed reproduces kernel-side RNG too.
    return int(torch.randint(0, 2 ** 62, (1,), device="cpu").item())


# ---------------------------------------------------------------------------
# Fake quantization (UniformQuantize parity)
# ---------------------------------------------------------------------------


class FakeQuant(torch.autograd.Function):
    """Clamp-quantize-dequantize with stochastic rounding; saturated STE bwd."""

    @staticmethod
    def forward(ctx, x, num_bits, min_value, max_value, stochastic):
        ctx.min_value = float(min_value)
        ctx.max_value = float(max_value)
        ctx.save_for_backward(x)
        if use_native(x):
            return ext().fake_quant_fwd(
                x, int(num_bits), float(min_value), float(max_value),
                float(stochastic), _next_seed(x.device))
        return ref.fake_quant_forward(x, num_bits, min_value, max_value, stochastic)

    @staticmethod
    def backward(ctx, grad_output):
        (x,) = ctx.saved_tensors
        if use_native(x):
            # ste_mask aligns layouts internally (x.suggest_memory_format);
            # a plain .contiguous() here would force an NCHW round-trip
            g = ext().ste_mask(grad_output, x, ctx.min_value, ctx.max_value)
        else:
            g = ref.fake_quant_backward(grad_output, x, ctx.min_value, ctx.max_value)
        return g, None, None, None, None


def fake_quant(x, num_bits, min_value, max_value, stochastic=0.0):
    return FakeQuant.apply(x, num_bits, min_value, max_value, stochastic)



def _factor_tensor(f, device):
    """Noise factor as a device-resident f32 scalar (no host sync)."""
    if isinstance(f, torch.Tensor):
        return f.detach().to(device=device, dtype=torch.float32).reshape(1)
    return torch.tensor([float(f)], device=device, dtype=torch.float32)


# ---------------------------------------------------------------------------
# Conv primitives: forward / dgrad / wgrad, each differentiable by composition
# ---------------------------------------------------------------------------


def _nhwc(t):
    return t.contiguous(memory_format=torch.channels_last)


def _patch_eligible(x, w, padding):
    """Mirror of csrc patch_eligible: image-resident LDS conv mode."""
    C = x.shape[1]
    R, S = w.shape[2], w.shape[3]
    c_pad = (C + 7) // 8 * 8
    wp = x.shape[3] + 2 * padding
    eb = x.element_size()
    return (R * S > 1 and C > 8
            and x.shape[2] * wp * c_pad * eb + 128 <= 96 * 1024)


def _conv_fwd_raw(x, w, bias, stride, padding):
    if use_native(x, w):
        R, S = w.shape[2], w.shape[3]
        if _patch_eligible(x, w, padding):
            empty = torch.empty(0, device=x.device, dtype=x.dtype)
            zero_f = torch.zeros(1, device=x.device, dtype=torch.float32)
            y = ext().conv_fwd_fused(_nhwc(x), _nhwc(w), _nhwc(w), empty,
                                     stride, padding, 0, zero_f, 0, False)[0]
        else:
            y = ext().conv_fwd(_nhwc(x), _nhwc(w), stride, padding)
        if bias is not None:
            y = y + bias.view(1, -1, 1, 1)
        return y
    return F.conv2d(x, w, bias, stride, padding)


def _conv_dgrad_raw(g, w, stride, padding, x_shape):
    if use_native(g, w):
        R, S = w.shape[2], w.shape[3]
        if stride == 1 and _patch_eligible(g, w, R - 1 - padding):
            # dgrad(stride 1) == conv of g with the flipped/transposed
            # filter at full-correlation padding -> image-patch kernel
            w2 = _nhwc(w.flip((2, 3)).transpose(0, 1))
            empty = torch.empty(0, device=g.device, dtype=g.dtype)
            zero_f = torch.zeros(1, device=g.device, dtype=torch.float32)
            dx = ext().conv_fwd_fused(_nhwc(g), w2, w2, empty, 1,
                                      R - 1 - padding, 0, zero_f, 0, False)[0]
            return dx
        if R * S == 1 and stride == 1 and padding == 0:
            # pointwise dgrad IS a GEMM: dx[M,C] = g[M,K] @ W[K,C]
            gn = _nhwc(g)
            K = gn.shape[1]
            m = gn.shape[0] * gn.shape[2] * gn.shape[3]
            g2 = gn.permute(0, 2, 3, 1).reshape(m, K)
            dx = ext().linear_dgrad(g2.contiguous(),
                                    w.reshape(w.shape[0], w.shape[1]))
            return dx.view(gn.shape[0], gn.shape[2], gn.shape[3],
                           w.shape[1]).permute(0, 3, 1, 2).contiguous(
                               memory_format=torch.channels_last)
        return ext().conv_dgrad(_nhwc(g), _nhwc(w), stride, padding,
                                x_shape[2], x_shape[3])
    return torch.nn.grad.conv2d_input(x_shape, w, g, stride, padding)


def _conv_wgrad_raw(g, x, stride, padding, w_shape, col=None):
    if use_native(g, x):
        R, S = w_shape[2], w_shape[3]
        if col is not None and R * S > 1 \
                and not _os.environ.get("NOISYNET_WGRAD_NO_COL"):
            # flat im2col matrix shared from the forward pass
            return ext().conv_wgrad_from_col(_nhwc(g), col, x.shape[1], R, S)
        if R * S > 1 and x.element_size() == 2:
            # patch wgrad: x operand gathered in-kernel from the NHWC
            # input -- no materialized im2col pass at all
            return ext().conv_wgrad_patch(_nhwc(g), _nhwc(x), stride,
                                          padding, R, S)
        if R * S == 1 and stride == 1 and padding == 0:
            # pointwise conv wgrad IS a GEMM wgrad: flat NHWC views into
            # the 128-deep pipelined mk kernels (the generic conv wgrad
            # kernel was 2 ms/step of MobileNetV2's expand/project convs)
            gn, xn = _nhwc(g), _nhwc(x)
            K = gn.shape[1]
            C_in = xn.shape[1]
            m = gn.shape[0] * gn.shape[2] * gn.shape[3]
            g2 = gn.permute(0, 2, 3, 1).reshape(m, K)
            x2 = xn.permute(0, 2, 3, 1).reshape(m, C_in)
            dw = ext().linear_wgrad(g2.contiguous(), x2.contiguous())
            return dw.view(K, C_in, 1, 1).contiguous(
                memory_format=torch.channels_last)
        if R * S > 1:
            # fp32: im2col-GEMM wgrad when the buffer is affordable (<2 GB)
            C_in = x.shape[1]
            cols_p = (R * S * C_in if C_in % 8 == 0
                      else (R * S * C_in + 31) // 32 * 32)
            m = g.shape[0] * g.shape[2] * g.shape[3]
            col_bytes = m * cols_p * x.element_size()
            if col_bytes <= 2 << 30:
                return ext().conv_wgrad_im2col(_nhwc(g), _nhwc(x), stride,
                                               padding, R, S)
        return ext().conv_wgrad(_nhwc(g), _nhwc(x), stride, padding, R, S)
    return torch.nn.grad.conv2d_weight(x, w_shape, g, stride, padding)


class ConvFwd(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, bias, stride, padding):
        ctx.stride, ctx.padding = stride, padding
        ctx.save_for_backward(x, w)
        ctx.has_bias = bias is not None
        return _conv_fwd_raw(x, w, bias, stride, padding)

    @staticmethod
    def backward(ctx, g):
        x, w = ctx.saved_tensors
        gx = gw = gb = None
        if ctx.needs_input_grad[0]:
            gx = ConvDgrad.apply(g, w, ctx.stride, ctx.padding, x.shape)
        if ctx.needs_input_grad[1]:
            gw = ConvWgrad.apply(g, x, ctx.stride, ctx.padding, w.shape, None)
        if ctx.has_bias and ctx.needs_input_grad[2]:
            gb = g.sum(dim=(0, 2, 3))
        return gx, gw, gb, None, None


class ConvDgrad(torch.autograd.Function):
    @staticmethod
    def forward(ctx, g, w, stride, padding, x_shape):
        ctx.stride, ctx.padding = stride, padding
        ctx.x_shape = tuple(x_shape)
        ctx.save_for_backward(g, w)
        return _conv_dgrad_raw(g, w, stride, padding, x_shape)

    @staticmethod
    def backward(ctx, gg):
        g, w = ctx.saved_tensors
        d_g = d_w = None
        if ctx.needs_input_grad[0]:
            d_g = ConvFwd.apply(gg, w, None, ctx.stride, ctx.padding)
        if ctx.needs_input_grad[1]:
            d_w = ConvWgrad.apply(g, gg, ctx.stride, ctx.padding, w.shape, None)
        return d_g, d_w, None, None, None


class ConvWgrad(torch.autograd.Function):
    @staticmethod
    def forward(ctx, g, x, stride, padding, w_shape, col=None):
        ctx.stride, ctx.padding = stride, padding
        ctx.w_shape = tuple(w_shape)
        ctx.save_for_backward(g, x)
        return _conv_wgrad_raw(g, x, stride, padding, w_shape, col)

    @staticmethod
    def backward(ctx, gw):
        g, x = ctx.saved_tensors
        d_g = d_x = None
        if ctx.needs_input_grad[0]:
            d_g = ConvFwd.apply(x, gw, None, ctx.stride, ctx.padding)
        if ctx.needs_input_grad[1]:
            d_x = ConvDgrad.apply(g, gw, ctx.stride, ctx.padding, x.shape)
        return d_g, d_x, None, None, None, None


def conv2d(x, w, bias=None, stride=1, padding=0):
    if isinstance(stride, (tuple, list)):
        stride = stride[0]
    if isinstance(padding, (tuple, list)):
        padding = padding[0]
    return ConvFwd.apply(x, w, bias, stride, padding)


# ---------------------------------------------------------------------------
# Fused noisy conv / linear: y = x*Wq (+bias) + N(0, sqrt(factor * x*f(|W|)))
# ---------------------------------------------------------------------------


class NoiseTelemetry:
    """Per-call telemetry matching hardware_model.py:55-88 (first-20-batch
    power / NSR / input-sparsity statistics). Values are 0-dim tensors."""

    __slots__ = ("power", "nsr", "input_sparsity")

    def __init__(self, power=None, nsr=None, input_sparsity=None):
        self.power = power
        self.nsr = nsr
        self.input_sparsity = input_sparsity


class _FusedNoisyConv(torch.autograd.Function):
    """Fused conv + sigma-conv + Gaussian noise.

    Forward computes, in ONE pass over the input tiles on GPU (dual MFMA
    accumulators; csrc/conv_mfma.hip):
        y      = conv(x, wq) [+ bias]
        sigma  = conv(x, f(|w_raw|)),  f per sigma_mode ('abs' | 'abs2')
        out    = y + eps * sqrt(factor * sigma),  eps ~ N(0,1) in-kernel
    Gradients flow through y only (noise is a detached additive term):
    hardware_model.py:23 wraps all noise math in no_grad.
    """

    @staticmethod
    def forward(ctx, x, wq, w_raw, bias, stride, padding, sigma_mode, factor,
                want_telemetry, telemetry_out, current, power_denom):
        ctx.stride, ctx.padding = stride, padding
        ctx.has_bias = bias is not None
        col = None
        if use_native(x, wq):
            y, tele, col = ext().conv_fwd_fused(
                _nhwc(x), _nhwc(wq), _nhwc(w_raw),
                bias if bias is not None else torch.empty(0, device=x.device, dtype=x.dtype),
                stride, padding, 1 if sigma_mode == "abs" else 2,
                _factor_tensor(factor, x.device), _next_seed(x.device),
                bool(want_telemetry))
            if want_telemetry:
                sum_sigma_abs = tele[0] / x.shape[0]
                sum_abs_noise = tele[1] / y.numel()
                max_y = tele[2]
        else:
            y = F.conv2d(x, wq, bias, stride, padding)
            with torch.no_grad():
                sig = ref.sigma_conv2d(x.detach(), w_raw, sigma_mode, stride, padding)
                noise = ref.vmm_noise(sig, factor)
                if want_telemetry:
                    if sigma_mode == "abs":
                        sig_abs = sig
                    else:
                        sig_abs = ref.sigma_conv2d(x.detach(), w_raw, "abs", stride, padding)
                    sum_sigma_abs = sig_abs.sum(dim=(1, 2, 3)).mean()
                    sum_abs_noise = noise.abs().mean()
                    max_y = y.detach().max()
            y = y + noise
        if want_telemetry and telemetry_out is not None:
            with torch.no_grad():
                telemetry_out.power = 1.2e-6 * current * sum_sigma_abs / power_denom
                telemetry_out.nsr = sum_abs_noise / max_y
                telemetry_out.input_sparsity = (x.detach() > 0).sum() / x.numel()
        if col is not None and col.numel():
            # the small-C route materialized the flat im2col matrix; keep
            # it so backward's wgrad skips the re-materialization
            ctx.save_for_backward(x, wq, col)
        else:
            ctx.save_for_backward(x, wq)
        return y

    @staticmethod
    def backward(ctx, g):
        x, wq = ctx.saved_tensors[0], ctx.saved_tensors[1]
        col = ctx.saved_tensors[2] if len(ctx.saved_tensors) > 2 else None
        gx = gw = gb = None
        if ctx.needs_input_grad[0]:
            gx = ConvDgrad.apply(g, wq, ctx.stride, ctx.padding, x.shape)
        if ctx.needs_input_grad[1]:
            gw = ConvWgrad.apply(g, x, ctx.stride, ctx.padding, wq.shape, col)
        if ctx.has_bias and ctx.needs_input_grad[3]:
            gb = g.sum(dim=(0, 2, 3))
        return (gx, gw, None, gb) + (None,) * 8


def fused_noisy_conv2d(x, wq, w_raw, bias, stride, padding, sigma_mode,
                       factor, current=0.0, power_denom=1.0,
                       want_telemetry=False, telemetry_out=None):
    return _FusedNoisyConv.apply(x, wq, w_raw, bias, stride, padding,
                                 sigma_mode, factor, want_telemetry,
                                 telemetry_out, current, power_denom)


# ---- sigma-only variants (noise tensor alone; used by the standalone
# add_noise_calculate_power API path where the clean output already exists) --


@torch.no_grad()
def sigma_noise_conv2d(x, w_raw, sigma_mode, factor, stride=1, padding=0,
                       want_sigma_abs=False):
    """Return (noise, mean-per-sample sum of sigma_abs or None).

    noise ~ N(0, sqrt(factor*sigma)); the second value is
    mean_over_batch(sum_over_outputs sigma_abs) -- the quantity the power
    formula needs (hardware_model.py:55-57). Native path runs the conv
    kernel with the y-accumulator disabled (sigma accumulator + in-kernel
    Philox Gaussian only) and returns the sum from the kernel's telemetry
    reduction.
    """
    if use_native(x, w_raw):
        noise, tele = ext().sigma_noise_conv(
            _nhwc(x), _nhwc(w_raw), stride, padding,
            1 if sigma_mode == "abs" else 2, _factor_tensor(factor, x.device),
            _next_seed(x.device), bool(want_sigma_abs))
        sig_mean = tele[0] / x.shape[0] if want_sigma_abs else None
        return noise, sig_mean
    sig = ref.sigma_conv2d(x, w_raw, sigma_mode, stride, padding)
    noise = ref.vmm_noise(sig, factor)
    sig_mean = None
    if want_sigma_abs:
        sig_abs = sig if sigma_mode == "abs" else ref.sigma_conv2d(x, w_raw, "abs", stride, padding)
        sig_mean = sig_abs.sum(dim=(1, 2, 3)).mean()
    return noise, sig_mean


@torch.no_grad()
def sigma_noise_linear(x, w_raw, sigma_mode, factor, want_sigma_abs=False):
    if use_native(x, w_raw):
        noise, tele = ext().sigma_noise_linear(
            x.contiguous(), w_raw.contiguous(),
            1 if sigma_mode == "abs" else 2, _factor_tensor(factor, x.device),
            _next_seed(x.device), bool(want_sigma_abs))
        sig_mean = tele[0] / x.shape[0] if want_sigma_abs else None
        return noise, sig_mean
    sig = ref.sigma_linear(x, w_raw, sigma_mode)
    noise = ref.vmm_noise(sig, factor)
    sig_mean = None
    if want_sigma_abs:
        sig_abs = sig if sigma_mode == "abs" else ref.sigma_linear(x, w_raw, "abs")
        sig_mean = sig_abs.sum(dim=1).mean()
    return noise, sig_mean


# ---- Linear twins ---------------------------------------------------------


def _linear_fwd_raw(x, w, bias):
    # Plain (un-fused) linear IS a library GEMM: hipBLASLt via F.linear
    # beats our MFMA kernel on the fc shapes (0.023 vs 0.172 ms for
    # 2048x3000 @ 390x3000 bf16 on MI355X). The custom kernel stays for
    # the FUSED noisy path (linear_fwd_fused) where sigma+noise ride the
    # same tiles, and is still exercised directly in tests/test_ops_gpu.
    return F.linear(x, w, bias)


class LinearFwd(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, bias):
        ctx.save_for_backward(x, w)
        ctx.has_bias = bias is not None
        return _linear_fwd_raw(x, w, bias)

    @staticmethod
    def backward(ctx, g):
        x, w = ctx.saved_tensors
        gx = gw = gb = None
        if ctx.needs_input_grad[0]:
            gx = LinearDgrad.apply(g, w)
        if ctx.needs_input_grad[1]:
            gw = LinearWgrad.apply(g, x)
        if ctx.has_bias and ctx.needs_input_grad[2]:
            gb = g.sum(dim=0)
        return gx, gw, gb


class LinearDgrad(torch.autograd.Function):
    """dx = g @ W  (g:[B,O], W:[O,I] -> dx:[B,I])."""

    @staticmethod
    def forward(ctx, g, w):
        ctx.save_for_backward(g, w)
        return g.matmul(w)  # plain GEMM -> hipBLASLt

    @staticmethod
    def backward(ctx, gg):
        g, w = ctx.saved_tensors
        d_g = d_w = None
        if ctx.needs_input_grad[0]:
            d_g = LinearFwd.apply(gg, w, None)
        if ctx.needs_input_grad[1]:
            d_w = LinearWgrad.apply(g, gg)
        return d_g, d_w


class LinearWgrad(torch.autograd.Function):
    """dW = g^T @ x  (g:[B,O], x:[B,I] -> dW:[O,I])."""

    @staticmethod
    def forward(ctx, g, x):
        ctx.save_for_backward(g, x)
        return g.t().matmul(x)  # plain GEMM -> hipBLASLt

    @staticmethod
    def backward(ctx, gw):
        g, x = ctx.saved_tensors
        d_g = d_x = None
        if ctx.needs_input_grad[0]:
            # d/dg of (g^T x) contracted with gw: x @ gw^T = linear(x, gw)
            d_g = LinearFwd.apply(x, gw, None)
        if ctx.needs_input_grad[1]:
            d_x = LinearDgrad.apply(g, gw)
        return d_g, d_x


def linear(x, w, bias=None):
    return LinearFwd.apply(x, w, bias)


class _FusedNoisyLinear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, wq, w_raw, bias, sigma_mode, factor,
                want_telemetry, telemetry_out, current, power_denom):
        ctx.save_for_backward(x, wq)
        ctx.has_bias = bias is not None
        if use_native(x, wq):
            y, tele = ext().linear_fwd_fused(
                x.contiguous(), wq.contiguous(), w_raw.contiguous(),
                bias if bias is not None else torch.empty(0, device=x.device, dtype=x.dtype),
                1 if sigma_mode == "abs" else 2,
                _factor_tensor(factor, x.device),
                _next_seed(x.device), bool(want_telemetry))
            if want_telemetry:
                sum_sigma_abs = tele[0] / x.shape[0]
                sum_abs_noise = tele[1] / y.numel()
                max_y = tele[2]
        else:
            y = F.linear(x, wq, bias)
            with torch.no_grad():
                sig = ref.sigma_linear(x.detach(), w_raw, sigma_mode)
                noise = ref.vmm_noise(sig, factor)
                if want_telemetry:
                    sig_abs = sig if sigma_mode == "abs" else ref.sigma_linear(x.detach(), w_raw, "abs")
                    sum_sigma_abs = sig_abs.sum(dim=1).mean()
                    sum_abs_noise = noise.abs().mean()
                    max_y = y.detach().max()
            y = y + noise
        if want_telemetry and telemetry_out is not None:
            with torch.no_grad():
                telemetry_out.power = 1.2e-6 * current * sum_sigma_abs / power_denom
                telemetry_out.nsr = sum_abs_noise / max_y
                telemetry_out.input_sparsity = (x.detach() > 0).sum() / x.numel()
        return y

    @staticmethod
    def backward(ctx, g):
        x, wq = ctx.saved_tensors
        gx = gw = gb = None
        if ctx.needs_input_grad[0]:
            gx = LinearDgrad.apply(g, wq)
        if ctx.needs_input_grad[1]:
            gw = LinearWgrad.apply(g, x)
        if ctx.has_bias and ctx.needs_input_grad[3]:
            gb = g.sum(dim=0)
        return (gx, gw, None, gb) + (None,) * 6


def fused_noisy_linear(x, wq, w_raw, bias, sigma_mode, factor, current=0.0,
                       power_denom=1.0, want_telemetry=False, telemetry_out=None):
    return _FusedNoisyLinear.apply(x, wq, w_raw, bias, sigma_mode, factor,
                                   want_telemetry, telemetry_out, current,
                                   power_denom)


# ---------------------------------------------------------------------------
# Simplified noise modes (uniform_ind / uniform_dep / normal_ind / normal_dep)
# hardware_model.py:24-41 -- train-time or --noise_test perturbations.
# ---------------------------------------------------------------------------


def simple_noise(output, mode, a):
    """Returns the noise tensor (detached). Modes per hardware_model.py:24-41."""
    with torch.no_grad():
        out = output.detach()
        if mode == "uniform_ind":
            amp = a * out.abs().max()
            return torch.empty_like(out).uniform_(-1, 1) * amp
        if mode == "uniform_dep":
            lo = torch.full_like(out, a)
            hi = torch.full_like(out, 1.0 / a)
            return lo + (hi - lo) * torch.rand_like(out)
        if mode == "normal_ind":
            s = a * out.abs().max()
            return torch.randn_like(out) * s
        if mode == "normal_dep":
            return torch.randn_like(out) * (a * out)  # signed scale, as reference
        raise ValueError(mode)


# ---------------------------------------------------------------------------
# Elementwise weight perturbation (AddNoise STE)
# ---------------------------------------------------------------------------


class MultUniformNoise(torch.autograd.Function):
    """out = w + w*U(-a,a); identity STE backward (hardware_model.py:291-307)."""

    @staticmethod
    def forward(ctx, w, a):
        if use_native(w):
            return ext().mult_uniform_noise(w, float(a), _next_seed(w.device))
        return ref.mult_uniform_noise(w, a)

    @staticmethod
    def backward(ctx, g):
        return g, None


def add_weight_noise(w, a):
    return MultUniformNoise.apply(w, a)


# ---------------------------------------------------------------------------
# Fused BatchNorm + ReLU + clip
# ---------------------------------------------------------------------------


class BnAct(torch.autograd.Function):
    """BN (train: batch stats; eval: running stats) fused with ReLU and an
    optional upper clip. Backward folds the activation mask into the BN
    gradient. Native path: csrc/bn_act.hip (per-channel partials + one fused
    normalize/act pass).

    ``sync=True`` is the fused SyncBN (main.py:786-796 equivalent): the
    per-rank (mean, E[x^2]) pairs are combined with ONE RCCL all-reduce
    before normalization, and the backward's per-channel sums are likewise
    all-reduced before the apply pass."""

    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var, training,
                momentum, eps, relu, act_max, sync=False):
        import torch.distributed as dist
        ws = dist.get_world_size() if (sync and dist.is_initialized()) else 1
        ctx.sync_ws = ws if training else 1
        if (training and ws == 1 and use_native(x)
                and (running_mean is None
                     or running_mean.dtype == torch.float32)):
            # fused single-rank path: stats + invstd + running update in
            # two kernels (the eager chain was ~8 tiny launches per layer)
            xc = _nhwc(x) if x.dim() == 4 else x.contiguous()
            empty = torch.empty(0, device=x.device, dtype=torch.float32)
            mean, invstd = ext().bn_stats_finalize(
                xc,
                running_mean if running_mean is not None else empty,
                running_var if running_var is not None else empty,
                float(momentum), float(eps))
            if use_native(x):
                y = ext().bn_act_fwd(xc, mean, invstd,
                                     weight.float().contiguous(),
                                     bias.float().contiguous(),
                                     bool(relu), float(act_max))
            ctx.save_for_backward(x, weight, mean, invstd, y)
            ctx.training = training
            ctx.relu, ctx.act_max = relu, act_max
            return y
        if training:
            if use_native(x):
                mean, var = ext().bn_stats(_nhwc(x) if x.dim() == 4 else x.contiguous())
            else:
                mean, var = ref.bn_stats(x)
            if ws > 1:
                stats = torch.stack([mean, var + mean * mean])
                dist.all_reduce(stats)
                stats /= ws
                mean = stats[0]
                var = (stats[1] - mean * mean).clamp_min(0)
            if running_mean is not None:
                with torch.no_grad():
                    n = x.numel() / x.shape[1] * ws
                    unbiased = var * (n / max(n - 1.0, 1.0))
                    running_mean.mul_(1 - momentum).add_(momentum * mean.to(running_mean.dtype))
                    running_var.mul_(1 - momentum).add_(momentum * unbiased.to(running_var.dtype))
        else:
            mean = running_mean
            var = running_var
        mean = mean.float()
        invstd = (var.float() + eps).rsqrt()
        if use_native(x):
            y = ext().bn_act_fwd(_nhwc(x) if x.dim() == 4 else x.contiguous(),
                                 mean.contiguous(), invstd.contiguous(),
                                 weight.float().contiguous(),
                                 bias.float().contiguous(),
                                 bool(relu), float(act_max))
        else:
            y = ref.bn_act_forward(x.float(), weight.float(), bias.float(),
                                   mean, invstd, act_max, relu).to(x.dtype)
        ctx.save_for_backward(x, weight, mean, invstd, y)
        ctx.training = training
        ctx.relu, ctx.act_max = relu, act_max
        return y

    @staticmethod
    def backward(ctx, g):
        import torch.distributed as dist
        x, weight, mean, invstd, y = ctx.saved_tensors
        ws = getattr(ctx, 'sync_ws', 1)
        if use_native(x):
            gc = _nhwc(g) if g.dim() == 4 else g.contiguous()
            xc = _nhwc(x) if x.dim() == 4 else x.contiguous()
            yc = _nhwc(y) if y.dim() == 4 else y.contiguous()
            if ws > 1:
                sum_g_loc, sum_gx_loc = ext().bn_act_bwd_reduce(
                    gc, xc, yc, mean.contiguous(), invstd.contiguous(),
                    bool(ctx.relu), float(ctx.act_max))
                pair = torch.stack([sum_g_loc, sum_gx_loc])
                dist.all_reduce(pair)
                sum_g, sum_gx = pair[0].contiguous(), pair[1].contiguous()
                count = x.numel() / x.shape[1] * ws
                gx = ext().bn_act_bwd_apply(
                    gc, xc, yc, mean.contiguous(), invstd.contiguous(),
                    weight.float().contiguous(), sum_g, sum_gx, float(count),
                    bool(ctx.training), bool(ctx.relu), float(ctx.act_max))
                # gamma/beta grads stay LOCAL sums: the data-parallel bucket
                # all-reduce averages them like torch SyncBatchNorm expects
                return (gx, sum_gx_loc.to(weight.dtype),
                        sum_g_loc.to(weight.dtype)) + (None,) * 8
            gx, g_gamma, g_beta = ext().bn_act_bwd(
                gc, xc, yc, mean.contiguous(), invstd.contiguous(),
                weight.float().contiguous(), bool(ctx.training),
                bool(ctx.relu), float(ctx.act_max))
            return (gx, g_gamma.to(weight.dtype),
                    g_beta.to(weight.dtype)) + (None,) * 8
        shape = (1, -1, 1, 1) if x.dim() == 4 else (1, -1)
        # activation mask: dy/dz = 1 where 0 < y (< act_max if clipped)
        mask = torch.ones_like(y)
        if ctx.relu:
            mask = mask * (y > 0).to(g.dtype)
        if ctx.act_max > 0:
            mask = mask * (y < ctx.act_max).to(g.dtype)
        g = (g * mask).float()
        dims = (0, 2, 3) if x.dim() == 4 else (0,)
        xhat = (x.float() - mean.view(shape)) * invstd.view(shape)
        g_gamma = (g * xhat).sum(dims)
        g_beta = g.sum(dims)
        if ws > 1:
            pair = torch.stack([g_beta, g_gamma])
            dist.all_reduce(pair)
            g_beta_r, g_gamma_r = pair[0], pair[1]
        else:
            g_beta_r, g_gamma_r = g_beta, g_gamma
        wf = weight.float()
        if ctx.training:
            n = x.numel() / x.shape[1] * ws
            gx = (wf.view(shape) * invstd.view(shape)) * (
                g - g_beta_r.view(shape) / n - xhat * g_gamma_r.view(shape) / n)
        else:
            gx = g * (wf.view(shape) * invstd.view(shape))
        return (gx.to(x.dtype), g_gamma.to(weight.dtype),
                g_beta.to(weight.dtype)) + (None,) * 8


def bn_act(x, weight, bias, running_mean, running_var, training, momentum,
           eps, relu=True, act_max=0.0, sync=False):
    return BnAct.apply(x, weight, bias, running_mean, running_var, training,
                       momentum, eps, relu, act_max, sync)


# ---------------------------------------------------------------------------
# Pooling
# ---------------------------------------------------------------------------


class MaxPool2x2(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        if use_native(x):
            y, idx = ext().maxpool2x2_fwd(_nhwc(x))
        else:
            y, idx = F.max_pool2d(x, 2, 2, return_indices=True)
        ctx.save_for_backward(idx)
        ctx.x_shape = x.shape
        return y

    @staticmethod
    def backward(ctx, g):
        (idx,) = ctx.saved_tensors
        if use_native(g):
            gx = ext().maxpool2x2_bwd(_nhwc(g), idx, ctx.x_shape[2], ctx.x_shape[3])
        else:
            gx = F.max_unpool2d(g, idx, 2, 2, output_size=ctx.x_shape[2:])
        return gx


def maxpool2x2(x):
    return MaxPool2x2.apply(x)


class MaxPoolNHWC(torch.autograd.Function):
    """Generic k x k / stride / pad max pool (ResNet 3x3 s2 stem etc.)."""

    @staticmethod
    def forward(ctx, x, k, stride, pad):
        if use_native(x):
            y, code = ext().maxpool_fwd(_nhwc(x), k, stride, pad)
        else:
            y, code = F.max_pool2d(x, k, stride, pad, return_indices=True)
        ctx.save_for_backward(code)
        ctx.meta = (x.shape, k, stride, pad)
        return y

    @staticmethod
    def backward(ctx, g):
        (code,) = ctx.saved_tensors
        x_shape, k, stride, pad = ctx.meta
        if use_native(g):
            gx = ext().maxpool_bwd(_nhwc(g), code, x_shape[2], x_shape[3],
                                   k, stride, pad)
        else:
            gx = F.max_unpool2d(g, code, k, stride, pad,
                                output_size=x_shape[2:])
        return gx, None, None, None


def maxpool_nhwc(x, k, stride, pad=0):
    return MaxPoolNHWC.apply(x, k, stride, pad)


class AvgPoolNHWC(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, k, stride, pad):
        ctx.meta = (x.shape, k, stride, pad)
        if use_native(x):
            return ext().avgpool_fwd(_nhwc(x), k, stride, pad)
        return F.avg_pool2d(x, k, stride, pad)

    @staticmethod
    def backward(ctx, g):
        x_shape, k, stride, pad = ctx.meta
        if use_native(g):
            gx = ext().avgpool_bwd(_nhwc(g), x_shape[2], x_shape[3], k,
                                   stride, pad)
        else:
            # distribute g/(k*k) over each window via conv_transpose of ones
            w = torch.ones(x_shape[1], 1, k, k, device=g.device,
                           dtype=g.dtype) / (k * k)
            gx = F.conv_transpose2d(g, w, None, stride, pad,
                                    groups=x_shape[1],
                                    output_padding=0)
            # pad to original size if needed
            if gx.shape[2] != x_shape[2] or gx.shape[3] != x_shape[3]:
                gx = F.pad(gx, (0, x_shape[3] - gx.shape[3],
                                0, x_shape[2] - gx.shape[2]))
        return gx, None, None, None


def avgpool_nhwc(x, k, stride, pad=0):
    return AvgPoolNHWC.apply(x, k, stride, pad)


# ---------------------------------------------------------------------------
# Depthwise conv (groups == channels): MobileNetV2 / EfficientNet dw layers
# ---------------------------------------------------------------------------


class DwConv2d(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, bias, stride, pad):
        ctx.stride, ctx.pad = stride, pad
        ctx.save_for_backward(x, w)
        ctx.has_bias = bias is not None
        if use_native(x, w):
            return ext().dwconv_fwd(
                _nhwc(x), w.contiguous(),
                bias if bias is not None else torch.empty(0, device=x.device,
                                                          dtype=x.dtype),
                stride, pad)
        return F.conv2d(x, w, bias, stride, pad, 1, x.shape[1])

    @staticmethod
    def backward(ctx, g):
        x, w = ctx.saved_tensors
        gx = gw = gb = None
        if use_native(g):
            if ctx.needs_input_grad[0]:
                gx = ext().dwconv_dgrad(_nhwc(g), w.contiguous(), ctx.stride,
                                        ctx.pad, x.shape[2], x.shape[3])
            if ctx.needs_input_grad[1]:
                gw = ext().dwconv_wgrad(_nhwc(g), _nhwc(x), ctx.stride,
                                        ctx.pad, w.shape[2], w.shape[3])
        else:
            if ctx.needs_input_grad[0]:
                gx = torch.nn.grad.conv2d_input(x.shape, w, g, ctx.stride,
                                                ctx.pad, 1, x.shape[1])
            if ctx.needs_input_grad[1]:
                gw = torch.nn.grad.conv2d_weight(x, w.shape, g, ctx.stride,
                                                 ctx.pad, 1, x.shape[1])
        if ctx.has_bias and ctx.needs_input_grad[2]:
            gb = g.sum(dim=(0, 2, 3))
        return gx, gw, gb, None, None


def depthwise_conv2d(x, w, bias=None, stride=1, padding=0):
    if isinstance(stride, (tuple, list)):
        stride = stride[0]
    if isinstance(padding, (tuple, list)):
        padding = padding[0]
    return DwConv2d.apply(x, w, bias, stride, padding)


# ---------------------------------------------------------------------------
# EfficientNet-family activations (memory-efficient analytic backward)
# ---------------------------------------------------------------------------

_ACT_IDS = {"swish": 0, "mish": 1, "hardswish": 2, "hardsigmoid": 3,
            "sigmoid": 4}


def _act_ref_fwd(x, act):
    if act == "swish":
        return x * torch.sigmoid(x)
    if act == "mish":
        return x * F.softplus(x).tanh()
    if act == "hardswish":
        return x * F.relu6(x + 3.0) / 6.0
    if act == "hardsigmoid":
        return F.relu6(x + 3.0) / 6.0
    return torch.sigmoid(x)


def _act_ref_bwd(g, x, act):
    if act == "swish":
        s = torch.sigmoid(x)
        return g * (s * (1 + x * (1 - s)))
    if act == "mish":
        sp = F.softplus(x)
        tsp = sp.tanh()
        s = torch.sigmoid(x)
        return g * (tsp + x * s * (1 - tsp * tsp))
    if act == "hardswish":
        d = torch.where(x <= -3.0, torch.zeros_like(x),
                        torch.where(x >= 3.0, torch.ones_like(x),
                                    (2 * x + 3.0) / 6.0))
        return g * d
    if act == "hardsigmoid":
        d = ((x > -3.0) & (x < 3.0)).to(g.dtype) / 6.0
        return g * d
    s = torch.sigmoid(x)
    return g * s * (1 - s)


class ActFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, act):
        ctx.act = act
        ctx.save_for_backward(x)
        if use_native(x):
            return ext().act_fwd(x, _ACT_IDS[act])
        return _act_ref_fwd(x, act)

    @staticmethod
    def backward(ctx, g):
        (x,) = ctx.saved_tensors
        if use_native(x):
            # the C++ wrapper aligns g to x's memory format; a plain
            # .contiguous() here forced an NCHW round-trip of every
            # channels_last activation grad (5.3 ms/step on EffNet-B0)
            return ext().act_bwd(g, x, _ACT_IDS[ctx.act]), None
        return _act_ref_bwd(g, x, ctx.act), None


def swish(x):
    return ActFn.apply(x, "swish")


def mish(x):
    return ActFn.apply(x, "mish")


def hard_swish(x):
    return ActFn.apply(x, "hardswish")


def hard_sigmoid(x):
    return ActFn.apply(x, "hardsigmoid")


def sigmoid(x):
    return ActFn.apply(x, "sigmoid")


# ---------------------------------------------------------------------------
# ReLU + clip (standalone, used where BN is off / merged)
# ---------------------------------------------------------------------------


class ReluClip(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, act_max, relu):
        if use_native(x):
            y = ext().relu_clip_fwd(x, bool(relu), float(act_max))
        else:
            y = F.relu(x) if relu else x
            if act_max > 0:
                y = y.clamp(max=act_max)
        ctx.save_for_backward(y)
        ctx.relu, ctx.act_max = relu, act_max
        return y

    @staticmethod
    def backward(ctx, g):
        (y,) = ctx.saved_tensors
        if use_native(y):
            # wrapper aligns g to y's memory format (no NCHW round-trip)
            return (ext().relu_clip_bwd(g, y, bool(ctx.relu),
                                        float(ctx.act_max)), None, None)
        mask = torch.ones_like(y)
        if ctx.relu:
            mask = mask * (y > 0).to(g.dtype)
        if ctx.act_max > 0:
            mask = mask * (y < ctx.act_max).to(g.dtype)
        return g * mask, None, None


def relu_clip(x, act_max=0.0, relu=True):
    return ReluClip.apply(x, act_max, relu)


# ---------------------------------------------------------------------------
# Dropout (Philox in-kernel on GPU)
# ---------------------------------------------------------------------------


class Dropout(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, p):
        if use_native(x):
            y, mask = ext().dropout_fwd(x, float(p), _next_seed(x.device))
        else:
            mask = (torch.rand_like(x) >= p).to(x.dtype) / (1.0 - p)
            y = x * mask
        ctx.save_for_backward(mask)
        return y

    @staticmethod
    def backward(ctx, g):
        (mask,) = ctx.saved_tensors
        return g * mask, None


def dropout(x, p, training):
    if not training or p <= 0:
        return x
    return Dropout.apply(x, p)


# ---------------------------------------------------------------------------
# Softmax cross-entropy (fused on GPU)
# ---------------------------------------------------------------------------


class SoftmaxXent(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, target):
        if use_native(logits):
            loss, softmax = ext().softmax_xent_fwd(logits.contiguous(), target.contiguous())
        else:
            logp = F.log_softmax(logits.float(), dim=1)
            loss = F.nll_loss(logp, target)
            softmax = logp.exp().to(logits.dtype)
        ctx.save_for_backward(softmax, target)
        return loss

    @staticmethod
    def backward(ctx, g):
        softmax, target = ctx.saved_tensors
        n, c = softmax.shape
        if use_native(softmax):
            if isinstance(g, torch.Tensor) and g.is_cuda:
                # device-resident scale: no host sync (graph-capturable)
                gx = ext().softmax_xent_bwd_t(
                    softmax, target, g.detach().to(torch.float32).reshape(1))
            else:
                gx = ext().softmax_xent_bwd(softmax, target, float(g))
        else:
            onehot = F.one_hot(target, c).to(softmax.dtype)
            gx = (softmax - onehot) * (g / n)
        return gx, None


def cross_entropy(logits, target):
    return SoftmaxXent.apply(logits, target)


# ---------------------------------------------------------------------------
# Fused optimizer steps (no_grad; clamp folded in)
# ---------------------------------------------------------------------------


def _grad_like(param, grad):
    """Raw-layout-match the gradient to the parameter (channels_last conv
    weights on GPU vs standard-contiguous grads, or vice versa)."""
    if param.dim() == 4 and param.is_contiguous(memory_format=torch.channels_last) \
            and not param.is_contiguous():
        return grad.contiguous(memory_format=torch.channels_last)
    return grad.contiguous()


@torch.no_grad()
def sgd_step(param, grad, momentum_buf, lr, momentum, weight_decay, nesterov,
             clamp_min=0.0, clamp_max=0.0):
    """SGD with L2, momentum, nesterov, and post-step weight clamping fused.

    Mirrors torch.optim.SGD semantics + the post-step p.clamp_(-w_max,w_max)
    of noisynet.py:1527-1542.
    """
    if use_native(param):
        ext().sgd_step(param, _grad_like(param, grad), momentum_buf,
                       float(lr), float(momentum),
                       float(weight_decay), bool(nesterov), float(clamp_min),
                       float(clamp_max))
        return
    g = grad
    if weight_decay != 0:
        g = g + weight_decay * param
    if momentum != 0:
        momentum_buf.mul_(momentum).add_(g)
        g = g + momentum * momentum_buf if nesterov else momentum_buf
    param.add_(g, alpha=-lr)
    if clamp_max > clamp_min:
        param.clamp_(clamp_min, clamp_max)


@torch.no_grad()
def adamw_step(param, grad, exp_avg, exp_avg_sq, step, lr, beta1, beta2, eps,
               weight_decay, clamp_min=0.0, clamp_max=0.0):
    """AdamW (decoupled weight decay) + fused post-step clamp."""
    if use_native(param):
        ext().adamw_step(param, _grad_like(param, grad), exp_avg, exp_avg_sq,
                         int(step), float(lr), float(beta1), float(beta2),
                         float(eps), float(weight_decay), float(clamp_min),
                         float(clamp_max))
        return
    param.mul_(1 - lr * weight_decay)
    exp_avg.mul_(beta1).add_(grad, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    denom = (exp_avg_sq / bc2).sqrt().add_(eps)
    param.addcdiv_(exp_avg, denom, value=-lr / bc1)
    if clamp_max > clamp_min:
        param.clamp_(clamp_min, clamp_max)


# ---------------------------------------------------------------------------
# Percentile calibration
# ---------------------------------------------------------------------------


def kth_percentile(x, pctl):
    if use_native(x):
        return ext().kth_percentile(x.contiguous().view(-1), float(pctl))
    return ref.kth_percentile(x, pctl)


# ---------------------------------------------------------------------------
# Per-sample (conditionally-parameterized) grouped conv
# ---------------------------------------------------------------------------


def per_sample_conv2d(x, weight, bias=None, stride=1, padding=0, groups=1):
    """Convolution where every sample has its OWN filter bank (CondConv,
    reference conv2d_layers.py:152-240).

    The reference runs this as one grouped conv with batch*groups groups --
    a shape cuDNN special-cases but that maps terribly to grouped-conv
    kernels. Here it is unfold + ONE batched GEMM: im2col columns
    [B, g, L, C/g*R*S] contracted against the per-sample filters
    [B, g, C/g*R*S, K/g] via a single rocBLAS strided-batched GEMM (plain
    library GEMMs are the one sanctioned library path). Fully
    differentiable (double-backward included) through torch autograd.

    x: [B, C, H, W]; weight: [B*K, C/g, R, S] (per-sample filters, sample-
    major as CondConv produces them); returns [B, K, OH, OW].
    """
    B, C, H, W = x.shape
    BK, Cg, R, S = weight.shape
    K = BK // B
    st = (stride, stride) if isinstance(stride, int) else tuple(stride)
    pd = (padding, padding) if isinstance(padding, int) else tuple(padding)
    OH = (H + 2 * pd[0] - R) // st[0] + 1
    OW = (W + 2 * pd[1] - S) // st[1] + 1
    cols = F.unfold(x, (R, S), padding=pd, stride=st)  # [B, C*R*S, L]
    L = cols.shape[-1]
    g = groups
    # columns laid out [g, C/g*R*S] along dim 1
    cols = cols.view(B, g, Cg * R * S, L)
    w = weight.view(B, g, K // g, Cg * R * S)
    out = torch.matmul(w, cols)                        # [B, g, K/g, L]
    out = out.reshape(B, K, OH, OW)
    if bias is not None:
        out = out + bias.view(B, K, 1, 1)
    return out


# ---------------------------------------------------------------------------
# Squeeze-Excite gating (fused broadcast-scale + per-(b,c) reduce backward)
# ---------------------------------------------------------------------------


_SE_GATES = {'sigmoid': 4, 'hard_sigmoid': 3}


class SeScale(torch.autograd.Function):
    """y = x * gate(s) with per-(batch, channel) logits s [B,C,1,1].

    Replaces the eager broadcast multiply + backward reduce of the SE
    block (reference efficientnet SqueezeExcite, models/efficientnet.py:
    449-466) with one fused pass each way."""

    @staticmethod
    def forward(ctx, x, s, gate_code):
        ctx.gate_code = gate_code
        ctx.save_for_backward(x, s)
        return ext().se_scale_fwd(x, s, gate_code)

    @staticmethod
    def backward(ctx, g):
        x, s = ctx.saved_tensors
        gx, gs = ext().se_scale_bwd(g, x, s, ctx.gate_code)
        return gx, gs.view_as(s), None


def se_scale(x, s, gate='sigmoid'):
    if use_native(x, s) and x.dim() == 4:
        return SeScale.apply(_nhwc(x), s, _SE_GATES[gate])
    gate_fn = torch.sigmoid if gate == 'sigmoid' \
        else lambda t: F.hardsigmoid(t)
    return x * gate_fn(s)
