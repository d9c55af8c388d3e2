"""Pure-PyTorch reference implementations of every native op.

These are the numerics oracles for the HIP kernels (tests compare the gfx950
kernels against these in fp32) and the execution path on CPU-only machines.

Semantics follow the reference framework exactly; each function cites the
behaviour it reproduces (file:line into /root/reference/).
"""

import torch
import torch.nn.functional as F

# ---------------------------------------------------------------------------
# Fake quantization (UniformQuantize chain)
# ---------------------------------------------------------------------------


def fake_quant_forward(x, num_bits, min_value, max_value, stochastic, gen=None):
    """Clamp-quantize-dequantize with optional stochastic rounding.

    Reproduces the exact in-place chain of hardware_model.py:148-170:
      scale = max((max-min)/(2^b-1), 1e-6)
      q = (x - min)/scale  [+ qmin=0]
      q += U(-stochastic, stochastic)   (if stochastic > 0)
      q = round(clamp(q, 0, 2^b-1))
      out = q*scale + min
    Note: the input is NOT pre-clamped; clamping happens in the quantized
    domain, which is equivalent for the deterministic path.
    """
    qmax = 2.0 ** num_bits - 1.0
    scale = max((max_value - min_value) / qmax, 1e-6)
    q = (x - min_value) / scale
    if stochastic > 0:
        noise = torch.empty_like(q)
        if gen is not None:
            noise.uniform_(-stochastic, stochastic, generator=gen)
        else:
            noise.uniform_(-stochastic, stochastic)
        q = q + noise
    q = q.clamp(0.0, qmax).round()
    return q * scale + min_value


def fake_quant_backward(grad_output, x, min_value, max_value):
    """Saturated STE: zero gradient outside [min_value, max_value].

    hardware_model.py:175-183 (strict inequalities).
    """
    mask = (x <= max_value) & (x >= min_value)
    return grad_output * mask.to(grad_output.dtype)


# ---------------------------------------------------------------------------
# Sigma convolutions / GEMMs for the analog noise model
# ---------------------------------------------------------------------------


def sigma_conv2d(x, weight, mode, stride=1, padding=0):
    """Second conv over transformed |W| used for per-pixel noise variance.

    mode='abs'  -> conv2d(x, |W|)          (hardware_model.py:49)
    mode='abs2' -> conv2d(x, |W|^2 + |W|)  (hardware_model.py:62-65)
    """
    aw = weight.abs()
    if mode == "abs2":
        aw = aw * aw + aw
    return F.conv2d(x, aw, None, stride, padding)


def sigma_linear(x, weight, mode):
    aw = weight.abs()
    if mode == "abs2":
        aw = aw * aw + aw
    return F.linear(x, aw, None)


def vmm_noise(sigmas, factor, gen=None):
    """Gaussian sample with per-element std sqrt(factor * sigmas).

    hardware_model.py:59,81: Normal(0, sqrt(0.1*(w_max/I)*sigmas)).sample().
    ``factor`` is the scalar 0.1*w_max/I (merged DAC) or 0.1*input_max/I.
    """
    s = (factor * sigmas).clamp_min(0).sqrt()
    n = torch.randn(sigmas.shape, device=sigmas.device, dtype=sigmas.dtype, generator=gen)
    return n * s


# ---------------------------------------------------------------------------
# Fused BN + activation (reference composition)
# ---------------------------------------------------------------------------


def bn_stats(x):
    """Per-channel biased mean/var over (N, H, W) for NCHW or (N,) for NC."""
    if x.dim() == 4:
        dims = (0, 2, 3)
    else:
        dims = (0,)
    mean = x.mean(dim=dims)
    var = x.var(dim=dims, unbiased=False)
    return mean, var


def bn_act_forward(x, weight, bias, mean, invstd, act_max=0.0, relu=True):
    """y = gamma*(x-mean)*invstd+beta, then ReLU, then clamp(<=act_max)."""
    shape = (1, -1, 1, 1) if x.dim() == 4 else (1, -1)
    y = (x - mean.view(shape)) * invstd.view(shape)
    if weight is not None:
        y = y * weight.view(shape)
    if bias is not None:
        y = y + bias.view(shape)
    if relu:
        y = F.relu(y)
    if act_max > 0:
        y = y.clamp(max=act_max)
    return y


# ---------------------------------------------------------------------------
# Elementwise perturbations
# ---------------------------------------------------------------------------


def mult_uniform_noise(x, noise, gen=None):
    """out = x + x*U(-noise, +noise): AddNoise fwd (hardware_model.py:293-301)
    and distort_weights (main.py:372-377)."""
    u = torch.empty_like(x)
    if gen is not None:
        u.uniform_(-noise, noise, generator=gen)
    else:
        u.uniform_(-noise, noise)
    return x + x * u


# ---------------------------------------------------------------------------
# Percentile (kthvalue) calibration
# ---------------------------------------------------------------------------


def kth_percentile(x, pctl):
    """k-th order statistic matching torch.kthvalue semantics used at
    hardware_model.py:233-249: k = int(numel * pctl / 100)."""
    flat = x.flatten()
    k = int(flat.numel() * pctl / 100.0)
    k = max(1, min(k, flat.numel()))
    val, _ = torch.kthvalue(flat.float(), k)
    return val.to(x.dtype)


# ---------------------------------------------------------------------------
# Pooling / loss (reference path = plain torch)
# ---------------------------------------------------------------------------


def maxpool2x2(x):
    return F.max_pool2d(x, 2, 2)


def softmax_xent(logits, target):
    return F.cross_entropy(logits, target)
