"""Tests for the analog noise/power model (SURVEY.md §2.3 math)."""

import numpy as np
import pytest
import torch
import torch.nn.functional as F

from noisynet_amd import ops
from noisynet_amd.hardware_model import (AddNoise, NoisyConv2d, NoisyLinear,
                                         add_noise_calculate_power)
from noisynet_amd.ops import reference as ref


def test_sigma_conv_matches_two_conv():
    torch.manual_seed(0)
    x = torch.rand(4, 3, 16, 16)
    w = torch.randn(8, 3, 5, 5) * 0.3
    sig = ref.sigma_conv2d(x, w, "abs")
    expect = F.conv2d(x, w.abs())
    assert torch.allclose(sig, expect, atol=1e-6)
    sig2 = ref.sigma_conv2d(x, w, "abs2")
    expect2 = F.conv2d(x, w.abs() ** 2 + w.abs())
    assert torch.allclose(sig2, expect2, atol=1e-5)


def test_vmm_noise_statistics():
    """Noise ~ N(0, sqrt(factor*sigma)): check var against the formula."""
    torch.manual_seed(0)
    sigmas = torch.full((500, 500), 2.0)
    factor = 0.05
    noise = ref.vmm_noise(sigmas, factor)
    assert abs(noise.mean().item()) < 2e-3
    expected_var = factor * 2.0
    assert abs(noise.var().item() - expected_var) / expected_var < 0.02


def test_add_noise_calculate_power_merged_dac():
    """Full merged-DAC path: telemetry power formula and clean-grad flow."""
    torch.manual_seed(0)

    class M(torch.nn.Module):
        pass

    model = M()
    model.train()
    model.power = [[] for _ in range(4)]
    model.nsr = [[] for _ in range(4)]
    model.input_sparsity = [[] for _ in range(4)]

    class A:
        distort_act = False
        uniform_ind = uniform_dep = normal_ind = normal_dep = 0.0
        noise_test = False
        layer_currents = [1.0, 1.0, 1.0, 1.0]
        noise = 0.0

    x = torch.rand(8, 3, 16, 16)
    w = torch.randn(8, 3, 5, 5, requires_grad=True) * 0.2
    w.retain_grad()
    y = F.conv2d(x, w)
    out = add_noise_calculate_power(model, A, [], x, w, y, layer_type='conv',
                                    i=0, layer_num=0, merged_dac=True)
    assert out.shape == y.shape
    # telemetry got recorded
    assert len(model.power[0]) == 1 and len(model.nsr[0]) == 1
    # power formula: 1.2e-6 * I * mean(sum sigmas) / (max(x) * max|w|)
    sig = F.conv2d(x, w.detach().abs())
    p_expect = 1.2e-6 * 1.0 * sig.sum(dim=(1, 2, 3)).mean() / (x.max() * w.detach().abs().max())
    assert abs(model.power[0][0] - float(p_expect)) < 1e-9
    # gradient flows through the clean output only (noise is additive const)
    out.sum().backward()
    assert w.grad is not None


def test_noise_scales_with_current():
    """Lower current => more noise (std ~ 1/sqrt(I))."""
    torch.manual_seed(0)
    x = torch.rand(16, 3, 16, 16)
    w = torch.randn(8, 3, 5, 5) * 0.2
    noises = {}
    for current in (1.0, 100.0):
        sig = ref.sigma_conv2d(x, w, "abs")
        factor = 0.1 * w.abs().max() / current
        noises[current] = ref.vmm_noise(sig, float(factor)).std().item()
    ratio = noises[1.0] / noises[100.0]
    assert 8 < ratio < 12  # sqrt(100) = 10


def test_addnoise_ste():
    w = torch.randn(50, 50, requires_grad=True)
    out = AddNoise().apply(w, 0.1, False)
    # multiplicative bound: |out - w| <= 0.1*|w|
    assert ((out - w).abs() <= 0.1 * w.abs() + 1e-6).all()
    out.sum().backward()
    assert torch.equal(w.grad, torch.ones_like(w))


def test_noisy_conv_weight_quant_path():
    torch.manual_seed(0)
    conv = NoisyConv2d(3, 8, kernel_size=3, num_bits=0, num_bits_weight=4,
                       noise=0, stochastic=0)
    conv.train()
    x = torch.rand(2, 3, 8, 8)
    y = conv(x)
    # quantized weights on the 4-bit grid in [-1, 1]
    wq, _ = conv.effective_weight()
    scale = 2.0 / 15
    grid = ((wq - (-1.0)) / scale).round()
    assert torch.allclose(wq, grid * scale - 1.0, atol=1e-6)
    assert y.shape == (2, 8, 6, 6)


def test_noisy_linear_train_noise():
    torch.manual_seed(0)
    lin = NoisyLinear(20, 10, num_bits=0, num_bits_weight=0, noise=0.2)
    lin.train()
    x = torch.rand(4, 20)
    y1 = lin(x)
    lin.eval()
    y2 = lin(x)  # eval: no weight noise
    expect = F.linear(x, lin.weight, lin.bias)
    assert torch.allclose(y2, expect)
    assert not torch.allclose(y1, expect)


def test_simple_noise_modes():
    torch.manual_seed(0)
    out = torch.randn(1000, 100)
    n1 = ops.simple_noise(out, "uniform_ind", 0.1)
    amp = 0.1 * out.abs().max()
    assert n1.abs().max() <= amp + 1e-5
    n2 = ops.simple_noise(out, "normal_ind", 0.1)
    s = 0.1 * out.abs().max()
    assert abs(n2.std().item() - s.item()) / s.item() < 0.05
    n3 = ops.simple_noise(out, "uniform_dep", 0.5)
    assert (n3 >= 0.5 - 1e-6).all() and (n3 <= 2.0 + 1e-6).all()
    n4 = ops.simple_noise(out, "normal_dep", 0.3)
    assert n4.shape == out.shape
