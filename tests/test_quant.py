"""Unit tests for the quantization core (reference chain parity + STE)."""

import numpy as np
import pytest
import torch

from noisynet_amd import ops
from noisynet_amd.ops import reference as ref
from noisynet_amd.quant import QuantMeasure, finish_calibration, start_calibration


def chain_reference(x, num_bits, min_value, max_value):
    """The exact reference chain (hardware_model.py:148-170), deterministic."""
    qmax = 2.0 ** num_bits - 1.0
    scale = max((max_value - min_value) / qmax, 1e-6)
    q = (x - min_value) / scale
    q = q.clamp(0.0, qmax).round()
    return q * scale + min_value


@pytest.mark.parametrize("bits", [2, 4, 8])
@pytest.mark.parametrize("rng", [(0.0, 1.0), (-1.0, 1.0), (0.0, 5.0)])
def test_fake_quant_deterministic(bits, rng):
    torch.manual_seed(0)
    x = torch.randn(64, 33) * 2
    mn, mx = rng
    out = ops.fake_quant(x, bits, mn, mx, 0.0)
    expected = chain_reference(x, bits, mn, mx)
    assert torch.allclose(out, expected)
    # output on the quantization grid
    scale = max((mx - mn) / (2 ** bits - 1), 1e-6)
    grid = ((out - mn) / scale).round()
    assert torch.allclose(out, grid * scale + mn, atol=1e-5)


def test_fake_quant_stochastic_statistics():
    """Stochastic rounding must be unbiased: E[q(x)] == x on the grid interior."""
    torch.manual_seed(0)
    val = 0.3  # between 4-bit grid points of [0,1]
    x = torch.full((200000,), val)
    out = ops.fake_quant(x, 4, 0.0, 1.0, 0.5)
    assert abs(out.mean().item() - val) < 2e-3
    # only the two adjacent grid points appear
    scale = 1.0 / 15
    levels = torch.unique((out / scale).round())
    assert levels.numel() <= 2


def test_fake_quant_ste_backward():
    x = torch.tensor([-2.0, -0.5, 0.0, 0.5, 0.99, 1.5], requires_grad=True)
    out = ops.fake_quant(x, 4, 0.0, 1.0, 0.0)
    out.sum().backward()
    # zero outside [0,1], one inside
    expected = torch.tensor([0.0, 0.0, 1.0, 1.0, 1.0, 0.0])
    assert torch.equal(x.grad, expected)


def test_quantmeasure_calibration_flow():
    torch.manual_seed(1)
    qm = QuantMeasure(num_bits=4, stochastic=0.0, pctl=99.98,
                      calculate_running=True)
    qm.train()
    for _ in range(5):
        qm(torch.rand(32, 100) * 3)
    assert len(qm.running_list) == 5

    class Holder(torch.nn.Module):
        def __init__(self, q):
            super().__init__()
            self.q = q

    m = Holder(qm)
    finish_calibration(m)
    assert not qm.calculate_running
    assert 2.0 < float(qm.running_max) <= 3.01

    qm.eval()
    out = qm(torch.rand(8, 100) * 3)
    assert out.max() <= float(qm.running_max) + 1e-5


def test_quantmeasure_negative_range_weights():
    torch.manual_seed(2)
    qm = QuantMeasure(num_bits=4, stochastic=0.0, min_value=-1.0,
                      max_value=1.0, pctl=99.0, calculate_running=True)
    qm.train()
    w = torch.randn(1000) * 0.3
    qm(w)
    assert not qm.calculate_running
    assert float(qm.running_min) < 0 < float(qm.running_max)


def test_kth_percentile_matches_torch():
    torch.manual_seed(3)
    x = torch.rand(10000)
    v = ops.kth_percentile(x, 99.0)
    k = int(x.numel() * 0.99)
    expect, _ = torch.kthvalue(x.flatten(), k)
    assert torch.allclose(v, expect)
