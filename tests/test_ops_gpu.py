"""GPU numerics tests: every HIP kernel vs the plain-PyTorch fp32 reference.

All marked @pytest.mark.gpu (run on MI355X via gpurun / at round end).
bf16 kernels compare against fp32 references with bf16-appropriate
tolerances (inputs are drawn bf16-representable where exactness matters).
"""

import os

import numpy as np
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

from noisynet_amd import ops  # noqa: E402
from noisynet_amd.ops import reference as ref  # noqa: E402


def dev():
    return torch.device("cuda")


def cl(t):
    return t.contiguous(memory_format=torch.channels_last)


def to_bf16_representable(x):
    return x.bfloat16().float()


# ---------------------------------------------------------------------------
# elementwise
# ---------------------------------------------------------------------------


def test_ext_loaded_native_path_required():
    assert ops.has_ext(), "HIP extension must be loadable on the GPU box"
    x = torch.randn(16, device=dev())
    # native op must run (raises if ext missing)
    ops.fake_quant(x, 4, 0.0, 1.0, 0.0)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_fake_quant_gpu_matches_reference(dtype):
    torch.manual_seed(0)
    x = torch.randn(4096, device=dev(), dtype=dtype) * 2
    out = ops.fake_quant(x, 4, 0.0, 1.0, 0.0)
    expect = ref.fake_quant_forward(x.float().cpu(), 4, 0.0, 1.0, 0.0)
    got = out.float().cpu()
    # rounding boundaries can differ by one level in bf16; compare levels
    atol = 1e-6 if dtype == torch.float32 else 1.5 / 15
    assert torch.allclose(got, expect, atol=atol)
    if dtype == torch.float32:
        assert torch.equal(got, expect)


def test_fake_quant_gpu_stochastic_unbiased():
    x = torch.full((1 << 20,), 0.3, device=dev())
    out = ops.fake_quant(x, 4, 0.0, 1.0, 0.5)
    assert abs(out.mean().item() - 0.3) < 2e-3


def test_ste_mask_gpu():
    x = torch.tensor([-2.0, 0.5, 1.5], device=dev())
    g = torch.ones_like(x)
    out = ops.ext().ste_mask(g, x, 0.0, 1.0)
    assert out.cpu().tolist() == [0.0, 1.0, 0.0]


def test_mult_uniform_noise_gpu_bounds():
    x = torch.randn(100000, device=dev())
    out = ops.ext().mult_uniform_noise(x, 0.1, 1234)
    d = (out - x).abs()
    assert (d <= 0.1 * x.abs() + 1e-5).all()
    # roughly uniform: mean |d| ~ 0.05 * |x|
    ratio = (d / x.abs().clamp_min(1e-8)).mean().item()
    assert 0.045 < ratio < 0.055


def test_dropout_gpu_stats():
    x = torch.ones(1 << 20, device=dev())
    y, mask = ops.ext().dropout_fwd(x, 0.25, 999)
    keep = (mask > 0).float().mean().item()
    assert abs(keep - 0.75) < 0.01
    assert abs(y.mean().item() - 1.0) < 0.01  # inverted dropout preserves mean


# ---------------------------------------------------------------------------
# conv / linear MFMA kernels
# ---------------------------------------------------------------------------


CONV_SHAPES = [
    # N, C, H, W, K, R, stride, pad      -- NoisyNet + ResNet shape classes
    (4, 3, 32, 32, 65, 5, 1, 0),
    (4, 65, 14, 14, 120, 5, 1, 0),
    (2, 16, 16, 16, 32, 3, 1, 1),
    (2, 8, 16, 16, 16, 3, 2, 1),
    (2, 64, 7, 7, 128, 1, 1, 0),
    (3, 33, 9, 9, 17, 3, 1, 1),
]


@pytest.mark.parametrize("shape", CONV_SHAPES)
def test_conv_fwd_matches_torch(shape):
    N, C, H, W, K, R, stride, pad = shape
    torch.manual_seed(0)
    x = to_bf16_representable(torch.randn(N, C, H, W)).to(dev()).bfloat16()
    w = to_bf16_representable(torch.randn(K, C, R, R) * 0.2).to(dev()).bfloat16()
    out = ops.ext().conv_fwd(cl(x), cl(w), stride, pad)
    expect = F.conv2d(x.float().cpu(), w.float().cpu(), None, stride, pad)
    got = out.float().cpu()
    err = (got - expect).abs().max().item()
    scale = expect.abs().max().item() + 1e-6
    assert err / scale < 0.02, f"rel err {err/scale}"


@pytest.mark.parametrize("shape", CONV_SHAPES)
def test_conv_dgrad_matches_torch(shape):
    N, C, H, W, K, R, stride, pad = shape
    torch.manual_seed(1)
    OH = (H + 2 * pad - R) // stride + 1
    g = to_bf16_representable(torch.randn(N, K, OH, OH)).to(dev()).bfloat16()
    w = to_bf16_representable(torch.randn(K, C, R, R) * 0.2).to(dev()).bfloat16()
    dx = ops.ext().conv_dgrad(cl(g), cl(w), stride, pad, H, W)
    expect = torch.nn.grad.conv2d_input((N, C, H, W), w.float().cpu(),
                                        g.float().cpu(), stride, pad)
    got = dx.float().cpu()
    err = (got - expect).abs().max().item()
    scale = expect.abs().max().item() + 1e-6
    assert err / scale < 0.02, f"rel err {err/scale}"


@pytest.mark.parametrize("shape", CONV_SHAPES)
def test_conv_wgrad_matches_torch(shape):
    N, C, H, W, K, R, stride, pad = shape
    torch.manual_seed(2)
    OH = (H + 2 * pad - R) // stride + 1
    g = to_bf16_representable(torch.randn(N, K, OH, OH) * 0.1).to(dev()).bfloat16()
    x = to_bf16_representable(torch.randn(N, C, H, W)).to(dev()).bfloat16()
    dw = ops.ext().conv_wgrad(cl(g), cl(x), stride, pad, R, R)
    expect = torch.nn.grad.conv2d_weight(x.float().cpu(), (K, C, R, R),
                                         g.float().cpu(), stride, pad)
    got = dw.float().cpu()
    err = (got - expect).abs().max().item()
    scale = expect.abs().max().item() + 1e-6
    assert err / scale < 0.03, f"rel err {err/scale}"


def test_linear_fwd_dgrad_wgrad():
    torch.manual_seed(3)
    B, I, O = 128, 3000, 390
    x = to_bf16_representable(torch.randn(B, I) * 0.5).to(dev()).bfloat16()
    w = to_bf16_representable(torch.randn(O, I) * 0.05).to(dev()).bfloat16()
    g = to_bf16_representable(torch.randn(B, O) * 0.1).to(dev()).bfloat16()

    y = ops.ext().linear_fwd(x, w)
    y_ref = x.float().cpu() @ w.float().cpu().t()
    rel = (y.float().cpu() - y_ref).abs().max() / (y_ref.abs().max() + 1e-6)
    assert rel < 0.02, rel

    dx = ops.ext().linear_dgrad(g, w)
    dx_ref = g.float().cpu() @ w.float().cpu()
    rel = (dx.float().cpu() - dx_ref).abs().max() / (dx_ref.abs().max() + 1e-6)
    assert rel < 0.02, rel

    dw = ops.ext().linear_wgrad(g, x)
    dw_ref = g.float().cpu().t() @ x.float().cpu()
    rel = (dw.float().cpu() - dw_ref).abs().max() / (dw_ref.abs().max() + 1e-6)
    assert rel < 0.03, rel


def test_conv_asymmetric_transpose_detection():
    """A=I with asymmetric W catches transposed output layouts (§5.4 r16)."""
    C = 16
    x = torch.zeros(1, C, 4, 4, device=dev()).bfloat16()
    # delta input at one pixel
    x[0, :, 1, 2] = torch.arange(C, device=dev()).bfloat16() / C
    w = torch.zeros(8, C, 1, 1, device=dev()).bfloat16()
    for k in range(8):
        w[k, (3 * k + 1) % C, 0, 0] = 1.0 + 0.125 * k
    out = ops.ext().conv_fwd(cl(x), cl(w), 1, 0)
    expect = F.conv2d(x.float().cpu(), w.float().cpu())
    assert torch.allclose(out.float().cpu(), expect, atol=1e-2)


def test_fused_conv_noise_statistics():
    """The fused kernel's noise must be N(0, sqrt(factor * conv(x, |w|)))."""
    torch.manual_seed(4)
    N, C, H, W, K, R = 64, 16, 12, 12, 32, 3
    x = torch.rand(N, C, H, W).to(dev()).bfloat16()
    w = (torch.randn(K, C, R, R) * 0.2).to(dev()).bfloat16()
    wq = w.clone()
    factor = 0.05
    out, tele = ops.ext().conv_fwd_fused(
        cl(x), cl(wq), cl(w), torch.empty(0, device=dev(), dtype=torch.bfloat16),
        1, 0, 1, torch.tensor([factor], device=dev()), 42, True)[:2]
    clean = F.conv2d(x.float().cpu(), w.float().cpu())
    noise = out.float().cpu() - clean
    sig = F.conv2d(x.float().cpu(), w.float().cpu().abs())
    # standardize: noise / sqrt(factor*sig) ~ N(0,1)
    z = noise / (factor * sig).clamp_min(1e-9).sqrt()
    assert abs(z.mean().item()) < 0.02
    assert abs(z.std().item() - 1.0) < 0.05
    # telemetry: sum sigma_abs
    assert abs(tele[0].item() - sig.sum().item()) / sig.sum().item() < 0.02
    # max of clean y
    assert abs(tele[2].item() - clean.max().item()) < 0.1


def test_fused_conv_abs2_mode():
    torch.manual_seed(5)
    N, C, H, W, K, R = 32, 8, 10, 10, 16, 3
    x = torch.rand(N, C, H, W).to(dev()).bfloat16()
    w = (torch.randn(K, C, R, R) * 0.3).to(dev()).bfloat16()
    factor = 0.02
    out, tele = ops.ext().conv_fwd_fused(
        cl(x), cl(w), cl(w), torch.empty(0, device=dev(), dtype=torch.bfloat16),
        1, 0, 2, torch.tensor([factor], device=dev()), 7, True)[:2]
    clean = F.conv2d(x.float().cpu(), w.float().cpu())
    noise = out.float().cpu() - clean
    aw = w.float().cpu().abs()
    sig2 = F.conv2d(x.float().cpu(), aw * aw + aw)
    z = noise / (factor * sig2).clamp_min(1e-9).sqrt()
    assert abs(z.std().item() - 1.0) < 0.05
    # telemetry sigma_abs uses |w| even in abs2 mode
    sig1 = F.conv2d(x.float().cpu(), aw)
    assert abs(tele[0].item() - sig1.sum().item()) / sig1.sum().item() < 0.02


def test_sigma_noise_only():
    torch.manual_seed(6)
    x = torch.rand(32, 8, 10, 10).to(dev()).bfloat16()
    w = (torch.randn(16, 8, 3, 3) * 0.3).to(dev()).bfloat16()
    noise, tele = ops.ext().sigma_noise_conv(cl(x), cl(w), 1, 0, 1,
        torch.tensor([0.05], device=dev()), 9, False)
    aw = w.float().cpu().abs()
    sig = F.conv2d(x.float().cpu(), aw)
    z = noise.float().cpu() / (0.05 * sig).clamp_min(1e-9).sqrt()
    assert abs(z.std().item() - 1.0) < 0.05
    assert abs(z.mean().item()) < 0.02


# ---------------------------------------------------------------------------
# bn / pool / loss / optim
# ---------------------------------------------------------------------------


def test_bn_stats_and_act():
    torch.manual_seed(7)
    x = torch.randn(32, 65, 14, 14, device=dev()).bfloat16()
    mean, var = ops.ext().bn_stats(cl(x))
    xe = x.float()
    em = xe.mean(dim=(0, 2, 3))
    ev = xe.var(dim=(0, 2, 3), unbiased=False)
    assert torch.allclose(mean, em, atol=2e-3, rtol=1e-2)
    assert torch.allclose(var, ev, atol=2e-3, rtol=2e-2)

    gamma = torch.randn(65, device=dev())
    beta = torch.randn(65, device=dev())
    invstd = (var + 1e-5).rsqrt()
    y = ops.ext().bn_act_fwd(cl(x), mean, invstd, gamma, beta, True, 2.0)
    ye = ref.bn_act_forward(xe, gamma, beta, mean, invstd, 2.0, True)
    assert (y.float() - ye).abs().max().item() < 0.05


def test_maxpool_gpu():
    torch.manual_seed(8)
    x = torch.randn(8, 65, 28, 28, device=dev()).bfloat16()
    y, code = ops.ext().maxpool2x2_fwd(cl(x))
    ye = F.max_pool2d(x.float(), 2, 2)
    assert torch.equal(y.float().cpu(), ye.cpu())
    g = torch.randn_like(y)
    gx = ops.ext().maxpool2x2_bwd(cl(g), code, 28, 28)
    # scatter sums equal
    assert torch.allclose(gx.sum().float().cpu(), g.sum().float().cpu(), rtol=1e-2)


def test_softmax_xent_gpu():
    torch.manual_seed(9)
    logits = torch.randn(512, 10, device=dev())
    target = torch.randint(0, 10, (512,), device=dev())
    loss, sm = ops.ext().softmax_xent_fwd(logits, target)
    le = F.cross_entropy(logits, target)
    assert abs(loss.item() - le.item()) < 1e-4
    grad = ops.ext().softmax_xent_bwd(sm, target, 1.0)
    logits2 = logits.clone().requires_grad_(True)
    F.cross_entropy(logits2, target).backward()
    assert torch.allclose(grad, logits2.grad, atol=1e-5)


def test_sgd_step_gpu_matches_torch():
    torch.manual_seed(10)
    p0 = torch.randn(1000, device=dev())
    g = torch.randn(1000, device=dev())
    # native
    p1 = p0.clone()
    buf1 = torch.zeros_like(p1)
    for _ in range(3):
        ops.ext().sgd_step(p1, g, buf1, 0.1, 0.9, 1e-4, True, -0.3, 0.3)
    # torch reference
    p2 = p0.clone().requires_grad_(True)
    opt = torch.optim.SGD([p2], lr=0.1, momentum=0.9, weight_decay=1e-4,
                          nesterov=True)
    for _ in range(3):
        p2.grad = g.clone()
        opt.step()
        with torch.no_grad():
            p2.clamp_(-0.3, 0.3)
    assert torch.allclose(p1, p2.detach(), atol=1e-5)


def test_adamw_step_gpu_matches_torch():
    torch.manual_seed(11)
    p0 = torch.randn(1000, device=dev())
    g = torch.randn(1000, device=dev())
    p1 = p0.clone()
    m = torch.zeros_like(p1)
    v = torch.zeros_like(p1)
    for step in range(1, 4):
        ops.ext().adamw_step(p1, g, m, v, step, 1e-3, 0.9, 0.999, 1e-8, 0.01,
                             0.0, 0.0)
    p2 = p0.clone().requires_grad_(True)
    opt = torch.optim.AdamW([p2], lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                            weight_decay=0.01)
    for _ in range(3):
        p2.grad = g.clone()
        opt.step()
    assert torch.allclose(p1, p2.detach(), atol=1e-6)


# ---------------------------------------------------------------------------
# end-to-end on GPU
# ---------------------------------------------------------------------------


def test_model_step_gpu_bf16():
    from noisynet_amd import utils
    from noisynet_amd.config import broadcast_per_layer, build_noisynet_parser
    from noisynet_amd.models.noisynet import Net
    from noisynet_amd import optim as native_optim
    from noisynet_amd.quant import finish_calibration, start_calibration

    args = build_noisynet_parser().parse_args(
        ['--current', '1', '--q_a', '4', '--act_max', '5', '--w_max1', '0.3',
         '--calculate_running'])
    broadcast_per_layer(args)
    torch.manual_seed(0)
    model = Net(args)
    utils.init_model(model, args)
    model = model.cuda().bfloat16()
    for m in model.modules():
        if isinstance(m, (torch.nn.BatchNorm1d, torch.nn.BatchNorm2d)):
            m.float()
    model = model.to(memory_format=torch.channels_last)

    x = (torch.randint(0, 16, (64, 3, 32, 32)).cuda().bfloat16() / 15.0)
    x = cl(x)
    y = torch.randint(0, 10, (64,)).cuda()

    start_calibration(model)
    with torch.no_grad():
        model(x, 0, 0)
    finish_calibration(model, 'cuda')

    opt = native_optim.SGD(model.parameters(), lr=0.01, momentum=0.9,
                           nesterov=True)
    model.train()
    losses = []
    for i in range(8):
        out = model(x, 0, 100 + i)
        loss = F.cross_entropy(out.float(), y)
        opt.zero_grad(set_to_none=False)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    # with I_max=1nA the per-step loss is dominated by injected noise;
    # require finiteness here (learning check: test_model_learns_gpu)
    assert all(np.isfinite(losses))


def test_model_gpu_matches_cpu_noisefree():
    """Noise-free fp32 model: GPU forward (HIP kernels) vs CPU forward."""
    from noisynet_amd import utils
    from noisynet_amd.config import broadcast_per_layer, build_noisynet_parser
    from noisynet_amd.models.noisynet import Net

    args = build_noisynet_parser().parse_args(['--batch_size', '16'])
    broadcast_per_layer(args)
    torch.manual_seed(0)
    model = Net(args)
    utils.init_model(model, args)
    model.eval()
    x = torch.rand(16, 3, 32, 32)
    with torch.no_grad():
        out_cpu = model(x)
        gm = model.cuda().to(memory_format=torch.channels_last)
        out_gpu = gm(cl(x.cuda()))
    err = (out_gpu.cpu() - out_cpu).abs().max().item()
    scale = out_cpu.abs().max().item()
    assert err / scale < 0.05, err / scale


def test_kth_percentile_gpu_matches_kthvalue():
    torch.manual_seed(12)
    for dtype in (torch.float32, torch.bfloat16):
        x = (torch.randn(100000) * 3).to(dev(), dtype)
        for pctl in (50.0, 99.0, 99.98):
            got = ops.ext().kth_percentile(x.view(-1), pctl).float().item()
            k = max(1, int(x.numel() * pctl / 100.0))
            expect, _ = torch.kthvalue(x.float().cpu().flatten(), k)
            assert got == pytest.approx(expect.item(), abs=1e-6), (dtype, pctl)


def test_model_learns_gpu_noisefree():
    """Noise-free bf16 model on the HIP kernels must overfit a fixed batch."""
    from noisynet_amd import utils
    from noisynet_amd.config import broadcast_per_layer, build_noisynet_parser
    from noisynet_amd.models.noisynet import Net
    from noisynet_amd import optim as native_optim

    args = build_noisynet_parser().parse_args(['--batch_size', '64'])
    broadcast_per_layer(args)
    torch.manual_seed(0)
    model = Net(args)
    utils.init_model(model, args)
    model = model.cuda().bfloat16()
    for m in model.modules():
        if isinstance(m, (torch.nn.BatchNorm1d, torch.nn.BatchNorm2d)):
            m.float()
    model = model.to(memory_format=torch.channels_last)
    x = cl(torch.rand(64, 3, 32, 32).cuda().bfloat16())
    y = torch.randint(0, 10, (64,)).cuda()
    opt = native_optim.SGD(model.parameters(), lr=0.02, momentum=0.9,
                           nesterov=True)
    model.train()
    losses = []
    for i in range(25):
        out = model(x, 0, 100 + i)
        loss = F.cross_entropy(out.float(), y)
        opt.zero_grad(set_to_none=False)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    first = sum(losses[:5]) / 5
    last = sum(losses[-5:]) / 5
    assert last < first * 0.7, (first, last)


def test_depthwise_conv_gpu_matches_torch():
    torch.manual_seed(13)
    # covers the win3 sliding-window wgrad (3x3/s1/p1), the 5x5 VEC4 path,
    # the stride-2 vec path and an odd-channel scalar fallback
    for stride, pad, k, C in ((1, 1, 3, 32), (2, 2, 5, 32), (2, 1, 3, 96),
                              (1, 1, 3, 24), (1, 1, 3, 25)):
        x = to_bf16_representable(torch.randn(4, C, 14, 14)).to(dev()).bfloat16()
        w = to_bf16_representable(torch.randn(C, 1, k, k) * 0.2).to(dev()).bfloat16()
        y = ops.ext().dwconv_fwd(cl(x), w.contiguous(),
                                 torch.empty(0, device=dev(), dtype=x.dtype),
                                 stride, pad)
        ye = F.conv2d(x.float().cpu(), w.float().cpu(), None, stride, pad, 1, C)
        rel = (y.float().cpu() - ye).abs().max() / (ye.abs().max() + 1e-6)
        assert rel < 0.02, rel
        g = to_bf16_representable(torch.randn_like(ye)).to(dev()).bfloat16()
        dx = ops.ext().dwconv_dgrad(cl(g), w.contiguous(), stride, pad, 14, 14)
        dxe = torch.nn.grad.conv2d_input((4, C, 14, 14), w.float().cpu(),
                                         g.float().cpu(), stride, pad, 1, C)
        rel = (dx.float().cpu() - dxe).abs().max() / (dxe.abs().max() + 1e-6)
        assert rel < 0.02, rel
        dw = ops.ext().dwconv_wgrad(cl(g), cl(x), stride, pad, k, k)
        dwe = torch.nn.grad.conv2d_weight(x.float().cpu(), (C, 1, k, k),
                                          g.float().cpu(), stride, pad, 1, C)
        rel = (dw.float().cpu() - dwe).abs().max() / (dwe.abs().max() + 1e-6)
        assert rel < 0.03, rel


def test_generic_pooling_gpu():
    torch.manual_seed(14)
    x = torch.randn(4, 64, 56, 56, device=dev()).bfloat16()
    y, code = ops.ext().maxpool_fwd(cl(x), 3, 2, 1)
    ye = F.max_pool2d(x.float(), 3, 2, 1)
    assert torch.equal(y.float().cpu(), ye.cpu())
    g = torch.randn_like(y)
    gx = ops.ext().maxpool_bwd(cl(g), code, 56, 56, 3, 2, 1)
    assert torch.allclose(gx.float().sum().cpu(), g.float().sum().cpu(),
                          rtol=1e-2)

    ya = ops.ext().avgpool_fwd(cl(x), 7, 1, 0)
    yae = F.avg_pool2d(x.float(), 7, 1)
    assert (ya.float() - yae).abs().max().item() < 0.02


def test_activations_gpu_match_reference():
    torch.manual_seed(15)
    x = torch.randn(10000, device=dev())
    for idx, fn in ((0, lambda v: v * torch.sigmoid(v)),
                    (2, F.hardswish), (3, F.hardsigmoid),
                    (4, torch.sigmoid)):
        y = ops.ext().act_fwd(x, idx)
        assert torch.allclose(y, fn(x), atol=1e-5), idx
    # analytic backward vs autograd (swish)
    g = torch.randn_like(x)
    gx = ops.ext().act_bwd(g, x, 0)
    x2 = x.clone().requires_grad_(True)
    (x2 * torch.sigmoid(x2)).backward(g)
    assert torch.allclose(gx, x2.grad, atol=1e-5)


@pytest.mark.parametrize("arch", ["resnet18", "mobilenet_v2"])
def test_imagenet_models_step_gpu(arch):
    from noisynet_amd.config import build_main_parser
    from noisynet_amd import optim as native_optim
    argv = ['-a', arch, '--q_a', '4', '--calculate_running']
    args = build_main_parser().parse_args(argv)
    torch.manual_seed(0)
    if arch == 'resnet18':
        from noisynet_amd.models.resnet import ResNet18
        m = ResNet18(args)
    else:
        from noisynet_amd.models.mobilenet import mobilenet_v2
        m = mobilenet_v2(args)
    m = m.cuda().bfloat16()
    for mod in m.modules():
        if isinstance(mod, (torch.nn.BatchNorm1d, torch.nn.BatchNorm2d)):
            mod.float()
    m = m.to(memory_format=torch.channels_last)
    x = cl(torch.rand(8, 3, 224, 224).cuda().bfloat16())
    y = torch.randint(0, 1000, (8,)).cuda()
    opt = native_optim.SGD(m.parameters(), lr=0.01, momentum=0.9)
    m.train()
    out = m(x)
    loss = F.cross_entropy(out.float(), y)
    opt.zero_grad(set_to_none=False)
    loss.backward()
    opt.step()
    assert torch.isfinite(loss)


def test_efficientnet_b0_step_gpu():
    from noisynet_amd.timm.models import create_model
    from noisynet_amd import optim as native_optim
    torch.manual_seed(0)
    m = create_model('efficientnet_b0', num_classes=1000)
    m = m.cuda().bfloat16()
    for mod in m.modules():
        if isinstance(mod, (torch.nn.BatchNorm1d, torch.nn.BatchNorm2d)):
            mod.float()
    m = m.to(memory_format=torch.channels_last)
    x = cl(torch.rand(8, 3, 224, 224).cuda().bfloat16())
    y = torch.randint(0, 1000, (8,)).cuda()
    opt = native_optim.SGD(m.parameters(), lr=0.01, momentum=0.9)
    m.train()
    loss = F.cross_entropy(m(x).float(), y)
    opt.zero_grad(set_to_none=False)
    loss.backward()
    opt.step()
    assert torch.isfinite(loss)


def test_patch_mode_conv_matches_torch():
    """Shapes that trigger the image-patch LDS-resident kernel (small
    spatial, C>8) vs torch, including the odd-C conv2 shape."""
    torch.manual_seed(21)
    for (N, C, H, W, K, R, stride, pad) in (
            (8, 65, 14, 14, 120, 5, 1, 0),   # NoisyNet conv2
            (4, 24, 16, 16, 32, 3, 1, 1),    # padded 3x3
            (4, 16, 15, 15, 24, 3, 2, 1)):   # strided
        x = to_bf16_representable(torch.randn(N, C, H, W)).to(dev()).bfloat16()
        w = to_bf16_representable(torch.randn(K, C, R, R) * 0.2).to(dev()).bfloat16()
        from noisynet_amd.ops.functional import _patch_eligible
        assert _patch_eligible(x, w, pad)
        empty = torch.empty(0, device=dev(), dtype=torch.bfloat16)
        zf = torch.zeros(1, device=dev())
        y = ops.ext().conv_fwd_fused(cl(x), cl(w), cl(w), empty,
                                     stride, pad, 0, zf, 0, False)[0]
        ye = F.conv2d(x.float().cpu(), w.float().cpu(), None, stride, pad)
        rel = (y.float().cpu() - ye).abs().max() / (ye.abs().max() + 1e-6)
        assert rel < 0.02, rel


def test_l3_gradient_penalty_double_backward_gpu():
    """L3 penalty (double backward) must compose through the HIP
    conv/GEMM dgrad/wgrad kernels (noisynet.py:1392-1476)."""
    from noisynet_amd.config import broadcast_per_layer, build_noisynet_parser
    from noisynet_amd.drivers import cifar
    from noisynet_amd.models.noisynet import Net
    from noisynet_amd import utils

    args = build_noisynet_parser().parse_args(['--L3', '0.01'])
    broadcast_per_layer(args)
    torch.manual_seed(0)
    m = Net(args)
    utils.init_model(m, args)
    m = m.cuda().to(memory_format=torch.channels_last)
    m.train()
    x = cl(torch.rand(16, 3, 32, 32).cuda())
    y = torch.randint(0, 10, (16,)).cuda()
    out = m(x, 0, 0)
    loss = F.cross_entropy(out, y)
    loss, retain = cifar.gradient_penalties(m, args, loss)
    loss.backward(retain_graph=retain)
    cifar.post_backward_penalties(m, args, loss)
    for name in ('conv1', 'conv2', 'linear1', 'linear2'):
        g = getattr(m, name).weight.grad
        assert g is not None and torch.isfinite(g.float()).all(), name


def test_distortion_harness_gpu():
    from noisynet_amd.config import broadcast_per_layer, build_noisynet_parser
    from noisynet_amd.harness import test_distortion as run_distortion
    from noisynet_amd.models.noisynet import Net
    from noisynet_amd import utils

    args = build_noisynet_parser().parse_args(
        ['--num_sims', '2', '--batch_size', '32'])
    broadcast_per_layer(args)
    args.stuck_at_weights = None
    args.test_temp = 0
    m = Net(args)
    utils.init_model(m, args)
    m = m.cuda().to(memory_format=torch.channels_last)
    m.eval()
    inputs = cl(torch.rand(64, 3, 32, 32).cuda())
    labels = torch.randint(0, 10, (64,)).cuda()
    w_before = m.conv1.weight.data.clone()
    run_distortion(m, args, val_loader=(inputs, labels), mode='weights',
                   vars=[0.1, 0.3])
    assert torch.equal(m.conv1.weight.data, w_before)
    args.stuck_at_weights = 'random_zero'
    run_distortion(m, args, val_loader=(inputs, labels), mode='weights',
                   vars=[0.2])


def test_seeded_determinism_gpu():
    """Same torch.manual_seed -> identical in-kernel Philox draws."""
    from noisynet_amd import ops as O
    x = torch.rand(32, 8, 10, 10).cuda().bfloat16()
    w = (torch.randn(16, 8, 3, 3) * 0.3).cuda().bfloat16()
    outs = []
    for _ in range(2):
        torch.manual_seed(1234)
        out = O.fused_noisy_conv2d(cl(x), cl(w), w.detach(), None, 1, 0,
                                   'abs', 0.05)
        outs.append(out)
    assert torch.equal(outs[0], outs[1])
    # and a different seed gives different noise
    torch.manual_seed(99)
    out3 = O.fused_noisy_conv2d(cl(x), cl(w), w.detach(), None, 1, 0,
                                'abs', 0.05)
    assert not torch.equal(outs[0], out3)


def test_cifar_driver_short_gpu(tmp_path, monkeypatch):
    """Two epochs of the flagship noisynet.py config end-to-end on GPU."""
    monkeypatch.chdir(tmp_path)
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from noisynet_amd.drivers import cifar
    out = cifar.main(['--nepochs', '2', '--n_train', '2048', '--n_test', '512',
                      '--batch_size', '256', '--current', '1', '--act_max', '5',
                      '--q_a', '4', '--calculate_running', '--LR', '0.005',
                      '--bf16', '--keep_bn_fp32'])
    assert out


@pytest.mark.parametrize("dtype", [torch.float32, torch.float16])
def test_conv_exact_dtype_paths(dtype):
    """fp32 convs use the exact f32-input MFMA (bitwise f32 fmaf chain,
    cdna_hip_programming.md §3); fp16 uses the f16 MFMA. Tight tolerances
    prove the compute really runs at the input precision (a bf16-internal
    path would miss by ~1e-2)."""
    torch.manual_seed(31)
    for (N, C, H, W, K, R, stride, pad) in (
            (4, 65, 14, 14, 120, 5, 1, 0),
            (2, 16, 16, 16, 32, 3, 1, 1),
            (4, 3, 32, 32, 65, 5, 1, 0)):
        x = torch.randn(N, C, H, W, dtype=dtype, device=dev())
        w = (torch.randn(K, C, R, R, dtype=dtype, device=dev()) * 0.2)
        out = ops.ext().conv_fwd(cl(x), cl(w), stride, pad)
        expect = F.conv2d(x.float().cpu(), w.float().cpu(), None, stride, pad)
        rel = (out.float().cpu() - expect).abs().max() / (expect.abs().max() + 1e-6)
        tol = 1e-5 if dtype == torch.float32 else 4e-3
        assert rel < tol, (dtype, (N, C, H, W, K, R), rel)

        # dgrad/wgrad too
        OH = (H + 2 * pad - R) // stride + 1
        g = torch.randn(N, K, OH, OH, dtype=dtype, device=dev())
        dx = ops.ext().conv_dgrad(cl(g), cl(w), stride, pad, H, W)
        dxe = torch.nn.grad.conv2d_input((N, C, H, W), w.float().cpu(),
                                         g.float().cpu(), stride, pad)
        rel = (dx.float().cpu() - dxe).abs().max() / (dxe.abs().max() + 1e-6)
        assert rel < (1e-5 if dtype == torch.float32 else 5e-3), (dtype, rel)

        dw = ops.ext().conv_wgrad(cl(g), cl(x), stride, pad, R, R)
        dwe = torch.nn.grad.conv2d_weight(x.float().cpu(), (K, C, R, R),
                                          g.float().cpu(), stride, pad)
        rel = (dw.float().cpu() - dwe).abs().max() / (dwe.abs().max() + 1e-6)
        assert rel < (1e-4 if dtype == torch.float32 else 2e-2), (dtype, rel)


def test_linear_fp32_exact():
    torch.manual_seed(32)
    x = torch.randn(64, 3000, device=dev())
    w = torch.randn(390, 3000, device=dev()) * 0.05
    y = ops.ext().linear_fwd(x, w)
    ye = x.cpu() @ w.cpu().t()
    rel = (y.cpu() - ye).abs().max() / (ye.abs().max() + 1e-6)
    assert rel < 1e-5, rel


def test_patch_conv_fp32_exact():
    torch.manual_seed(33)
    x = torch.randn(4, 24, 14, 14, device=dev())
    w = torch.randn(32, 24, 3, 3, device=dev()) * 0.2
    from noisynet_amd.ops.functional import _patch_eligible
    assert _patch_eligible(x, w, 1)
    empty = torch.empty(0, device=dev())
    zf = torch.zeros(1, device=dev())
    y = ops.ext().conv_fwd_fused(cl(x), cl(w), cl(w), empty, 1, 1, 0, zf,
                                 0, False)[0]
    ye = F.conv2d(x.cpu(), w.cpu(), None, 1, 1)
    rel = (y.cpu() - ye).abs().max() / (ye.abs().max() + 1e-6)
    assert rel < 1e-5, rel


@pytest.mark.gpu
def test_im2col_materialize_flat_layout():
    """im2col produces the FLAT (r,s,c)-flattened row layout, zero-padded
    to a multiple of 8 columns, for both the aligned (C%8==0) and the
    decode (C%8!=0) kernels."""
    torch.manual_seed(0)
    for C in (3, 8, 13, 16):
        x = torch.randn(4, C, 9, 9, device="cuda", dtype=torch.bfloat16)
        xc = x.contiguous(memory_format=torch.channels_last)
        col = ops.ext().im2col_materialize(xc, 7, 1, 2, 3, 3)
        ref_u = F.unfold(x.float(), 3, padding=2)         # [N, C*9, L]
        N, _, L = ref_u.shape
        ref_u = (ref_u.view(N, C, 9, L).permute(0, 3, 2, 1)
                 .reshape(N * L, 9 * C))                  # [(m), (r,s,c)]
        rsc = 9 * C
        exp_cols = rsc if C % 8 == 0 else (rsc + 31) // 32 * 32
        assert col.shape == (N * L, exp_cols)
        assert torch.allclose(col[:, :rsc].float().cpu(), ref_u.cpu(),
                              atol=1e-1, rtol=1e-2), C
        assert (col[:, rsc:] == 0).all()


@pytest.mark.gpu
def test_linear_wgrad_wide_tile_variant():
    """K*C >= 128k routes to the 128x128-tile wgrad kernel; verify vs
    fp32 matmul, plus the narrow-tile (64x64) path on an aligned shape."""
    torch.manual_seed(3)
    for M, K, C in ((256, 128, 1024), (300, 16, 8200), (4096, 72, 80)):
        g = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        x = torch.randn(M, C, device="cuda", dtype=torch.bfloat16)
        dw = ops.ext().linear_wgrad(g, x).float()
        ref_dw = g.float().t() @ x.float()
        err = (dw - ref_dw).abs().max() / (ref_dw.abs().max() + 1e-6)
        assert err < 5e-2, (M, K, C, err)


@pytest.mark.gpu
def test_wgrad_deterministic():
    """Per-slice partial buffers give bit-identical wgrad across runs."""
    torch.manual_seed(4)
    g = torch.randn(200000, 72, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(200000, 80, device="cuda", dtype=torch.bfloat16)
    a = ops.ext().linear_wgrad(g, x)
    b = ops.ext().linear_wgrad(g, x)
    assert torch.equal(a, b)


@pytest.mark.gpu
def test_bn_finalize_running_stats_match_torch():
    """bn_stats_finalize updates running mean/var exactly like
    torch.nn.BatchNorm2d (momentum semantics + unbiased var)."""
    torch.manual_seed(5)
    x = torch.randn(8, 24, 10, 10, device="cuda", dtype=torch.bfloat16)
    xc = cl(x)
    bn = torch.nn.BatchNorm2d(24, momentum=0.1, eps=1e-5).cuda().float()
    bn.train()
    bn(x.float())
    rm = torch.zeros(24, device="cuda")
    rv = torch.ones(24, device="cuda")
    mean, invstd = ops.ext().bn_stats_finalize(xc, rm, rv, 0.1, 1e-5)
    assert torch.allclose(rm, bn.running_mean, atol=5e-3), \
        (rm - bn.running_mean).abs().max()
    assert torch.allclose(rv, bn.running_var, atol=5e-3, rtol=1e-2)
    ref_mean = x.float().mean(dim=(0, 2, 3))
    assert torch.allclose(mean, ref_mean, atol=5e-3)


@pytest.mark.gpu
def test_maxpool2x2_odd_dims():
    """Odd H/W leave a tail row/column outside every window; backward must
    zero them (the all-cells-write rewrite handles them explicitly)."""
    torch.manual_seed(7)
    for H, W in ((9, 9), (8, 9), (9, 8), (10, 10)):
        x = torch.randn(3, 13, H, W, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True)
        xc = cl(x.detach()).requires_grad_(True)
        y = ops.maxpool2x2(xc)
        g = torch.randn_like(y)
        y.backward(g)
        xr = x.detach().float().cpu().requires_grad_(True)
        yr = F.max_pool2d(xr, 2, 2)
        yr.backward(g.float().cpu())
        assert torch.allclose(xc.grad.float().cpu(), xr.grad, atol=1e-2), (H, W)


@pytest.mark.gpu
def test_bn_stats_odd_channels():
    """Channel counts around the 64-lane span boundary (the C=65 tail
    channel was a 1-lane critical path before per-block packing)."""
    torch.manual_seed(8)
    for C in (1, 24, 63, 64, 65, 100, 128, 130):
        x = torch.randn(16, C, 7, 7, device="cuda", dtype=torch.bfloat16)
        mean, var = ops.ext().bn_stats(cl(x))
        ref_mean = x.float().mean(dim=(0, 2, 3))
        ref_var = x.float().var(dim=(0, 2, 3), unbiased=False)
        assert torch.allclose(mean.cpu(), ref_mean.cpu(), atol=5e-3), C
        assert torch.allclose(var.cpu(), ref_var.cpu(), atol=5e-3, rtol=1e-2), C


def test_conv_wgrad_patch_matches_reference():
    """Patch wgrad (in-kernel im2col gather) vs torch fp32 conv2d_weight."""
    torch.manual_seed(21)
    shapes = [
        # (N, C, H, W, K, R, stride, pad): conv2-like, conv1-like, padded,
        # strided (ResNet downsample shape)
        (16, 65, 14, 14, 120, 5, 1, 0),
        (8, 3, 32, 32, 65, 5, 1, 0),
        (4, 24, 16, 16, 32, 3, 1, 1),
        (4, 32, 16, 16, 64, 3, 2, 1),
    ]
    for (N, C, H, W, K, R, stride, pad) in shapes:
        x = to_bf16_representable(torch.randn(N, C, H, W) * 0.5)
        OH = (H + 2 * pad - R) // stride + 1
        gy = to_bf16_representable(torch.randn(N, K, OH, OH) * 0.5)
        got = ops.ext().conv_wgrad_patch(
            cl(gy.cuda().bfloat16()), cl(x.cuda().bfloat16()),
            stride, pad, R, R)
        ref = torch.nn.grad.conv2d_weight(
            x.float(), (K, C, R, R), gy.float(), stride, pad)
        err = (got.float().cpu() - ref).abs().max().item()
        scale = ref.abs().max().item()
        assert err / scale < 0.02, ((N, C, H, W, K, R, stride, pad),
                                    err / scale)


def test_conv_wgrad_patch_deterministic():
    torch.manual_seed(3)
    x = cl(torch.randn(8, 65, 14, 14).cuda().bfloat16())
    gy = cl(torch.randn(8, 120, 10, 10).cuda().bfloat16())
    a = ops.ext().conv_wgrad_patch(gy, x, 1, 0, 5, 5)
    b = ops.ext().conv_wgrad_patch(gy, x, 1, 0, 5, 5)
    assert torch.equal(a, b)


def test_se_scale_matches_eager():
    """Fused SE gating (fwd + both backward outputs) vs eager broadcast."""
    torch.manual_seed(5)
    for gate in ('sigmoid', 'hard_sigmoid'):
        x = to_bf16_representable(torch.randn(4, 32, 7, 7))
        s = to_bf16_representable(torch.randn(4, 32, 1, 1))
        xg = cl(x.cuda().bfloat16()).requires_grad_(True)
        sg = s.cuda().bfloat16().requires_grad_(True)
        y = ops.se_scale(xg, sg, gate)
        g = to_bf16_representable(torch.randn_like(y.cpu().float())).cuda().bfloat16()
        y.backward(cl(g))

        xe = x.float().requires_grad_(True)
        se = s.float().requires_grad_(True)
        gate_fn = torch.sigmoid if gate == 'sigmoid' else F.hardsigmoid
        ye = xe * gate_fn(se)
        ye.backward(g.float().cpu())

        assert torch.allclose(y.float().cpu(), ye.detach(), atol=2e-2)
        assert torch.allclose(xg.grad.float().cpu(), xe.grad, atol=2e-2)
        assert torch.allclose(sg.grad.float().cpu(), se.grad,
                              atol=0.05, rtol=0.05), gate


def test_native_batchnorm2d_matches_torch():
    """NativeBatchNorm2d (fused kernel) vs nn.BatchNorm2d: output, grads,
    running stats, train and eval."""
    from noisynet_amd.models.conv2d_layers import NativeBatchNorm2d
    torch.manual_seed(6)
    C = 24
    ours = NativeBatchNorm2d(C).cuda()
    ref = torch.nn.BatchNorm2d(C).cuda()
    ref.load_state_dict(ours.state_dict())
    x = torch.randn(8, C, 9, 9).cuda()
    xo = cl(x.clone()).requires_grad_(True)
    xr = x.clone().requires_grad_(True)
    for train in (True, False):
        ours.train(train)
        ref.train(train)
        yo = ours(xo)
        yr = ref(xr)
        assert torch.allclose(yo.float(), cl(yr).float(), atol=1e-4), train
        yo.sum().backward()
        yr.sum().backward()
        assert torch.allclose(xo.grad.float(), cl(xr.grad).float(), atol=1e-4)
        assert torch.allclose(ours.weight.grad, ref.weight.grad, atol=1e-3)
        xo.grad = None
        xr.grad = None
        ours.zero_grad()
        ref.zero_grad()
    assert torch.allclose(ours.running_mean, ref.running_mean, atol=1e-5)
    assert torch.allclose(ours.running_var, ref.running_var, atol=1e-5)
    assert int(ours.num_batches_tracked) == int(ref.num_batches_tracked)


def test_pointwise_conv_grads_match_reference():
    """1x1 conv routes dgrad/wgrad through the GEMM kernels; grads must
    match the fp32 torch oracle."""
    torch.manual_seed(9)
    x = to_bf16_representable(torch.randn(6, 32, 9, 9) * 0.5)
    w = to_bf16_representable(torch.randn(48, 32, 1, 1) * 0.2)
    xg = cl(x.cuda().bfloat16()).requires_grad_(True)
    wg = cl(w.cuda().bfloat16()).requires_grad_(True)
    y = ops.conv2d(xg, wg, None, 1, 0)
    g = to_bf16_representable(torch.randn_like(y.float().cpu()))
    y.backward(cl(g.cuda().bfloat16()))

    xr = x.float().requires_grad_(True)
    wr = w.float().requires_grad_(True)
    yr = F.conv2d(xr, wr, None, 1, 0)
    yr.backward(g.float())
    assert torch.allclose(y.float().cpu(), yr.detach(), atol=2e-2)
    assert torch.allclose(xg.grad.float().cpu(), cl(xr.grad).float().cpu(),
                          atol=2e-2, rtol=2e-2)
    assert torch.allclose(wg.grad.float().cpu(), cl(wr.grad).float().cpu(),
                          atol=5e-2, rtol=5e-2)


def test_condconv_per_sample_gpu():
    """CondConv per-sample route (unfold + batched GEMM) on GPU, bf16,
    channels_last: matches the fp32 grouped-conv oracle."""
    from noisynet_amd.models.conv2d_layers import CondConv2d
    torch.manual_seed(17)
    m = CondConv2d(16, 24, kernel_size=3, stride=1, padding=1, groups=1,
                   bias=True, num_experts=4).cuda().bfloat16()
    x = cl(torch.randn(5, 16, 10, 10).cuda().bfloat16()).requires_grad_(True)
    rw = torch.softmax(torch.randn(5, 4), dim=1).cuda().bfloat16()
    out = m(x, rw)
    B = 5
    w = torch.matmul(rw.float(), m.weight.float()).view(
        B * m.out_channels, 16, 3, 3)
    b = torch.matmul(rw.float(), m.bias.float()).view(B * m.out_channels)
    ref = F.conv2d(x.detach().float().contiguous().view(1, B * 16, 10, 10),
                   w, b, stride=1, padding=1, groups=B)
    ref = ref.permute([1, 0, 2, 3]).reshape(B, 24, 10, 10)
    rel = (out.float() - ref).abs().max() / (ref.abs().max() + 1e-6)
    assert rel < 0.05, rel.item()
    out.sum().backward()
    assert x.grad is not None and m.weight.grad is not None



def test_maxpool2x2_even_dims_odd_channels():
    """2x2 pool fwd+bwd vs torch on even spatial dims with odd channel
    counts (the NHWC alignment corner)."""
    torch.manual_seed(31)
    for N, C, H, W in ((4, 65, 28, 28), (8, 120, 10, 10), (2, 7, 8, 12)):
        x = to_bf16_representable(torch.randn(N, C, H, W)).cuda().bfloat16()
        y, codev = ops.ext().maxpool2x2_fwd(cl(x))
        ye = F.max_pool2d(x.float().cpu(), 2)
        assert torch.allclose(y.float().cpu(), ye, atol=1e-2), (N, C, H, W)
        g = to_bf16_representable(torch.randn_like(ye)).cuda().bfloat16()
        gx = ops.ext().maxpool2x2_bwd(cl(g), codev, H, W)
        xr = x.float().cpu().requires_grad_(True)
        F.max_pool2d(xr, 2).backward(g.float().cpu())
        assert torch.allclose(gx.float().cpu(), xr.grad, atol=1e-2), (N, C)


def test_fused_optimizers_channels_last_alignment():
    """Fused SGD/AdamW on CHANNELS_LAST params must update each element
    with ITS OWN gradient: regression for the wrapper's plain
    .contiguous() silently re-ordering channels_last grads to NCHW (every
    cl-trained model mis-learned; the NCHW driver was unaffected)."""
    torch.manual_seed(41)
    from noisynet_amd import optim as native_optim
    for opt_cls, torch_cls, kw in (
            (native_optim.SGD, torch.optim.SGD,
             dict(lr=0.05, momentum=0.9, nesterov=True)),
            (native_optim.AdamW, torch.optim.AdamW, dict(lr=0.05))):
        w0 = torch.randn(16, 8, 3, 3)
        p_cl = torch.nn.Parameter(
            w0.clone().cuda().contiguous(memory_format=torch.channels_last))
        p_ref = torch.nn.Parameter(w0.clone().cuda())
        opt = opt_cls([p_cl], **kw)
        ref = torch_cls([p_ref], **kw)
        for i in range(3):
            g = torch.randn(16, 8, 3, 3).cuda()
            # grads arrive channels_last (autograd mirrors the producer)
            p_cl.grad = g.contiguous(memory_format=torch.channels_last)
            p_ref.grad = g.clone()
            opt.step()
            ref.step()
        d = (p_cl.detach().float().contiguous() -
             p_ref.detach().float()).abs().max().item()
        assert d < 1e-4, (opt_cls.__name__, d)
        # and NCHW grads on cl params must also align
        g = torch.randn(16, 8, 3, 3).cuda()
        p_cl.grad = g.clone()
        p_ref.grad = g.clone()
        opt.step()
        ref.step()
        d = (p_cl.detach().float().contiguous() -
             p_ref.detach().float()).abs().max().item()
        assert d < 1e-4, (opt_cls.__name__, 'nchw-grad', d)
