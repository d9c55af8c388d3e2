"""CPU tests for the model zoo: ResNet-18, MobileNetV2, EfficientNet family,
conv variant layers, activations, timm subsystems."""

import numpy as np
import pytest
import torch

from noisynet_amd.config import build_main_parser


def margs(extra=None):
    argv = ['-a', 'resnet18', '--q_a', '0']
    if extra:
        argv += extra
    return build_main_parser().parse_args(argv)


def test_resnet18_forward_backward():
    from noisynet_amd.models.resnet import ResNet18
    args = margs(['--q_a', '4', '--calculate_running'])
    m = ResNet18(args)
    m.train()
    x = torch.rand(2, 3, 224, 224)
    out = m(x)
    assert out.shape == (2, 1000)
    out.sum().backward()
    assert m.conv1.weight.grad is not None
    assert m.layer4[1].conv2.weight.grad is not None


def test_resnet18_act_max_hardtanh():
    from noisynet_amd.models.resnet import ResNet18
    args = margs(['--act_max', '4'])
    m = ResNet18(args)
    m.eval()
    x = torch.rand(2, 3, 224, 224)
    with torch.no_grad():
        m(x)  # runs the clip path


def test_mobilenet_v2_forward_backward():
    from noisynet_amd.models.mobilenet import mobilenet_v2
    args = margs(['-a', 'mobilenet_v2', '--q_a', '4', '--calculate_running'])
    m = mobilenet_v2(args)
    m.train()
    x = torch.rand(2, 3, 224, 224)
    out = m(x)
    assert out.shape == (2, 1000)
    out.sum().backward()
    assert m.features[0].conv.weight.grad is not None


def test_efficientnet_mini_local():
    from noisynet_amd.models.efficientnet import efficientnet_b0
    args = margs(['-a', 'efficientnet_b0'])
    m = efficientnet_b0(args)
    m.train()
    x = torch.rand(2, 3, 64, 64)
    out = m(x)
    assert out.shape == (2, 1000)
    out.sum().backward()


def test_timm_registry_and_family():
    from noisynet_amd.timm.models import create_model, is_model, list_models
    assert is_model('efficientnet_b0')
    models = list_models('efficientnet*')
    assert 'efficientnet_b0' in models and 'efficientnet_b7' in models
    m0 = create_model('efficientnet_b0', num_classes=10)
    m1 = create_model('efficientnet_b1', num_classes=10)
    # b1 is deeper than b0
    n0 = sum(1 for _ in m0.blocks.modules())
    n1 = sum(1 for _ in m1.blocks.modules())
    assert n1 > n0
    out = m0(torch.rand(2, 3, 96, 96))
    assert out.shape == (2, 10)


def test_efficientnet_gradients_healthy_at_init():
    """Regression: dw-conv goog init (fan_out//groups); grads bounded."""
    from noisynet_amd.timm.models import create_model
    torch.manual_seed(0)
    m = create_model('efficientnet_b0', num_classes=1000)
    m.train()
    out = m(torch.rand(2, 3, 224, 224))
    loss = torch.nn.functional.cross_entropy(out, torch.randint(0, 1000, (2,)))
    loss.backward()
    worst = max(p.grad.abs().max().item() for p in m.parameters()
                if p.grad is not None)
    # the dw-conv fan_out bug produced ~1e14; healthy init lands around
    # 1e5-1e6 depending on the random draw, so the bound is order-of-
    # magnitude, not tight
    assert worst < 1e8, worst


def test_efficientnet_features_backbone():
    from noisynet_amd.timm.models.efficientnet import EfficientNetFeatures
    m = EfficientNetFeatures(out_indices=(0, 2, 4))
    m.eval()
    feats = m(torch.rand(1, 3, 64, 64))
    assert len(feats) == 3
    assert feats[0].shape[2] > feats[1].shape[2] > feats[2].shape[2]


def test_mobilenetv3_and_edge():
    from noisynet_amd.timm.models import create_model
    for name in ('mobilenetv3_large_100', 'efficientnet_es'):
        m = create_model(name, num_classes=10)
        m.eval()
        out = m(torch.rand(1, 3, 64, 64))
        assert out.shape == (1, 10)


def test_condconv_and_mixedconv():
    from noisynet_amd.models.conv2d_layers import (CondConv2d, MixedConv2d,
                                                   Conv2dSame, select_conv2d)
    x = torch.rand(3, 16, 14, 14)
    cc = CondConv2d(16, 24, 3, padding='same', num_experts=4)
    routing = torch.softmax(torch.rand(3, 4), dim=1)
    out = cc(x, routing)
    assert out.shape == (3, 24, 14, 14)
    # per-sample experts: different routing -> different output
    routing2 = torch.softmax(torch.rand(3, 4), dim=1)
    out2 = cc(x, routing2)
    assert not torch.allclose(out, out2)

    mc = MixedConv2d(16, 16, [3, 5], padding='', depthwise=True)
    assert mc(x).shape[1] == 16

    cs = Conv2dSame(16, 8, 3, stride=2)
    assert cs(x).shape == (3, 8, 7, 7)

    c = select_conv2d(16, 8, 3, padding='same', stride=1)
    assert c(x).shape == (3, 8, 14, 14)


def test_activations_match_torch():
    import torch.nn.functional as F
    from noisynet_amd import ops
    x = torch.randn(100, requires_grad=True)
    y = ops.swish(x)
    assert torch.allclose(y, x * torch.sigmoid(x), atol=1e-6)
    y.sum().backward()
    x2 = x.detach().clone().requires_grad_(True)
    (x2 * torch.sigmoid(x2)).sum().backward()
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)

    x = torch.randn(100)
    assert torch.allclose(ops.hard_swish(x), F.hardswish(x), atol=1e-6)
    assert torch.allclose(ops.hard_sigmoid(x), F.hardsigmoid(x), atol=1e-6)
    assert torch.allclose(ops.mish(x), F.mish(x), atol=1e-5)


def test_timm_optimizers():
    from types import SimpleNamespace
    from noisynet_amd.timm.optim import create_optimizer
    model = torch.nn.Linear(10, 10)
    for opt_name in ('sgd', 'adam', 'adamw', 'radam', 'nadam', 'novograd',
                     'rmsproptf', 'lookahead_sgd'):
        args = SimpleNamespace(opt=opt_name, lr=0.01, weight_decay=1e-4,
                               momentum=0.9, opt_eps=1e-8)
        opt = create_optimizer(args, model)
        loss = model(torch.rand(4, 10)).sum()
        opt.zero_grad()
        loss.backward()
        opt.step()


def test_timm_schedulers():
    from types import SimpleNamespace
    from noisynet_amd.timm.scheduler import create_scheduler
    model = torch.nn.Linear(4, 4)
    for sched in ('cosine', 'tanh', 'step', 'plateau'):
        opt = torch.optim.SGD(model.parameters(), lr=0.1)
        args = SimpleNamespace(epochs=10, sched=sched, min_lr=1e-5,
                               decay_rate=0.1, warmup_lr=1e-4,
                               warmup_epochs=2, cooldown_epochs=1,
                               decay_epochs=3, patience_epochs=2)
        s, n = create_scheduler(args, opt)
        s.step(1, 0.5)
        if sched == 'cosine':
            s.step_update(5)


def test_timm_losses():
    from noisynet_amd.timm.loss import (LabelSmoothingCrossEntropy,
                                        SoftTargetCrossEntropy)
    x = torch.randn(8, 10)
    y = torch.randint(0, 10, (8,))
    l1 = LabelSmoothingCrossEntropy(0.1)(x, y)
    assert torch.isfinite(l1)
    soft = torch.softmax(torch.randn(8, 10), dim=1)
    l2 = SoftTargetCrossEntropy()(x, soft)
    assert torch.isfinite(l2)


def test_timm_utils_checkpoint_saver(tmp_path):
    from types import SimpleNamespace
    from noisynet_amd.timm.utils import CheckpointSaver, ModelEma
    model = torch.nn.Linear(4, 4)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    args = SimpleNamespace(model='testnet')
    saver = CheckpointSaver(checkpoint_dir=str(tmp_path),
                            recovery_dir=str(tmp_path), max_history=2)
    for epoch, metric in enumerate([10.0, 30.0, 20.0]):
        saver.save_checkpoint(model, opt, args, epoch, metric=metric)
    import os
    files = os.listdir(tmp_path)
    assert 'model_best.pth.tar' in files
    # top-2 history kept
    assert len([f for f in files if f.startswith('checkpoint-')]) == 2
    assert saver.best_metric == 30.0
    saver.save_recovery(model, opt, args, 3, batch_idx=7)
    assert saver.find_recovery()

    ema = ModelEma(model, decay=0.5)
    with torch.no_grad():
        model.weight.add_(1.0)
    ema.update(model)
    assert not torch.allclose(ema.ema.weight, model.weight)


def test_timm_mixup_and_sampler():
    from noisynet_amd.timm.data import FastCollateMixup, mixup_target
    from noisynet_amd.timm.data.distributed_sampler import OrderedDistributedSampler
    t = mixup_target(torch.tensor([1, 2]), 10, lam=0.7, smoothing=0.1,
                     device='cpu')
    assert t.shape == (2, 10)
    assert torch.allclose(t.sum(1), torch.ones(2))

    collate = FastCollateMixup(mixup_alpha=1.0, label_smoothing=0.1,
                               num_classes=10)
    batch = [(np.random.randint(0, 255, (3, 8, 8), dtype=np.uint8), i % 10)
             for i in range(4)]
    x, y = collate(batch)
    assert x.shape == (4, 3, 8, 8) and y.shape == (4, 10)

    class DS:
        def __len__(self):
            return 10
    s = OrderedDistributedSampler(DS(), num_replicas=4, rank=1)
    assert len(list(iter(s))) == 3


def test_timm_dataset_synthetic_loader():
    from noisynet_amd.timm.data import Dataset, create_loader, fast_collate
    ds = Dataset('')  # no dir -> synthetic
    ds._synthetic.num_samples = 8
    ds.samples = [('synthetic', 0)] * 8
    loader = create_loader(ds, input_size=(3, 32, 32), batch_size=4,
                           is_training=True, use_prefetcher=False,
                           num_workers=0)
    x, y = next(iter(loader))
    assert x.shape[0] == 4


def test_main_driver_efficientnet_arch(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    from noisynet_amd.drivers import imagenet
    imagenet.main(['-a', 'efficientnet_b0', '--epochs', '1',
                   '--batch_size', '2', '--synthetic_batches', '2',
                   '--q_a', '0', '-p', '1'])


def test_train_efficientnet_driver(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    from noisynet_amd.drivers import efficientnet_train
    efficientnet_train.main(['--model', 'efficientnet_b0', '--epochs', '1',
                             '-b', '2', '--synthetic-batches', '2',
                             '--sched', 'step', '--cooldown-epochs', '0',
                             '--no-prefetcher', '--workers', '0',
                             '--output', str(tmp_path / 'out')])


def test_chip_mnist_driver(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    from noisynet_amd.drivers import mnist
    res = mnist.main(['--epochs', '1', '--n_train', '500', '--n_test', '100',
                      '--L3', '0.0'])
    assert res


def test_auto_augment_and_randaugment():
    from noisynet_amd.timm.data import (AutoAugment, rand_augment_transform)
    img = np.random.randint(0, 255, (3, 32, 32), dtype=np.uint8)
    aa = AutoAugment()
    out = aa(img.copy())
    assert out.shape == img.shape and out.dtype == np.uint8
    ra = rand_augment_transform('rand-m9-mstd0.5-n2')
    out = ra(img.copy())
    assert out.shape == img.shape and out.dtype == np.uint8
    # individual ops sanity
    from noisynet_amd.timm.data.auto_augment import (equalize, posterize,
                                                     rotate, solarize)
    assert solarize(img, 128).shape == img.shape
    assert posterize(img, 4).max() <= 255
    assert rotate(img, 15.0).shape == img.shape
    assert equalize(img).shape == img.shape


def test_median_pool_and_feature_hooks():
    from noisynet_amd.timm.models.median_pool import MedianPool2d
    x = torch.rand(1, 3, 8, 8)
    mp = MedianPool2d(3, 1, 1)
    assert mp(x).shape == x.shape

    from noisynet_amd.timm.models.feature_hooks import FeatureHooks
    net = torch.nn.Sequential(torch.nn.Conv2d(3, 4, 3), torch.nn.ReLU())
    hooks = FeatureHooks([{'name': '0'}], net.named_modules())
    y = net(x)
    feats = hooks.get_output(x.device)
    assert len(feats) == 1 and feats[0].shape[1] == 4


def test_gpu_augment_cpu_path():
    """The CIFAR train-loop augmentation (random 32x32 crop from the 40x40
    padded tensor + hflip, noisynet.py:1264-1269) runs on any device."""
    from noisynet_amd import data as data_mod
    x = torch.rand(8, 3, 40, 40)
    out = data_mod.gpu_augment(x)
    assert out.shape == (8, 3, 32, 32)
    assert torch.isfinite(out).all()


def test_prefetch_loader_cpu():
    """PrefetchLoader normalizes and yields batches; without CUDA it runs
    the same pipeline synchronously."""
    from noisynet_amd.timm.data import SyntheticImageDataset, create_loader
    ds = SyntheticImageDataset(num_samples=16, size=32, num_classes=10)
    loader = create_loader(ds, input_size=(3, 32, 32), batch_size=8,
                           is_training=True, use_prefetcher=True,
                           num_workers=0)
    batches = list(loader)
    assert len(batches) >= 1
    x, y = batches[0]
    assert x.shape[1:] == (3, 32, 32)


def test_random_erasing_modes():
    from noisynet_amd.timm.data import RandomErasing
    for mode in ('const', 'rand', 'pixel'):
        re_op = RandomErasing(probability=1.0, mode=mode, device='cpu')
        x = torch.rand(4, 3, 16, 16)
        out = re_op(x.clone())
        assert out.shape == x.shape


def test_dataset_tar_scanner(tmp_path):
    """DatasetTar indexes a class-per-dir image tarball and serves samples."""
    import io
    import tarfile

    import numpy as np
    from PIL import Image

    from noisynet_amd.timm.data import DatasetTar

    tar_path = tmp_path / "imgs.tar"
    with tarfile.open(tar_path, "w") as tf:
        for cls in ("dogs", "cats"):
            for i in range(3):
                buf = io.BytesIO()
                Image.fromarray(
                    np.full((8, 8, 3), i * 20, dtype=np.uint8)).save(
                        buf, format="PNG")
                data = buf.getvalue()
                info = tarfile.TarInfo("%s/im%d.png" % (cls, i))
                info.size = len(data)
                tf.addfile(info, io.BytesIO(data))

    ds = DatasetTar(str(tar_path))
    assert len(ds) == 6
    assert ds.class_to_idx == {"cats": 0, "dogs": 1}
    img, target = ds[0]
    assert target in (0, 1)
    assert img.size == (8, 8)
    raw, _ = DatasetTar(str(tar_path), load_bytes=True)[1]
    assert isinstance(raw, bytes) and raw[:4] == b"\x89PNG"
    names = ds.filenames(basename=True)
    assert "im0.png" in names


def test_nvnovograd_step():
    """NvNovoGrad reduces a quadratic and matches reference semantics
    (layer-wise scalar second moment, first step seeded with grad norm)."""
    import torch

    from noisynet_amd.timm.optim import NvNovoGrad

    torch.manual_seed(0)
    w = torch.nn.Parameter(torch.randn(4, 4))
    opt = NvNovoGrad([w], lr=0.05, weight_decay=0.01)
    losses = []
    for _ in range(30):
        loss = (w ** 2).sum()
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < 0.2 * losses[0]
    state = opt.state[w]
    assert state['exp_avg_sq'].dim() == 0  # layer-wise scalar moment


def test_condconv_per_sample_matches_grouped_oracle():
    """CondConv2d's unfold+batched-GEMM path (K13 native route) must equal
    the reference's batch*groups grouped-conv formulation."""
    import torch
    import torch.nn.functional as F

    from noisynet_amd.models.conv2d_layers import CondConv2d, conv2d_same

    torch.manual_seed(0)
    for groups, pad in [(1, ''), (1, 1), (4, 1)]:
        m = CondConv2d(8, 16, kernel_size=3, stride=1, padding=pad,
                       groups=groups, bias=True, num_experts=4)
        x = torch.randn(5, 8, 10, 10, requires_grad=True)
        rw = torch.softmax(torch.randn(5, 4), dim=1)
        out = m(x, rw)
        B = 5
        w = torch.matmul(rw, m.weight).view(
            B * m.out_channels, m.in_channels // m.groups, 3, 3)
        b = torch.matmul(rw, m.bias).view(B * m.out_channels)
        x2 = x.view(1, B * 8, 10, 10)
        if m.dynamic_padding:
            ref = conv2d_same(x2, w, b, stride=m.stride, padding=m.padding,
                              dilation=(1, 1), groups=m.groups * B)
        else:
            ref = F.conv2d(x2, w, b, stride=m.stride, padding=m.padding,
                           dilation=(1, 1), groups=m.groups * B)
        ref = ref.permute([1, 0, 2, 3]).reshape(B, m.out_channels,
                                                ref.shape[-2], ref.shape[-1])
        assert (out - ref).abs().max().item() < 1e-5, (groups, pad)
        out.sum().backward()
        assert x.grad is not None and m.weight.grad is not None


def test_plot_surface_roundtrip(tmp_path):
    """--write artifacts (layers/array_names/input_sizes/layer_power .npy)
    round-trip through the multi-model comparison plot (reference
    plot_histograms.py:513-605)."""
    import numpy as np
    import torch

    from noisynet_amd import plot_histograms as ph

    # synthesize two "result dirs" of capture artifacts
    dirs = []
    for run in range(2):
        d = tmp_path / ("run%d" % run)
        d.mkdir()
        torch.manual_seed(run)
        layers = []
        sizes = []
        for l in range(4):
            per_layer = []
            x = torch.randn(2, 3, 8, 8)
            w = torch.randn(4, 3, 3, 3) * 0.2
            y = torch.nn.functional.conv2d(x, w)
            ph.get_layers(per_layer, x, w, y, stride=1, padding=0,
                          layer='conv')
            layers.append(per_layer)
            sizes.append(int(np.prod(w.shape[1:])))
        names = ph._CAPTURE_NAMES[:len(layers[0])]
        np.save(d / 'layers.npy', np.array(layers, dtype=object),
                allow_pickle=True)
        np.save(d / 'array_names.npy', np.array(names, dtype=object),
                allow_pickle=True)
        np.save(d / 'input_sizes.npy', np.array(sizes), allow_pickle=True)
        np.save(d / 'layer_power.npy', np.arange(4, dtype=np.float64),
                allow_pickle=True)
        (d / 'model_epoch_3_acc_77.10.pth').write_bytes(b'')
        dirs.append(str(d))

    out = ph.compare_result_dirs(dirs, var='seed', vars=[0, 1])
    assert out is not None
    import os
    assert os.path.exists(out)
    # and the single-capture grid path
    out2 = ph.plot_layers(num_layers=4, models=[dirs[0]], epoch=1, i=2,
                          layers=np.load(os.path.join(dirs[0], 'layers.npy'),
                                         allow_pickle=True),
                          names=list(np.load(
                              os.path.join(dirs[0], 'array_names.npy'),
                              allow_pickle=True)), acc=50.0, tag='t')
    assert os.path.exists(out2)


def test_tf_preprocessing_transform():
    """TF-exact preprocessing without TF: padded center crop + bicubic
    (eval) and distorted-bbox random crop (train), CHW uint8 out."""
    import numpy as np
    from PIL import Image

    from noisynet_amd.timm.data.tf_preprocessing import (
        TfPreprocessTransform, _center_crop_box)

    img = Image.fromarray(
        (np.random.RandomState(0).rand(300, 400, 3) * 255).astype('uint8'))
    ev = TfPreprocessTransform(is_training=False, size=224)
    out = ev(img)
    assert out.shape == (3, 224, 224) and out.dtype == np.uint8
    # the eval crop is the reference's size/(size+32)*min(h,w) center box
    box = _center_crop_box(400, 300, 224)
    assert box[2] - box[0] == box[3] - box[1] == int(224 / 256 * 300)
    tr = TfPreprocessTransform(is_training=True, size=224)
    out2 = tr(img)
    assert out2.shape == (3, 224, 224)
    # bytes input path (the reference feeds encoded jpeg bytes)
    import io
    buf = io.BytesIO()
    img.save(buf, format='JPEG')
    out3 = ev(buf.getvalue())
    assert out3.shape == (3, 224, 224)


def test_optim_factory_fused_names():
    """apex Fused* names resolve to the native fused-kernel optimizers
    (reference optim_factory.py:75-97 requires apex; ours ARE fused)."""
    import types

    import torch

    from noisynet_amd.timm.optim.optim_factory import create_optimizer

    m = torch.nn.Linear(4, 4)
    expected = {'fusedsgd': 'SGD', 'fusedadam': 'Adam',
                'fusedadamw': 'AdamW', 'fusednovograd': 'NvNovoGrad'}
    for name, cls in expected.items():
        args = types.SimpleNamespace(opt=name, weight_decay=1e-4, lr=0.01,
                                     momentum=0.9, opt_eps=1e-8)
        opt = create_optimizer(args, m)
        assert type(opt).__name__ == cls, name
        loss = m(torch.randn(2, 4)).sum()
        opt.zero_grad()
        loss.backward()
        opt.step()


def test_nvnovograd_amsgrad_branch():
    """NvNovoGrad AMSGrad variant: the max-of-second-moments denominator
    (reference nvnovograd.py:100-106)."""
    import torch

    from noisynet_amd.timm.optim import NvNovoGrad

    torch.manual_seed(1)
    w = torch.nn.Parameter(torch.randn(3, 3))
    opt = NvNovoGrad([w], lr=0.05, amsgrad=True)
    losses = []
    for _ in range(25):
        loss = (w ** 2).sum()
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < 0.3 * losses[0]
    st = opt.state[w]
    assert 'max_exp_avg_sq' in st
    assert float(st['max_exp_avg_sq']) >= float(st['exp_avg_sq']) - 1e-8


def test_scheduler_step_update_per_iteration():
    """Cosine scheduler's per-iteration step_update lowers LR smoothly
    between epoch boundaries (reference cosine_lr.py + per-iteration
    t_in_epochs=False use, train_efficientnet.py:544-545)."""
    import types

    import torch

    from noisynet_amd.timm.scheduler import create_scheduler

    m = torch.nn.Linear(2, 2)
    opt = torch.optim.SGD(m.parameters(), lr=0.4)
    args = types.SimpleNamespace(sched='cosine', epochs=4, min_lr=1e-5,
                                 warmup_lr=1e-4, warmup_epochs=1,
                                 decay_rate=0.1, decay_epochs=1,
                                 cooldown_epochs=0, lr_cycle_mul=1.0,
                                 lr_cycle_limit=1, seed=0, lr=0.4,
                                 patience_epochs=2)
    sched, num_epochs = create_scheduler(args, opt)
    assert num_epochs >= 4
    sched.step(0)
    lrs = []
    for update in range(0, 40, 10):
        sched.step_update(update)
        lrs.append(opt.param_groups[0]['lr'])
    # warmup: LR should be rising early and stay bounded by base lr
    assert lrs[0] <= lrs[-1] <= 0.4 + 1e-9
