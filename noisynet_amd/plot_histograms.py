"""Histogram capture / plotting of layer internals.

Capability parity with reference plot_histograms.py: per-layer capture of
input/weights/VMM outputs plus positive/negative-separated partial sums and
per-block "source current" products (get_layers, plot_histograms.py:12-239),
and matplotlib histogram grids (plot/place_fig/plot_grid/plot_layers,
:379-586). matplotlib is imported lazily with the Agg backend so headless
and matplotlib-less environments can still run capture + .npy export.
"""

import os

import numpy as np
import torch
import torch.nn.functional as F


def _np(t):
    return t.detach().cpu().numpy() if isinstance(t, torch.Tensor) else np.asarray(t)


def get_layers(arrays, input, weight, output, stride=1, padding=0,
               layer='conv', basic=False, debug=False, block_size=None):
    """Capture input / weights / vmm (+ pos/neg-separated partials and
    blocked source-current sums) for one layer into ``arrays``.

    Mirrors reference get_layers semantics: appends, in order,
    [input], [weights], [vmm] and (unless basic) [vmm diff] = pos-neg
    separated difference and per-block weighted input sums for
    block sizes {full, 128, 64, 32} (or the single requested block_size).
    """
    x = input.detach()
    w = weight.detach()
    y = output.detach()
    arrays.append([_np(x.half())])
    arrays.append([_np(w.half())])
    arrays.append([_np(y.half())])
    if basic:
        return

    w_pos = w.clamp(min=0)
    w_neg = (-w).clamp(min=0)
    if layer == 'conv':
        out_pos = F.conv2d(x, w_pos, None, stride, padding)
        out_neg = F.conv2d(x, w_neg, None, stride, padding)
    else:
        out_pos = F.linear(x, w_pos)
        out_neg = F.linear(x, w_neg)
    vmm_diff = out_pos - out_neg
    arrays.append([_np(vmm_diff.half())])

    # "source current" products: per-block sums of |W| columns times input
    if layer == 'conv':
        x2 = F.unfold(x, w.shape[-1], padding=padding, stride=stride)  # B, C*R*S, L
        x2 = x2.transpose(1, 2).reshape(-1, w.shape[1] * w.shape[2] * w.shape[3])
        wf = w.reshape(w.shape[0], -1)
    else:
        x2 = x.reshape(-1, w.shape[1])
        wf = w
    sizes = [0, 128, 64, 32] if block_size is None else [block_size]
    for bs in sizes:
        fan_in = wf.shape[1]
        b = fan_in if bs == 0 else min(bs, fan_in)
        srcs = []
        srcs_diff = []
        for start in range(0, fan_in, b):
            blk_w = wf[:, start:start + b]
            blk_x = x2[:, start:start + b]
            srcs.append(blk_x.matmul(blk_w.abs().t()))
            srcs_diff.append(blk_x.matmul(blk_w.clamp(min=0).t())
                             - blk_x.matmul((-blk_w).clamp(min=0).t()))
        srcs = torch.stack(srcs)
        srcs_diff = torch.stack(srcs_diff)
        arrays.append([_np(srcs.half())])
        if block_size is None:
            pass
        else:
            arrays.append([_np(srcs_diff.half())])
    if block_size is None:
        # diffs for each block size, appended after the four source arrays
        for bs in sizes:
            fan_in = wf.shape[1]
            b = fan_in if bs == 0 else min(bs, fan_in)
            diffs = []
            for start in range(0, fan_in, b):
                blk_w = wf[:, start:start + b]
                blk_x = x2[:, start:start + b]
                diffs.append(blk_x.matmul(blk_w.clamp(min=0).t())
                             - blk_x.matmul((-blk_w).clamp(min=0).t()))
            arrays.append([_np(torch.stack(diffs).half())])


def _plt():
    import matplotlib
    matplotlib.use('Agg')
    import matplotlib.pyplot as plt
    return plt


def plot(values, values2=None, bins=120, range_=None, labels=None, title='',
         log=True, path=None):
    """Single/dual overlaid histogram (reference plot_histograms.py:379)."""
    plt = _plt()
    plt.figure(figsize=(8, 6))
    v = np.asarray(values).astype(np.float32).ravel()
    if range_ is None:
        lo, hi = float(v.min()), float(v.max())
        if values2 is not None:
            v2 = np.asarray(values2).astype(np.float32).ravel()
            lo, hi = min(lo, float(v2.min())), max(hi, float(v2.max()))
        range_ = (lo, hi)
    alpha = 0.5 if values2 is not None else 1.0
    plt.hist(v, bins=bins, range=range_, alpha=alpha, color='b',
             label=(labels[0] if labels else None))
    if values2 is not None:
        plt.hist(np.asarray(values2).astype(np.float32).ravel(), bins=bins,
                 range=range_, alpha=alpha, color='r',
                 label=(labels[1] if labels and len(labels) > 1 else None))
    plt.title(title, fontsize=16)
    if labels:
        plt.legend(loc='upper right')
    if log:
        plt.semilogy()
    if path:
        plt.savefig(path if path.endswith('.png') else path + '.png',
                    dpi=120, bbox_inches='tight')
    plt.close()


_COLORS = ['blue', 'red', 'green', 'black', 'magenta', 'cyan', 'orange',
           'yellow', 'gray']


def place_fig(arrays, rows=1, columns=1, r=0, c=0, bins=100, range_=None,
              title=None, name=None, infos=None, labels=None, log=True):
    """One grid cell: overlaid step-histograms of one quantity across
    models/variants (reference place_fig :413-454). Each curve is labelled
    with its (min, max) span."""
    plt = _plt()
    ax = plt.subplot2grid((rows, columns), (r, c))
    flat = [np.asarray(a).astype(np.float32).ravel() for a in arrays]
    flat = [f[np.isfinite(f)] for f in flat]
    flat = [f for f in flat if f.size]
    if not flat:
        return ax
    if range_ is None and len(flat) > 1:
        range_ = (min(float(f.min()) for f in flat),
                  max(float(f.max()) for f in flat))
    histtype = 'bar' if len(flat) == 1 else 'step'
    for f, color in zip(flat, _COLORS):
        ax.hist(f, bins=bins, range=range_, histtype=histtype, color=color,
                linewidth=1.5,
                label='({:.1f}, {:.1f})'.format(float(f.min()),
                                                float(f.max())))
    ax.set_title((title or '') + (name or ''), fontsize=14)
    if log:
        ax.semilogy()
    ax.legend(loc='best', prop={'size': 9})
    return ax


def plot_grid(layers, names, path, filename='', info=None, pctl=99.9,
              labels=None, normalize=False):
    """Rows = layers, columns = quantities; each cell overlays all models.

    ``normalize`` reproduces the reference's crossbar normalization
    (plot_grid :457-510): inputs scaled by max input, weights by max |w|,
    weight-sum diffs by the same |w| threshold, input sums by max input,
    everything else by their product.
    """
    plt = _plt()
    rows = len(layers)
    columns = min(len(names), len(layers[0]))
    plt.figure(figsize=(columns * 7, rows * 6))
    for r, layer in enumerate(layers):
        max_input = thr = 1.0
        for c in range(columns):
            name = names[c]
            cell = [np.asarray(a, dtype=np.float32) for a in layer[c]]
            if normalize:
                if name == 'input':
                    max_input = max(float(np.max(np.abs(cell[0]))), 1e-8)
                    cell = [a / max_input for a in cell]
                elif name == 'weights':
                    thr = max(float(np.max(np.abs(cell[0]))), 1e-8)
                    cell = [a / thr for a in cell]
                elif 'weight sums diff' in name:
                    cell = [a / thr for a in cell]
                elif 'input sums' in name or 'source' in name:
                    cell = [a / max_input for a in cell]
                else:
                    cell = [a / (max_input * thr) for a in cell]
            place_fig(cell, rows=rows, columns=columns, r=r, c=c,
                      title='layer%d ' % r, name=name,
                      infos=(info[r] if info else None), labels=labels)
    out = os.path.join(path, filename) if filename else path
    plt.savefig(out, dpi=100, bbox_inches='tight')
    plt.close()
    print('saved histogram grid to', out)
    return out


def _acc_from_dir(model_dir):
    """Best accuracy encoded in the checkpoint filename
    (model_epoch_{e}_acc_{a}.pth, reference noisynet.py:1636)."""
    try:
        for fname in os.listdir(model_dir):
            if 'model' in fname and fname.endswith('.pth'):
                return float(fname.rsplit('_', 1)[-1][:-4])
    except OSError:
        pass
    return 0.0


def plot_layers(num_layers=4, models=None, epoch=0, i=0, layers=None,
                names=None, var='', vars=None, infos=None, pctl=99.9,
                acc=0.0, tag='', normalize=False):
    """Histogram grid for one capture, or a comparison across several saved
    result dirs (reference plot_layers :513-586).

    Multi-model mode (len(models) > 1): each dir must contain the .npy
    artifacts written by ``--write`` (layers.npy, array_names.npy,
    input_sizes.npy, optionally layer_power.npy); the grids are overlaid
    per cell and labelled "var value (acc%)".
    """
    vars = vars if vars is not None else [0.0]
    accs = [acc]
    infos_out = infos

    if models is not None and len(models) > 1:
        names = list(np.load(os.path.join(models[0], 'array_names.npy'),
                             allow_pickle=True))
        layers = [[[] for _ in names] for _ in range(num_layers)]
        accs, input_sizes, powers = [], [], []
        for model_dir in models:
            accs.append(_acc_from_dir(model_dir))
            saved = np.load(os.path.join(model_dir, 'layers.npy'),
                            allow_pickle=True)
            input_sizes.append(np.load(
                os.path.join(model_dir, 'input_sizes.npy'),
                allow_pickle=True))
            ppath = os.path.join(model_dir, 'layer_power.npy')
            if os.path.exists(ppath):
                powers.append(np.load(ppath, allow_pickle=True))
            for l in range(num_layers):
                for col in range(min(len(names), len(saved[l]))):
                    layers[l][col].append(saved[l][col][0])
        infos_out = []
        for l in range(num_layers):
            row = []
            for mi in range(len(models)):
                entry = ['%d inputs\n' % int(input_sizes[mi][l])]
                if powers:
                    entry.append('%.2fmW ' % float(powers[mi][l]))
                row.append(entry)
            infos_out.append(row)

    labels = ['%s %s (%.1f%%)' % (var, str(v), a)
              for v, a in zip(list(vars) + [0.0] * len(accs), accs)]

    if models is not None and len(models) > 1:
        filename = 'comparison_of_%s%s.png' % (var, tag)
        out_dir = models[0]
    else:
        filename = 'epoch_%d_iter_%d_acc_%.2f_%s.png' % (epoch, i, acc, tag)
        out_dir = (models[0] if models else '.')
    return plot_grid(layers, names, out_dir, filename=filename,
                     info=infos_out, pctl=pctl, labels=labels,
                     normalize=normalize)


_CAPTURE_NAMES = ['input', 'weights', 'vmm', 'vmm diff',
                  'input sums', 'input sums 128', 'input sums 64',
                  'input sums 32', 'weight sums diff',
                  'weight sums diff 128', 'weight sums diff 64',
                  'weight sums diff 32']


def capture_and_emit(model, args, arrays, epoch, i, s, acc):
    """Driver-side hook for Net._forward_reference (reference
    noisynet.py:601-694): run get_layers over the four layers, then plot
    (--plot) and/or persist the .npy artifact set (--write) that
    multi-model ``plot_layers`` comparison consumes."""
    if not (args.plot or args.write):
        return
    specs = [
        (model.input, model.conv1.weight, model.conv1_, 'conv'),
        (model.relu1, model.conv2.weight, model.conv2_, 'conv'),
        (model.relu2, model.linear1.weight, model.linear1_, 'linear'),
        (model.relu3, model.linear2.weight, model.linear2_, 'linear'),
    ]
    layers = []
    input_sizes = []
    for x, w, y, kind in specs:
        per_layer = []
        get_layers(per_layer, x.float(), w.float(), y.float(), stride=1,
                   padding=0, layer=kind)
        layers.append(per_layer)
        input_sizes.append(int(np.prod(w.shape[1:])))
    names = _CAPTURE_NAMES[:len(layers[0])]

    out_dir = getattr(args, 'checkpoint_dir', '') or 'results/plots'
    os.makedirs(out_dir, exist_ok=True)
    if args.write:
        np.save(os.path.join(out_dir, 'layers.npy'),
                np.array(layers, dtype=object), allow_pickle=True)
        np.save(os.path.join(out_dir, 'array_names.npy'),
                np.array(names, dtype=object), allow_pickle=True)
        np.save(os.path.join(out_dir, 'input_sizes.npy'),
                np.array(input_sizes), allow_pickle=True)
        power = [float(np.mean(p)) if len(p) else 0.0
                 for p in getattr(model, 'power', [])]
        if power:
            np.save(os.path.join(out_dir, 'layer_power.npy'),
                    np.array(power), allow_pickle=True)
        print('arrays saved to', os.path.join(out_dir, 'layers.npy'))
    if args.plot:
        try:
            plot_layers(num_layers=len(layers), models=[out_dir],
                        epoch=epoch, i=i, layers=layers, names=names,
                        acc=acc, tag='capture')
        except Exception as exc:  # headless/matplotlib-less environments
            print('plotting unavailable (%r); use --write for .npy export'
                  % (exc,))


def compare_result_dirs(dirs, out='comparison.png', var='', vars=None,
                        pctl=99.9, normalize=False, tag=''):
    """Compare several result dirs saved with --write (reference
    plot_histograms.py __main__ :589-605)."""
    dirs = [d for d in dirs
            if os.path.isfile(os.path.join(d, 'layers.npy'))]
    if len(dirs) < 2:
        print('need >= 2 result dirs containing layers.npy')
        return None
    return plot_layers(num_layers=4, models=dirs, var=var,
                       vars=vars if vars is not None else list(range(len(dirs))),
                       pctl=pctl, tag=tag or '_cmp', normalize=normalize)


if __name__ == '__main__':
    import sys
    compare_result_dirs(sys.argv[1:] or ['.'])
