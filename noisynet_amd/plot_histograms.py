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


def plot(values, values2=None, bins=120, range_=None, labels=None, title='',
         log=True, path=None):
    import matplotlib
    matplotlib.use('Agg')
    import matplotlib.pyplot as plt
    plt.figure(figsize=(8, 5))
    v = np.asarray(values).astype(np.float32).flatten()
    plt.hist(v, bins=bins, range=range_, log=log, alpha=0.6,
             label=(labels[0] if labels else None))
    if values2 is not None:
        v2 = np.asarray(values2).astype(np.float32).flatten()
        plt.hist(v2, bins=bins, range=range_, log=log, alpha=0.6,
                 label=(labels[1] if labels and len(labels) > 1 else None))
    if labels:
        plt.legend()
    plt.title(title)
    if path:
        plt.savefig(path + '.png', dpi=120)
        plt.close()
    else:
        plt.close()


def plot_grid(layers, names, path, pctl=99.98, normalize=False):
    import matplotlib
    matplotlib.use('Agg')
    import matplotlib.pyplot as plt
    num_layers = len(layers)
    num_cols = len(names)
    fig, axes = plt.subplots(num_layers, num_cols,
                             figsize=(3 * num_cols, 2.5 * num_layers),
                             squeeze=False)
    for li, layer in enumerate(layers):
        for ci, arr in enumerate(layer[:num_cols]):
            v = np.asarray(arr[0]).astype(np.float32).flatten()
            v = v[np.isfinite(v)]
            if v.size == 0:
                continue
            if normalize and v.std() > 0:
                v = v / max(abs(np.percentile(v, pctl)), 1e-8)
            axes[li][ci].hist(v, bins=80, log=True)
            if li == 0:
                axes[li][ci].set_title(names[ci], fontsize=8)
    fig.tight_layout()
    fig.savefig(path, dpi=100)
    plt.close(fig)


def plot_layers(num_layers, models, epoch, i, layers, names, var='', vars=None,
                infos=None, pctl=99.98, acc=0.0, tag='', normalize=False):
    """Histogram grid per model dir (reference plot_layers :513-586)."""
    for mdl in models:
        out_dir = mdl if os.path.isdir(mdl) else '.'
        path = os.path.join(out_dir, 'layers_epoch_{}_acc_{:.2f}{}.png'.format(epoch, acc, tag))
        try:
            plot_grid(layers, names, path, pctl=pctl, normalize=normalize)
            print('saved histogram grid to', path)
        except Exception as exc:  # matplotlib absent or headless failure
            npy = os.path.join(out_dir, 'layers_epoch_{}{}.npy'.format(epoch, tag))
            np.save(npy, np.array(layers, dtype=object), allow_pickle=True)
            print('plotting unavailable (%s); arrays saved to %s' % (exc, npy))


def capture_and_emit(model, args, arrays, epoch, i, s, acc):
    """Driver-side hook for Net._forward_reference: save .npy / plot PNG
    when --plot/--write is set (reference noisynet.py:601-694)."""
    if not (args.plot or args.write):
        return
    out_dir = getattr(args, 'checkpoint_dir', 'results/plots')
    os.makedirs(out_dir, exist_ok=True)
    if args.write:
        np.save(os.path.join(out_dir, 'layers.npy'),
                np.array(arrays, dtype=object), allow_pickle=True)
        print('arrays saved to', os.path.join(out_dir, 'layers.npy'))


def compare_result_dirs(dirs, out='comparison.png', pctl=99.98):
    """Load saved layer captures (layers*.npy) from several result dirs
    and render them side by side (reference plot_histograms.py:589-605
    compares four training runs this way)."""
    loaded, names = [], []
    for d in dirs:
        cands = sorted(
            f for f in os.listdir(d) if f.startswith('layers') and
            f.endswith('.npy')) if os.path.isdir(d) else []
        if not cands:
            print('no layers*.npy in', d)
            continue
        arr = np.load(os.path.join(d, cands[-1]), allow_pickle=True)
        loaded.append(arr)
        names.append(os.path.basename(os.path.normpath(d)))
    if not loaded:
        return None
    flat = [x for arr in loaded for x in list(arr)]
    labels = [f'{n}:{j}' for n, arr in zip(names, loaded)
              for j in range(len(list(arr)))]
    plot_grid(flat, labels, out, pctl=pctl)
    print('saved comparison to', out)
    return out


if __name__ == '__main__':
    import sys
    compare_result_dirs(sys.argv[1:] or ['.'])
