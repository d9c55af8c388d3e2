"""Mixup: convex combination of a batch with its reversed self, plus
label-smoothed soft targets.

Capability parity with the reference's timm mixup surface
(timm/data/mixup.py:5-43): ``mixup_target`` builds the smoothed two-hot
target, ``FastCollateMixup`` applies mixup inside collate on the raw uint8
numpy batch (so the GPU prefetch stream sees already-mixed frames), and
``mixup_batch`` is the tensor-level variant used outside the prefetcher.

Implementation is vectorized tensor code throughout -- no per-sample
Python loop in the collate path.
"""

import numpy as np
import torch


def _smoothed_one_hot(labels, num_classes, smoothing, device):
    """(B,) int labels -> (B, C) float targets with label smoothing."""
    off = smoothing / num_classes
    out = torch.full((labels.numel(), num_classes), off,
                     dtype=torch.float32, device=device)
    out.scatter_(1, labels.long().view(-1, 1).to(device),
                 1.0 - smoothing + off)
    return out


def one_hot(x, num_classes, on_value=1., off_value=0., device='cuda'):
    out = torch.full((x.numel(), num_classes), off_value,
                     dtype=torch.float32, device=device)
    out.scatter_(1, x.long().view(-1, 1).to(device), on_value)
    return out


def mixup_target(target, num_classes, lam=1., smoothing=0.0, device='cuda'):
    """Soft target for a batch mixed with its flip: lam*y + (1-lam)*y_flip."""
    y = _smoothed_one_hot(target, num_classes, smoothing, device)
    # lerp(y_flip, y, lam) == lam*y + (1-lam)*y_flip
    return torch.lerp(y.flip(0), y, float(lam))


def mixup_batch(input, target, alpha=0.2, num_classes=1000, smoothing=0.1,
                disable=False):
    """In-place mixup of a device-resident batch with its reversed self."""
    lam = 1.0 if disable else float(np.random.beta(alpha, alpha))
    input = torch.lerp(input.flip(0), input, lam)
    target = mixup_target(target, num_classes, lam, smoothing,
                          device=str(input.device))
    return input, target


class FastCollateMixup:
    """Collate a list of (uint8 HWC/CHW numpy array, label) samples into a
    mixed uint8 batch + soft-target tensor.

    Mixing happens on the stacked numpy batch in one vectorized pass (the
    reference loops per sample); rounding to nearest keeps the uint8
    contract so the prefetch stream's normalize kernel is unchanged.
    """

    def __init__(self, mixup_alpha=1., label_smoothing=0.1, num_classes=1000):
        self.mixup_alpha = mixup_alpha
        self.label_smoothing = label_smoothing
        self.num_classes = num_classes
        self.mixup_enabled = True

    def _draw_lam(self):
        if not self.mixup_enabled:
            return 1.0
        return float(np.random.beta(self.mixup_alpha, self.mixup_alpha))

    def __call__(self, batch):
        lam = self._draw_lam()
        labels = torch.tensor([label for _, label in batch],
                              dtype=torch.int64)
        target = mixup_target(labels, self.num_classes, lam,
                              self.label_smoothing, device='cpu')
        frames = np.stack([np.asarray(img) for img, _ in batch])
        mixed = frames.astype(np.float32)
        mixed *= lam
        mixed += frames[::-1].astype(np.float32) * (1.0 - lam)
        np.rint(mixed, out=mixed)
        return torch.from_numpy(mixed.astype(np.uint8)), target
