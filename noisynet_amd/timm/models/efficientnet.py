"""Full EfficientNet family (B0-B8 + lite/edge variants) on the shared
builder (reference timm/models/efficientnet.py: full 7-stage arch_def at
:1049-1057, entrypoints :1072+, EfficientNetFeatures :940-1001)."""

import torch.nn as nn

from ...models.activations import HardSwish, Swish
from ...models.efficientnet_builder import (EfficientNet, decode_arch_def,
                                            round_channels)
from .registry import register_model

__all__ = []

default_cfgs = {}

_B0_ARCH_DEF = [
    ['ds_r1_k3_s1_e1_c16_se0.25'],
    ['ir_r2_k3_s2_e6_c24_se0.25'],
    ['ir_r2_k5_s2_e6_c40_se0.25'],
    ['ir_r3_k3_s2_e6_c80_se0.25'],
    ['ir_r3_k5_s1_e6_c112_se0.25'],
    ['ir_r4_k5_s2_e6_c192_se0.25'],
    ['ir_r1_k3_s1_e6_c320_se0.25'],
]


def _gen_efficientnet(channel_multiplier=1.0, depth_multiplier=1.0,
                      num_classes=1000, drop_rate=0.2, drop_connect_rate=0.2,
                      **kwargs):
    kwargs.pop('pretrained', None)
    kwargs.pop('in_chans', None)
    bn_out = kwargs.pop('bn_out', False)
    block_args = decode_arch_def(_B0_ARCH_DEF, depth_multiplier)
    model = EfficientNet(
        block_args, num_classes=num_classes, stem_size=32,
        channel_multiplier=channel_multiplier,
        num_features=round_channels(1280, channel_multiplier, 8, None),
        act_layer=Swish, drop_rate=drop_rate,
        drop_connect_rate=drop_connect_rate, bn_out=bn_out)
    return model


@register_model
def efficientnet_b0(pretrained=False, **kwargs):
    return _gen_efficientnet(1.0, 1.0, **kwargs)


@register_model
def efficientnet_b1(pretrained=False, **kwargs):
    return _gen_efficientnet(1.0, 1.1, **kwargs)


@register_model
def efficientnet_b2(pretrained=False, **kwargs):
    return _gen_efficientnet(1.1, 1.2, **kwargs)


@register_model
def efficientnet_b3(pretrained=False, **kwargs):
    return _gen_efficientnet(1.2, 1.4, **kwargs)


@register_model
def efficientnet_b4(pretrained=False, **kwargs):
    return _gen_efficientnet(1.4, 1.8, **kwargs)


@register_model
def efficientnet_b5(pretrained=False, **kwargs):
    return _gen_efficientnet(1.6, 2.2, **kwargs)


@register_model
def efficientnet_b6(pretrained=False, **kwargs):
    return _gen_efficientnet(1.8, 2.6, **kwargs)


@register_model
def efficientnet_b7(pretrained=False, **kwargs):
    return _gen_efficientnet(2.0, 3.1, **kwargs)


@register_model
def efficientnet_b8(pretrained=False, **kwargs):
    return _gen_efficientnet(2.2, 3.6, **kwargs)


@register_model
def efficientnet_es(pretrained=False, **kwargs):
    """EfficientNet-EdgeTPU small."""
    arch_def = [
        ['er_r1_k3_s1_e4_c24_noskip'],
        ['er_r2_k3_s2_e8_c32'],
        ['er_r4_k3_s2_e8_c48'],
        ['ir_r5_k5_s2_e8_c96'],
        ['ir_r4_k5_s1_e8_c144'],
        ['ir_r2_k5_s2_e8_c192'],
    ]
    kwargs.pop('pretrained', None)
    bn_out = kwargs.pop('bn_out', False)
    block_args = decode_arch_def(arch_def, 1.0)
    return EfficientNet(block_args, num_classes=kwargs.get('num_classes', 1000),
                        stem_size=32, channel_multiplier=1.0,
                        num_features=1280, act_layer=nn.ReLU,
                        drop_rate=0.2, bn_out=bn_out)


@register_model
def mobilenetv3_large_100(pretrained=False, **kwargs):
    """MobileNetV3-Large (timm-style generator, HardSwish activations)."""
    arch_def = [
        ['ds_r1_k3_s1_e1_c16_nre'],
        ['ir_r1_k3_s2_e4_c24_nre', 'ir_r1_k3_s1_e3_c24_nre'],
        ['ir_r3_k5_s2_e3_c40_se0.25_nre'],
        ['ir_r1_k3_s2_e6_c80', 'ir_r1_k3_s1_e2.5_c80', 'ir_r2_k3_s1_e2.3_c80'],
        ['ir_r2_k3_s1_e6_c112_se0.25'],
        ['ir_r3_k5_s2_e6_c160_se0.25'],
        ['cn_r1_k1_s1_c960'],
    ]
    kwargs.pop('pretrained', None)
    bn_out = kwargs.pop('bn_out', False)
    block_args = decode_arch_def(arch_def, 1.0)
    return EfficientNet(block_args, num_classes=kwargs.get('num_classes', 1000),
                        stem_size=16, channel_multiplier=1.0,
                        num_features=1280, act_layer=HardSwish,
                        drop_rate=0.2, bn_out=bn_out)


class EfficientNetFeatures(nn.Module):
    """Backbone feature extractor (reference timm/models/efficientnet.py:940-1001):
    returns the intermediate stage outputs instead of logits."""

    def __init__(self, out_indices=(0, 1, 2, 3, 4), channel_multiplier=1.0,
                 depth_multiplier=1.0, **kwargs):
        super().__init__()
        self.out_indices = out_indices
        model = _gen_efficientnet(channel_multiplier, depth_multiplier, **kwargs)
        self.conv_stem = model.conv_stem
        self.bn1 = model.bn1
        self.act1 = model.act1
        self.blocks = model.blocks

    def forward(self, x):
        x = self.act1(self.bn1(self.conv_stem(x)))
        features = []
        for i, stage in enumerate(self.blocks):
            x = stage(x)
            if i in self.out_indices:
                features.append(x)
        return features
