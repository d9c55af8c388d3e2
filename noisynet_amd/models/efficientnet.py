"""Local EfficientNet-B0 entrypoint (reference models/efficientnet.py:732-737).

The reference deliberately truncates the arch_def to ONE stage
([['ds_r1_k3_s1_e1_c16_se0.25']], models/efficientnet.py:717) -- a mini
model for chip experiments -- and adds the trailing bn_out BatchNorm1d.
The full B0-B8 family lives in noisynet_amd.timm.models.efficientnet.
"""

from .efficientnet_builder import EfficientNet, decode_arch_def
from .activations import Swish


def efficientnet_b0(parameters):
    args = parameters
    arch_def = [['ds_r1_k3_s1_e1_c16_se0.25']]  # truncated, as the reference
    block_args = decode_arch_def(arch_def, depth_multiplier=1.0)
    model = EfficientNet(block_args, num_classes=1000, stem_size=32,
                         channel_multiplier=1.0, num_features=1280,
                         act_layer=Swish, drop_rate=0.2,
                         bn_out=getattr(args, 'bn_out', False))
    model.args = args
    return model
