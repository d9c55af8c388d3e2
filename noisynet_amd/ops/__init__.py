from . import reference  # noqa: F401
from ._ext import ext, has_ext, use_native  # noqa: F401
from .functional import (  # noqa: F401
    adamw_step,
    add_weight_noise,
    bn_act,
    conv2d,
    cross_entropy,
    dropout,
    fake_quant,
    fused_noisy_conv2d,
    fused_noisy_linear,
    avgpool_nhwc,
    kth_percentile,
    linear,
    maxpool2x2,
    maxpool_nhwc,
    NoiseTelemetry,
    relu_clip,
    sgd_step,
    sigma_noise_conv2d,
    sigma_noise_linear,
    simple_noise,
)
