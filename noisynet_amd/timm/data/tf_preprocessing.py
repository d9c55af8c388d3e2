"""TF-exact preprocessing stub.

The reference's timm/data/tf_preprocessing.py runs Tensorflow's exact
EfficientNet eval preprocessing through a tf.Session (reference
:199-227). Tensorflow is not installed in this environment; the entry
point is kept for API parity and raises with a clear message.
"""


class TfPreprocessTransform:
    def __init__(self, is_training=False, size=224, interpolation='bicubic'):
        raise ImportError(
            'TF-exact preprocessing requires tensorflow, which is not '
            'available in this environment. Use the numpy/torch transforms '
            'in noisynet_amd.timm.data.transforms instead.')
