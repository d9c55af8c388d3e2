from .cross_entropy import LabelSmoothingCrossEntropy, SoftTargetCrossEntropy  # noqa: F401
