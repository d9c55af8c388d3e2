from .cross_entropy import (FusedCrossEntropy, LabelSmoothingCrossEntropy,  # noqa: F401
                            SoftTargetCrossEntropy)
