"""Capability-equivalent subset of the reference's vendored timm tree
(SURVEY.md §2.1 rows timm/*): model registry/factory, the EfficientNet
family on the shared builder, data pipeline (prefetch stream, mixup,
random erasing, distributed samplers), optimizers, schedulers, losses,
and training utilities."""

from .models import create_model, is_model, list_models  # noqa: F401
