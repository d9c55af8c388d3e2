from .noisynet import Net, noisynet  # noqa: F401

__all__ = ['Net', 'noisynet']
