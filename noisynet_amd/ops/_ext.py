"""Loader for the in-tree HIP extension (``noisynet_hip``).

The extension is built ahead of time into ``noisynet_amd/ops/`` by
``__graft_entry__.build()`` (hipcc, --offload-arch=gfx950) so the ``.so``
travels with the repo snapshot to GPU machines. There is deliberately NO
silent fallback on GPU: if a CUDA/HIP tensor reaches an op and the extension
is missing, we raise. CPU tensors always use the pure-PyTorch reference path
(``noisynet_amd.ops.reference``), which doubles as the numerics oracle.
"""

import os

import torch

_EXT = None
_EXT_ERR = None


def _try_load():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return
    try:
        import importlib

        _EXT = importlib.import_module("noisynet_amd.ops.noisynet_hip")
    except Exception as exc:  # pragma: no cover - exercised only sans .so
        _EXT_ERR = exc


def has_ext():
    _try_load()
    return _EXT is not None


def ext():
    """Return the extension module, raising with a clear message if absent."""
    _try_load()
    if _EXT is None:
        raise RuntimeError(
            "noisynet_hip HIP extension is not built/loadable but a GPU tensor "
            "reached a native op. Build it with `python __graft_entry__.py build` "
            "(hipcc --offload-arch=gfx950). Original import error: %r" % (_EXT_ERR,)
        )
    return _EXT


def use_native(*tensors):
    """HIP kernels run for CUDA(ROCm) tensors; CPU tensors use the reference path.

    On a GPU box the native path is mandatory: if any input is on a HIP device
    and the extension is missing, ``ext()`` (called by the op) raises rather
    than silently falling back to eager PyTorch.
    """
    if os.environ.get("NOISYNET_FORCE_REFERENCE", "0") == "1":
        return False
    return any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))
