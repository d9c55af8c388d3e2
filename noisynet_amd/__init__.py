"""NoisyNet-MI355X: an AMD MI355X-native framework for noise-aware quantized CNN training.

A from-scratch rebuild of the capabilities of michaelklachko/NoisyNet
(reference layout documented in SURVEY.md): analog current-noise injection
(Gaussian, scaled by I_max), 2-8 bit clamp-quantize training with stochastic
rounding and percentile calibration, and a robustness-evaluation harness --
implemented MI355X-first:

* hot-path ops are hand-written CDNA4 (gfx950) HIP kernels: MFMA implicit-GEMM
  conv/GEMM with fused weight-quantize + dual-accumulator sigma computation +
  in-kernel Philox Gaussian noise, fused BN+activation(+clip) epilogues,
  fused SGD/AdamW updates with weight clamping (see ``csrc/``);
* data-parallel scaling is one process per GPU with RCCL over xGMI:
  bucketed all-reduce overlapped with backward on a dedicated HIP stream
  (``noisynet_amd.distributed``);
* CLI flags, entrypoints (noisynet.py / main.py / train_efficientnet.py /
  chip_mnist.py) and checkpoint formats stay compatible with the reference.

On a machine without a GPU every op falls back to a pure-PyTorch reference
implementation (``noisynet_amd.ops.reference``) which is also the numerics
oracle for the HIP kernels' unit tests. On a GPU the HIP extension is
required: ops raise if it cannot be loaded (no silent eager fallback).
"""

__version__ = "0.1.0"

from . import ops  # noqa: F401
