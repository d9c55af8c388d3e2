"""hipGraph capture of the training step.

The secondary models (ResNet-18 / MobileNetV2 / EfficientNet-B0) launch
~900 small kernels per step and were measured host-launch-bound at ~40%
GPU busy (profiles/noisynet_bench_kernel_breakdown.md). Capturing one full
training step (forward + loss + backward + optimizer) in a hipGraph turns
~900 Python-driven launches into ONE graph launch per step.

RNG correctness across replays: kernel arguments are frozen at capture, so
by-value Philox seeds would replay identical noise. The extension instead
reads a 1-element int64 "step counter" device buffer (csrc/common.h
graph_seed): ``GraphedTrainStep`` installs the buffer, captures a
``counter.add_(1)`` node at the top of the graph, and every RNG kernel
mixes the live counter with its per-launch salt -- each replay draws fresh
noise, deterministic under --seed.

Constraints on the captured step (standard hipGraph rules):
  * all tensors the step reads must be the SAME storage every replay --
    copy fresh batches into the static input buffers, then ``replay()``;
  * no host synchronisation inside the step (QuantMeasure caches its
    calibrated range as a Python float after ``finish_calibration`` so the
    hot path never reads a device scalar);
  * collectives are not captured -- graph mode is single-process (the
    multi-GPU path keeps eager launches with overlapped RCCL all-reduce).
"""

import torch

from . import ops


class GraphedTrainStep:
    """Capture ``step_fn`` (a closure over static tensors that runs one full
    training step and returns the loss tensor) into a hipGraph.

    Usage:
        gstep = GraphedTrainStep(step_fn)         # warms up + captures
        ...
        static_x.copy_(next_batch)                # refresh static inputs
        loss = gstep.replay()                     # one graph launch
    """

    def __init__(self, step_fn, warmup=3):
        if not torch.cuda.is_available():
            raise RuntimeError("GraphedTrainStep requires a GPU")
        self.step_fn = step_fn
        # live step counter read by RNG kernels (kept alive by this object)
        self.seed_buf = torch.zeros(1, dtype=torch.int64, device="cuda")
        ops.ext().set_seed_buffer(self.seed_buf)
        self._owns_seed_buffer = True

        # warm up on a side stream (allocator settles, cuDNN-style lazy init)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup):
                step_fn()
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.seed_buf.add_(1)  # fresh RNG stream every replay
            self.loss = step_fn()

    def replay(self):
        self.graph.replay()
        return self.loss

    def close(self):
        """Detach the RNG step counter from the extension (call before
        dropping the object if eager RNG kernels will still run)."""
        if getattr(self, "_owns_seed_buffer", False):
            try:
                ops.ext().clear_seed_buffer()
            except Exception:
                pass
            self._owns_seed_buffer = False

    def __del__(self):
        self.close()
