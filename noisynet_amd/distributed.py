"""Data-parallel training over RCCL/xGMI: one process per GPU.

MI355X-native replacement for the reference's torch/apex DDP wiring
(main.py:737-803): gradient synchronisation is implemented here directly --
parameters are packed into flat buckets and all-reduced with async RCCL
collectives that overlap with the remaining backward pass (the xGMI
full-mesh gives each GPU 7 point-to-point links; RCCL builds multi-ring
schedules over them, so a handful of large buckets beats many small ones).

``delay_allreduce=True`` reproduces apex DDP's non-overlapped single-shot
mode (main.py:798) for parity testing.

Everything works with the gloo backend on CPU (world_size>1 multi-process
CPU tests run in CI); on ROCm the "nccl" backend IS RCCL.
"""

import os
from datetime import timedelta

import torch
import torch.distributed as dist


def env_rank():
    return int(os.environ.get("RANK", os.environ.get("LOCAL_RANK", 0)))


def env_world_size():
    return int(os.environ.get("WORLD_SIZE", 1))


def env_local_rank():
    return int(os.environ.get("LOCAL_RANK", 0))


def init_distributed(backend=None, timeout_s=300):
    """env:// rendezvous with the torchrun env contract (WORLD_SIZE, RANK,
    LOCAL_RANK, MASTER_ADDR/PORT), matching main.py:737-755."""
    if env_world_size() <= 1:
        return False
    if dist.is_initialized():
        return True
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if torch.cuda.is_available():
        torch.cuda.set_device(env_local_rank())
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    dist.init_process_group(backend=backend, init_method="env://",
                            timeout=timedelta(seconds=timeout_s))
    return True


def reduce_tensor(tensor, world_size=None):
    """all-reduce mean of a metric tensor (timm/utils.py:202-206)."""
    if not dist.is_initialized():
        return tensor
    ws = world_size or dist.get_world_size()
    rt = tensor.clone()
    dist.all_reduce(rt, op=dist.ReduceOp.SUM)
    rt /= ws
    return rt


class GradBucket:
    __slots__ = ("params", "numel", "bytes", "dtype", "flat", "work", "ready")

    def __init__(self, dtype):
        self.params = []
        self.numel = 0
        self.bytes = 0
        self.dtype = dtype
        self.flat = None      # persistent flat buffer, allocated on first use
        self.work = None
        self.ready = 0


class DataParallel:
    """Bucketed, overlapped gradient all-reduce.

    Buckets are built in REVERSE parameter order (gradients become ready
    back-to-front during backward) and are homogeneous in dtype, each up to
    ``bucket_cap_mb`` of ACTUAL grad bytes (bf16 params get bf16-sized
    buckets). When the last gradient of a bucket lands
    (post-accumulate-grad hook), the grads are packed into the bucket's
    persistent flat buffer and an async all-reduce is launched immediately
    -- RCCL runs it on its own stream, overlapping the remaining backward
    compute. ``finish()`` waits for all works and scatters the averaged
    gradients back.

    A parameter whose grad is absent at reduce time (e.g. manually-updated
    clip params like ``w_max1`` under ``--train_w_max``, drivers/cifar.py)
    contributes ZEROS to its bucket segment, so the collective shape is
    identical on every rank regardless of which params got autograd grads;
    the averaged values are scattered back only into grads that exist.

    Gradient accumulation: wrap non-final backwards in ``no_sync()`` so the
    hooks skip launching; the final (unwrapped) backward reduces the fully
    accumulated grads. One ``finish()`` per synchronized backward.

    With ``delay_allreduce=True`` hooks are skipped and ``finish()`` does
    one bucketed all-reduce sweep after backward (apex parity mode,
    reference main.py:798).
    """

    def __init__(self, model, bucket_cap_mb=25, delay_allreduce=False,
                 process_group=None):
        self.model = model
        self.group = process_group
        self.delay = delay_allreduce
        self.enabled = dist.is_initialized() and dist.get_world_size() > 1
        self.world_size = dist.get_world_size() if self.enabled else 1
        self._hooks = []
        self._sync = True

        params = [p for p in model.parameters() if p.requires_grad]
        cap = int(bucket_cap_mb * 1024 * 1024)
        self.buckets = []
        self.param_bucket = {}
        bucket = None
        for p in reversed(params):
            bytes_p = p.numel() * p.element_size()
            if bucket is None or bucket.dtype != p.dtype or \
                    (bucket.params and bucket.bytes + bytes_p > cap):
                if bucket is not None and bucket.params:
                    self.buckets.append(bucket)
                bucket = GradBucket(p.dtype)
            bucket.params.append(p)
            bucket.numel += p.numel()
            bucket.bytes += bytes_p
            self.param_bucket[p] = bucket
        if bucket is not None and bucket.params:
            self.buckets.append(bucket)

        if self.enabled:
            self.sync_parameters()
            if not self.delay:
                for p in params:
                    h = p.register_post_accumulate_grad_hook(self._hook)
                    self._hooks.append(h)

    # --------------------------------------------------------------
    def sync_parameters(self):
        """Broadcast rank-0 parameters and buffers (DDP init semantics)."""
        if not self.enabled:
            return
        for t in list(self.model.parameters()) + list(self.model.buffers()):
            if t.dtype in (torch.float32, torch.float16, torch.bfloat16,
                           torch.float64, torch.int64, torch.int32):
                dist.broadcast(t.data, src=0, group=self.group)

    def no_sync(self):
        """Context manager: skip gradient sync (accumulation steps)."""
        import contextlib

        @contextlib.contextmanager
        def ctx():
            self._sync = False
            try:
                yield
            finally:
                self._sync = True
        return ctx()

    def _launch(self, bucket):
        # pack grads into the persistent flat buffer (no per-step torch.cat
        # allocation); absent grads contribute zeros so every rank reduces
        # the same shape.
        dev = bucket.params[0].device
        for p in bucket.params:
            if p.grad is not None:
                dev = p.grad.device
                break
        if bucket.flat is None or bucket.flat.device != dev:
            bucket.flat = torch.empty(bucket.numel, dtype=bucket.dtype,
                                      device=dev)
        flat = bucket.flat
        off = 0
        inv = 1.0 / self.world_size
        for p in bucket.params:
            n = p.numel()
            seg = flat[off:off + n]
            if p.grad is not None:
                seg.copy_(p.grad.reshape(-1))
            else:
                seg.zero_()
            off += n
        flat.mul_(inv)
        bucket.work = dist.all_reduce(flat, op=dist.ReduceOp.SUM,
                                      group=self.group, async_op=True)

    def _hook(self, p):
        if not self._sync:
            return
        bucket = self.param_bucket[p]
        bucket.ready += 1
        if bucket.ready == len(bucket.params):
            self._launch(bucket)

    def finish(self):
        """Wait for (or run) all bucket reductions; scatter averages back."""
        if not self.enabled:
            return
        for bucket in self.buckets:
            if bucket.work is None:
                # delay mode, or a bucket whose last hook never fired
                # (params with manually-written grads / unused params)
                self._launch(bucket)
            bucket.work.wait()
            off = 0
            for p in bucket.params:
                n = p.numel()
                if p.grad is not None:
                    p.grad.copy_(bucket.flat[off:off + n].view_as(p.grad))
                off += n
            bucket.work = None
            bucket.ready = 0

    def remove(self):
        for h in self._hooks:
            h.remove()
        self._hooks = []
