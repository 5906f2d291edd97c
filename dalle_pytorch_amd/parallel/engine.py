"""Pure-RCCL data-parallel engine over xGMI.

Replaces the reference's pluggable backend zoo (DeepSpeed / Horovod / Dummy,
distributed_backends/*) with one engine designed for an 8x MI355X node:

* one process per GPU, ``torch.distributed`` with the ``nccl`` backend
  (which IS RCCL on ROCm), rendezvous from the standard launcher env;
* parameter/buffer broadcast from rank 0 at wrap time (the Horovod path's
  C5 collective, horovod_backend.py:49-52);
* gradient all-reduce in large flat buckets (default 64 MiB — xGMI links
  are point-to-point at ~153 GB/s, so fewer, larger collectives win),
  launched asynchronously as soon as each bucket's grads are final, so
  communication overlaps the remaining backward compute (the DeepSpeed
  engine's C3 collective);
* native gradient accumulation (``no_sync``) and scalar loss averaging
  (C1, deepspeed_backend.py:165-171).

Single-process runs need no initialization at all — every helper degrades
to a no-op, which is the reference DummyBackend's contract
(dummy_backend.py:4-52).
"""

import os
from contextlib import contextmanager

import torch
import torch.distributed as dist

_DEFAULT_BUCKET_BYTES = 64 << 20


def is_distributed():
    return dist.is_available() and dist.is_initialized()


def init_distributed(backend=None, timeout_minutes=30):
    """Initialize from torchrun/launcher env. Returns (rank, world, local)."""
    if is_distributed():
        return get_rank(), get_world_size(), get_local_rank()
    world = int(os.environ.get('WORLD_SIZE', '1'))
    if world <= 1 and 'RANK' not in os.environ:
        return 0, 1, 0
    if backend is None:
        backend = 'nccl' if torch.cuda.is_available() else 'gloo'
    import datetime
    dist.init_process_group(backend=backend,
                            timeout=datetime.timedelta(minutes=timeout_minutes))
    local = get_local_rank()
    if torch.cuda.is_available():
        torch.cuda.set_device(local)
    return get_rank(), get_world_size(), local


def get_rank():
    return dist.get_rank() if is_distributed() else 0


def get_world_size():
    return dist.get_world_size() if is_distributed() else 1


def get_local_rank():
    return int(os.environ.get('LOCAL_RANK', '0'))


def barrier():
    if is_distributed():
        dist.barrier()


def average_scalar(value):
    """All-reduce-average a scalar tensor across ranks (collective C1)."""
    if not is_distributed():
        return value
    t = value.detach().clone() if torch.is_tensor(value) else torch.tensor(value)
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t / get_world_size()


class _Bucket:
    __slots__ = ('flat', 'params', 'pending', 'handle', 'offsets', 'wire')

    def __init__(self, params, device, dtype):
        self.params = params
        numel = sum(p.numel() for p in params)
        self.flat = torch.zeros(numel, device=device, dtype=dtype)
        self.offsets = []
        off = 0
        for p in params:
            self.offsets.append(off)
            off += p.numel()
        self.pending = len(params)
        self.handle = None
        self.wire = None    # compressed send buffer when wire_dtype is set


class DataParallelEngine:
    """Wraps an on-device model for synchronous data parallelism.

    Usage::

        engine = DataParallelEngine(model)          # broadcast + hook grads
        loss = model(...); loss.backward()
        engine.finish_gradient_sync()               # wait bucketed allreduce
        clip/step/zero via engine.zero_grad()
    """

    def __init__(self, model, bucket_bytes=_DEFAULT_BUCKET_BYTES,
                 grad_average=True, broadcast=True, wire_dtype='bf16'):
        """wire_dtype: dtype gradients travel in over xGMI.
        'bf16' (default) halves the 1.19 GiB/step fp32 payload of the
        flagship config: grads accumulate locally in fp32, are pre-divided
        by world size (a power of two — an exact exponent shift), cast to
        bf16 for the all-reduce, and written back to the fp32 master grads.
        None / 'fp32' reduces in the accumulation dtype. fp32 buckets only;
        bf16-param models already travel at wire width."""
        self.model = model
        self.world_size = get_world_size()
        self.rank = get_rank()
        self.grad_average = grad_average
        self._sync_enabled = True
        self._buckets = []
        self._param_bucket = {}
        if wire_dtype in (None, 'fp32', torch.float32):
            self.wire_dtype = None
        else:
            self.wire_dtype = (torch.bfloat16 if wire_dtype == 'bf16'
                               else wire_dtype)

        if self.world_size > 1 and broadcast:
            self.broadcast_parameters()

        params = [p for p in model.parameters() if p.requires_grad]
        if not params:
            return
        # bucket in reverse creation order: last layers' grads are ready
        # first in backward, so their all-reduce launches earliest
        groups, cur, cur_bytes = [], [], 0
        for p in reversed(params):
            nbytes = p.numel() * p.element_size()
            if cur and (cur_bytes + nbytes > bucket_bytes
                        or p.dtype != cur[0].dtype or p.device != cur[0].device):
                groups.append(cur)
                cur, cur_bytes = [], 0
            cur.append(p)
            cur_bytes += nbytes
        if cur:
            groups.append(cur)

        for g in groups:
            b = _Bucket(g, g[0].device, g[0].dtype)
            self._buckets.append(b)
            for p, off in zip(g, b.offsets):
                # grads accumulate directly into the flat bucket storage:
                # no gather/copy before the collective
                p.grad = b.flat.narrow(0, off, p.numel()).view_as(p)
                self._param_bucket[p] = b
                p.register_post_accumulate_grad_hook(self._grad_ready)

    # ------------------------------------------------------------------

    def broadcast_parameters(self, root=0):
        if not is_distributed():
            return
        with torch.no_grad():
            for t in list(self.model.parameters()) + list(self.model.buffers()):
                if t.is_floating_point() or t.dtype in (torch.int64, torch.int32,
                                                        torch.bool, torch.uint8):
                    dist.broadcast(t.data, src=root)

    @contextmanager
    def no_sync(self):
        """Skip gradient all-reduce (gradient-accumulation microbatches)."""
        prev = self._sync_enabled
        self._sync_enabled = False
        try:
            yield
        finally:
            self._sync_enabled = prev

    def _use_wire(self, b):
        return self.wire_dtype is not None and b.flat.dtype == torch.float32

    def _launch_reduce(self, b):
        if self._use_wire(b):
            if self.grad_average:
                # exact: world size is a power of two on this topology
                b.flat.div_(self.world_size)
            b.wire = b.flat.to(self.wire_dtype)
            b.handle = dist.all_reduce(b.wire, op=dist.ReduceOp.SUM,
                                       async_op=True)
        else:
            b.handle = dist.all_reduce(b.flat, op=dist.ReduceOp.SUM,
                                       async_op=True)

    def _grad_ready(self, param):
        b = self._param_bucket[param]
        b.pending -= 1
        if b.pending == 0:
            b.pending = len(b.params)
            if self._sync_enabled and self.world_size > 1:
                self._launch_reduce(b)

    def finish_gradient_sync(self):
        """Wait for in-flight bucket collectives; average. Call between
        ``loss.backward()`` and the optimizer step."""
        if self.world_size <= 1 or not self._sync_enabled:
            return
        for b in self._buckets:
            if b.handle is None:
                # a bucket whose hook never completed (unused params):
                # reduce it now — zeros contribute nothing
                self._launch_reduce(b)
        for b in self._buckets:
            b.handle.wait()
            b.handle = None
            if b.wire is not None:
                b.flat.copy_(b.wire)    # averaged already (pre-divided)
                b.wire = None
            elif self.grad_average:
                b.flat.div_(self.world_size)

    def clip_grad_norm_(self, max_norm, eps=1e-6):
        """Global grad-norm clip computed on the flat buckets: one norm and
        one scale kernel per 64 MiB bucket instead of the per-parameter
        reduce kernels ``torch.nn.utils.clip_grad_norm_`` launches (~60 per
        step at the flagship param count). Semantics match torch's default
        (L2 norm over all grads, coef = max_norm/(total+eps) clamped to 1).
        """
        if not self._buckets:
            return torch.nn.utils.clip_grad_norm_(
                self.model.parameters(), max_norm)
        norms = torch.stack(
            [torch.linalg.vector_norm(b.flat, 2) for b in self._buckets])
        total = torch.linalg.vector_norm(norms, 2)
        coef = (max_norm / (total + eps)).clamp(max=1.0)
        for b in self._buckets:
            b.flat.mul_(coef)
        return total

    def zero_grad(self):
        for b in self._buckets:
            b.flat.zero_()
            b.pending = len(b.params)
        # params outside buckets (none normally) fall back to .grad = None
        for p in self.model.parameters():
            if p.requires_grad and p not in self._param_bucket and p.grad is not None:
                p.grad = None

    @property
    def is_root(self):
        return self.rank == 0

    def average_all(self, value):
        return average_scalar(value)
