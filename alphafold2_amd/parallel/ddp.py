"""Data-parallel training over RCCL / xGMI.

The reference has an empty distributed-training slot
(reference training_scripts/deepspeed.py + lightning.py are 0-byte
stubs; zero collective call sites in the tree — SURVEY.md §2.16).  This
module is the MI355X-native replacement: one process per GPU,
`torch.distributed` with the nccl backend (RCCL on ROCm), and a
bucketed gradient all-reduce overlapped with backward.

xGMI-aware choices:
* the MI355X node fabric is point-to-point (7 links x ~153 GB/s per
  GPU), so a ring all-reduce is per-link bound: buckets default to
  64 MiB — large enough to amortize RCCL launch + ring pipelining on a
  per-link-bound fabric, small enough that several buckets are in
  flight before backward finishes (overlap).
* gradients are flattened per-bucket into one contiguous buffer so each
  collective moves one large message instead of many small ones.
* collectives are issued from gradient-ready hooks
  (post-accumulate-grad), i.e. in reverse parameter order = backward
  order, so communication of early buckets overlaps the rest of
  backward on the compute stream.

Works identically over gloo on CPU (multi-process CPU tests).
"""
import os
from typing import List, Optional

import torch
import torch.distributed as dist


def _force_dist():
    """AF2AMD_FORCE_DIST=1: initialize a process group and run the full
    DDP machinery even at world_size 1.  This is the single-GPU
    rehearsal mode for the multi-GPU path — it exercises RCCL init,
    broadcast, bucketed async all-reduce and finalize ordering (incl.
    under hipGraph capture) without needing a multi-GPU lease."""
    return os.environ.get('AF2AMD_FORCE_DIST') == '1'


def init_distributed(backend: Optional[str] = None):
    """Initialize torch.distributed from torchrun env vars; no-op when
    single-process.  Returns (rank, world_size, local_rank)."""
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size(), \
            int(os.environ.get('LOCAL_RANK', 0))
    world_size = int(os.environ.get('WORLD_SIZE', '1'))
    if world_size <= 1 and not _force_dist():
        return 0, 1, 0
    if backend is None:
        backend = 'nccl' if torch.cuda.is_available() else 'gloo'
    local_rank = int(os.environ.get('LOCAL_RANK', '0'))
    if backend == 'nccl':
        torch.cuda.set_device(local_rank)
    os.environ.setdefault('MASTER_ADDR', '127.0.0.1')
    os.environ.setdefault('MASTER_PORT', '29511')
    dist.init_process_group(backend=backend, world_size=world_size,
                            rank=int(os.environ.get('RANK', '0')))
    return dist.get_rank(), dist.get_world_size(), local_rank


def is_distributed():
    return dist.is_available() and dist.is_initialized() \
        and dist.get_world_size() > 1


def get_rank():
    return dist.get_rank() if is_distributed() else 0


def get_world_size():
    return dist.get_world_size() if is_distributed() else 1


def all_reduce_mean(t: torch.Tensor):
    """Mean-all-reduce a metric tensor across ranks (validation)."""
    if not is_distributed():
        return t
    t = t.clone()
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t / get_world_size()


class _Bucket:
    __slots__ = ('params', 'flat', 'numel', 'ready', 'work', 'offsets',
                 'seen', 'pending', 'skipped')

    def __init__(self):
        self.params: List[torch.nn.Parameter] = []
        self.flat: Optional[torch.Tensor] = None
        self.numel = 0
        self.ready = 0
        self.work = None
        self.offsets = {}
        self.seen = set()
        # all params got grads this step; the bucket is eligible to
        # launch (launches happen strictly in bucket-index order)
        self.pending = False
        # skip_unused_buckets marked this bucket as reduce-less this step
        self.skipped = False


class DataParallelEngine:
    """Bucketed-overlap data parallelism.

    Usage:
        engine = DataParallelEngine(model)
        ...
        loss.backward()
        engine.finalize()          # wait for in-flight all-reduces
        optimizer.step()
        engine.zero_grad()
    """

    def __init__(self, model: torch.nn.Module, bucket_cap_mb: float = 64,
                 process_group=None, grad_dtype=None,
                 skip_unused_buckets: bool = False):
        self.model = model
        self.group = process_group
        self.world_size = get_world_size()
        # forced-dist rehearsal: run the full collective machinery even
        # at world_size 1 (see _force_dist)
        self.enabled = is_distributed() or \
            (dist.is_available() and dist.is_initialized() and _force_dist())
        self.grad_dtype = grad_dtype
        # skip_unused_buckets=True suppresses the all-reduce of buckets
        # in which NO param has a grad.  Only safe when every rank runs
        # the SAME graph every step (optional inputs like templates /
        # extra-MSA present-or-absent uniformly across ranks): a rank
        # skipping a bucket that another rank reduces deadlocks the job.
        # Default False = always reduce every bucket (collective-
        # consistent by construction, like torch DDP).
        self.skip_unused_buckets = skip_unused_buckets
        self._hooks = []
        self._buckets: List[_Bucket] = []
        self._param_bucket = {}
        self._sync = True
        # index of the next bucket allowed to launch its all-reduce.
        # Buckets launch strictly in index order regardless of the order
        # their hooks complete, so ranks whose backward visits modules
        # in different orders still issue identical collective sequences.
        self._next_launch = 0

        if self.enabled:
            self._broadcast_parameters()
            self._build_buckets(int(bucket_cap_mb * 1024 * 1024))
            self._register_hooks()

    # -- setup ------------------------------------------------------------

    def _broadcast_parameters(self):
        for p in self.model.parameters():
            dist.broadcast(p.data, src=0, group=self.group)
        for b in self.model.buffers():
            if b.dtype.is_floating_point or b.dtype in (torch.int64, torch.int32):
                dist.broadcast(b.data, src=0, group=self.group)

    def _build_buckets(self, cap_bytes: int):
        # reverse order approximates gradient-ready (backward) order
        params = [p for p in self.model.parameters() if p.requires_grad]
        current = _Bucket()
        for p in reversed(params):
            bytes_ = p.numel() * p.element_size()
            if current.numel > 0 and \
                    (current.numel * p.element_size() + bytes_) > cap_bytes:
                self._buckets.append(current)
                current = _Bucket()
            current.offsets[p] = current.numel
            current.params.append(p)
            current.numel += p.numel()
            self._param_bucket[p] = current
        if current.numel > 0:
            self._buckets.append(current)

    def _register_hooks(self):
        for p in self._param_bucket:
            h = p.register_post_accumulate_grad_hook(self._on_grad_ready)
            self._hooks.append(h)

    # -- steady state -----------------------------------------------------

    def _ensure_flat(self, bucket: _Bucket, like: torch.Tensor):
        if bucket.flat is None or bucket.flat.device != like.device:
            dtype = self.grad_dtype or like.dtype
            bucket.flat = torch.zeros(bucket.numel, device=like.device,
                                      dtype=dtype)

    def no_sync(self):
        """Context manager suppressing gradient reduction (use for all
        but the last micro-batch under gradient accumulation)."""
        import contextlib

        @contextlib.contextmanager
        def ctx():
            self._sync = False
            try:
                yield
            finally:
                self._sync = True
        return ctx()

    def _on_grad_ready(self, p: torch.nn.Parameter):
        if not self._sync:
            return
        bucket = self._param_bucket[p]
        self._ensure_flat(bucket, p.grad)
        off = bucket.offsets[p]
        bucket.flat[off:off + p.numel()].copy_(p.grad.reshape(-1))
        bucket.seen.add(p)
        bucket.ready += 1
        if bucket.ready == len(bucket.params):
            bucket.pending = True
            self._launch_in_order()

    def _launch_in_order(self):
        """Launch pending bucket all-reduces strictly in bucket-index
        order.  One large message per bucket; async so backward keeps
        going while early buckets reduce."""
        while self._next_launch < len(self._buckets) \
                and self._buckets[self._next_launch].pending:
            bucket = self._buckets[self._next_launch]
            if not bucket.skipped:
                bucket.work = dist.all_reduce(
                    bucket.flat, op=dist.ReduceOp.SUM,
                    group=self.group, async_op=True)
            self._next_launch += 1

    def finalize(self):
        """Flush buckets the hooks didn't complete, wait for in-flight
        collectives, average, write grads back."""
        if not self.enabled:
            return
        inv = 1.0 / self.world_size
        # fill + mark every not-yet-pending bucket, then launch the
        # remainder in index order (identical sequence on every rank)
        for bucket in self._buckets:
            if bucket.pending:
                continue
            if self.skip_unused_buckets \
                    and not any(p.grad is not None for p in bucket.params):
                bucket.skipped = True
            else:
                self._fill_flat(bucket)
            bucket.pending = True
        self._launch_in_order()
        for bucket in self._buckets:
            if bucket.work is not None:
                bucket.work.wait()
                bucket.work = None
                bucket.flat.mul_(inv)
                for p in bucket.params:
                    if p.grad is None:
                        continue
                    off = bucket.offsets[p]
                    p.grad.copy_(
                        bucket.flat[off:off + p.numel()].view_as(p.grad))
            bucket.ready = 0
            bucket.pending = False
            bucket.skipped = False
            bucket.seen.clear()
        self._next_launch = 0

    def _fill_flat(self, bucket: _Bucket):
        """Fill the flat buffer of a bucket some of whose params never
        fired their hook this step.  A param can hold a REAL accumulated
        grad and still not fire (it got its grad during a no_sync
        micro-batch and was unused in the final one), so copy every
        un-seen grad into the flat buffer — and zero the slice of truly
        grad-less params so a previous step's values never leak into the
        collective."""
        if bucket.flat is None:
            any_grad = next(
                (p.grad for p in bucket.params if p.grad is not None), None)
            like = any_grad if any_grad is not None \
                else next(self.model.parameters())
            self._ensure_flat(bucket, like)
        for p in bucket.params:
            if p in bucket.seen:
                continue
            off = bucket.offsets[p]
            dst = bucket.flat[off:off + p.numel()]
            if p.grad is not None:
                dst.copy_(p.grad.reshape(-1))
            else:
                dst.zero_()
        bucket.ready = len(bucket.params)

    def zero_grad(self, set_to_none: bool = True):
        self.model.zero_grad(set_to_none=set_to_none)
        for bucket in self._buckets:
            bucket.ready = 0
            bucket.work = None
            bucket.pending = False
            bucket.skipped = False
            bucket.seen.clear()
        self._next_launch = 0

    def remove(self):
        for h in self._hooks:
            h.remove()
        self._hooks.clear()
