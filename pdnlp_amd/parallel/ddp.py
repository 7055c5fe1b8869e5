"""Our own DistributedDataParallel: bucketed gradient all-reduce over RCCL,
overlapped with backward on a side HIP stream.

MI355X-native equivalent of the torch DDP C++ reducer (reference
wraps at multi-gpu-distributed-cls.py:340-341). Design choices for the xGMI
fabric: bucket size is a first-class tunable (default 50 MB —
bigger than DDP's 25 MB because each of the 7 point-to-point links is
per-link bound at ≈153 GB/s and larger buckets amortize ring latency; sweep
with ``bucket_cap_mb``), gradients live as views into pre-allocated flat
buckets (no flatten copy per step), and each bucket's all-reduce launches on
a dedicated comm stream as soon as its last grad is produced, overlapping the
rest of backward.

Also folds in the Horovod-capability options: gradient
compression (= comm in bf16/fp16 while grads are fp32) and construction-time
parameter broadcast.
"""

from __future__ import annotations

import contextlib
from typing import Dict, List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn


class _Bucket:
    __slots__ = ("index", "params", "flat", "comm_flat", "views", "ready",
                 "expect", "work", "event")

    def __init__(self, index: int):
        self.index = index
        self.params: List[torch.nn.Parameter] = []
        self.flat: Optional[torch.Tensor] = None       # grad dtype
        self.comm_flat: Optional[torch.Tensor] = None  # compressed dtype or alias
        self.views: Dict[int, torch.Tensor] = {}
        self.ready = 0
        self.expect = 0
        self.work = None
        self.event: Optional[torch.cuda.Event] = None


class DistributedDataParallel(nn.Module):
    def __init__(self, module: nn.Module, bucket_cap_mb: float = 50.0,
                 grad_compression: str = "none",
                 overlap_comm: bool = True,
                 process_group=None,
                 broadcast_params: bool = True,
                 average_grads: bool = True):
        super().__init__()
        self.module = module
        self.pg = process_group
        self.world_size = dist.get_world_size(self.pg) if dist.is_initialized() else 1
        self.overlap = overlap_comm and dist.is_initialized()
        self.average = average_grads
        self.compression = grad_compression
        self._require_sync = True
        self._comm_stream = (torch.cuda.Stream()
                             if torch.cuda.is_available() and self.overlap else None)
        self._hooks = []
        self._buckets: List[_Bucket] = []
        self._param_bucket: Dict[int, _Bucket] = {}
        self._next_to_launch = 0

        if dist.is_initialized() and broadcast_params:
            self._broadcast_params()
        if dist.is_initialized():
            self._build_buckets(int(bucket_cap_mb * 1024 * 1024))
            self._register_hooks()

    # ------------------------------------------------------------------
    def _broadcast_params(self):
        """Construction-time rank-0 broadcast (what torch DDP and
        hvd.broadcast_parameters both do)."""
        for t in list(self.module.state_dict().values()):
            if isinstance(t, torch.Tensor) and t.numel() > 0:
                dist.broadcast(t.data, src=0, group=self.pg)

    def _build_buckets(self, cap_bytes: int):
        params = [p for p in self.module.parameters() if p.requires_grad]
        # reverse registration order approximates backward completion order
        params = params[::-1]
        cur = _Bucket(0)
        size = 0
        for p in params:
            nbytes = p.numel() * p.element_size()
            if size > 0 and size + nbytes > cap_bytes:
                self._buckets.append(cur)
                cur = _Bucket(len(self._buckets))
                size = 0
            cur.params.append(p)
            size += nbytes
        if cur.params:
            self._buckets.append(cur)
        comm_dtype = {"none": None, "bf16": torch.bfloat16,
                      "fp16": torch.float16}[self.compression]
        for b in self._buckets:
            total = sum(p.numel() for p in b.params)
            dev = b.params[0].device
            gdtype = b.params[0].dtype
            b.flat = torch.zeros(total, dtype=gdtype, device=dev)
            if comm_dtype is not None and comm_dtype != gdtype:
                b.comm_flat = torch.empty(total, dtype=comm_dtype, device=dev)
            else:
                b.comm_flat = b.flat
            off = 0
            for p in b.params:
                n = p.numel()
                view = b.flat[off:off + n].view_as(p)
                b.views[id(p)] = view
                p.grad = view
                off += n
                self._param_bucket[id(p)] = b
            b.expect = len(b.params)

    def _register_hooks(self):
        for b in self._buckets:
            for p in b.params:
                h = p.register_post_accumulate_grad_hook(self._grad_ready)
                self._hooks.append(h)

    # ------------------------------------------------------------------
    def _grad_ready(self, p: torch.nn.Parameter):
        if not self._require_sync:
            return
        b = self._param_bucket[id(p)]
        if p.grad is not b.views[id(p)]:
            # autograd allocated a fresh grad (e.g. after set_to_none);
            # fold it into the bucket view and restore the aliasing.
            b.views[id(p)].add_(p.grad)
            p.grad = b.views[id(p)]
        b.ready += 1
        if b.ready == b.expect:
            self._maybe_launch()

    def _maybe_launch(self):
        while (self._next_to_launch < len(self._buckets)
               and self._buckets[self._next_to_launch].ready
               == self._buckets[self._next_to_launch].expect):
            self._launch(self._buckets[self._next_to_launch])
            self._next_to_launch += 1

    def _launch(self, b: _Bucket):
        scale = 1.0 / self.world_size if self.average else 1.0
        from ..ops.functional import dw_stream
        if self._comm_stream is not None:
            cur = torch.cuda.current_stream()
            self._comm_stream.wait_stream(cur)
            if dw_stream.stream is not None:
                # side-stream dW GEMMs must land before the bucket reduces
                self._comm_stream.wait_stream(dw_stream.stream)
            with torch.cuda.stream(self._comm_stream):
                self._reduce_bucket(b, scale)
            b.event = torch.cuda.Event()
            b.event.record(self._comm_stream)
        else:
            dw_stream.join()
            self._reduce_bucket(b, scale)

    def _reduce_bucket(self, b: _Bucket, scale: float):
        if b.comm_flat is not b.flat:
            b.comm_flat.copy_(b.flat)
            dist.all_reduce(b.comm_flat, op=dist.ReduceOp.SUM, group=self.pg)
            b.flat.copy_(b.comm_flat)
        else:
            dist.all_reduce(b.flat, op=dist.ReduceOp.SUM, group=self.pg)
        if scale != 1.0:
            b.flat.mul_(scale)

    # ------------------------------------------------------------------
    def forward(self, *args, **kwargs):
        if dist.is_initialized() and self._require_sync:
            for b in self._buckets:
                b.ready = 0
                b.work = None
                b.event = None
            self._next_to_launch = 0
        return self.module(*args, **kwargs)

    def finalize_backward(self):
        """Block the compute stream on the last bucket's all-reduce. Call
        after ``loss.backward()`` and before ``optimizer.step()``."""
        if not dist.is_initialized() or not self._require_sync:
            return
        self._maybe_launch()
        if self._next_to_launch < len(self._buckets):
            missing = [i for i in range(self._next_to_launch, len(self._buckets))
                       if self._buckets[i].ready != self._buckets[i].expect]
            raise RuntimeError(
                f"buckets {missing} never became ready — a parameter did not "
                "receive a gradient this step (unused parameter?)")
        if self._comm_stream is not None:
            cur = torch.cuda.current_stream()
            for b in self._buckets:
                if b.event is not None:
                    cur.wait_event(b.event)

    @contextlib.contextmanager
    def no_sync(self):
        """Skip gradient sync (grad-accumulation micro-steps)."""
        old = self._require_sync
        self._require_sync = False
        try:
            yield
        finally:
            self._require_sync = old

    def zero_grad_buffers(self, set_to_none: bool = False):
        if not self._buckets:
            self.module.zero_grad(set_to_none=set_to_none)
            return
        for b in self._buckets:
            b.flat.zero_()

    def zero_grad(self, set_to_none: bool = True):  # match nn.Module API
        self.zero_grad_buffers(set_to_none=False)

    # passthroughs -----------------------------------------------------
    def state_dict(self, *a, **kw):
        return self.module.state_dict(*a, **kw)

    def load_state_dict(self, sd, *a, **kw):
        return self.module.load_state_dict(sd, *a, **kw)

    @property
    def config(self):
        return getattr(self.module, "config", None)
