"""Optimizer-hooked all-reduce (the Horovod capability).

Reference: ``hvd.DistributedOptimizer`` wraps the optimizer so gradient
all-reduce happens inside ``optimizer.step`` with tensor fusion and fp16
compression; parameters and optimizer state broadcast from rank 0 at start
(multi-gpu-horovod-cls.py:338-349). Here: ``DistributedOptimizer`` fuses
grads into flat buffers (bf16/fp16 compression options), all-reduces over
RCCL, then runs the wrapped optimizer.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist


def broadcast_parameters(module: torch.nn.Module, root_rank: int = 0,
                         group=None) -> None:
    if not dist.is_initialized():
        return
    for t in module.state_dict().values():
        if isinstance(t, torch.Tensor) and t.numel() > 0:
            dist.broadcast(t.data, src=root_rank, group=group)


def broadcast_optimizer_state(optimizer: torch.optim.Optimizer,
                              root_rank: int = 0, group=None) -> None:
    if not dist.is_initialized():
        return
    for state in optimizer.state.values():
        for v in state.values():
            if isinstance(v, torch.Tensor) and v.numel() > 0:
                dist.broadcast(v.data, src=root_rank, group=group)


class Compression:
    """hvd.Compression equivalent: dtype used on the wire."""
    none = None
    fp16 = torch.float16
    bf16 = torch.bfloat16


class DistributedOptimizer:
    def __init__(self, optimizer: torch.optim.Optimizer,
                 compression: Optional[torch.dtype] = None,
                 fusion_mb: float = 64.0, group=None, average: bool = True):
        self.optimizer = optimizer
        self.compression = compression
        self.fusion_bytes = int(fusion_mb * 1024 * 1024)
        self.group = group
        self.average = average
        self.param_groups = optimizer.param_groups
        self.state = optimizer.state

    def _params_with_grads(self):
        for g in self.optimizer.param_groups:
            for p in g["params"]:
                if p.grad is not None:
                    yield p

    @torch.no_grad()
    def _allreduce_grads(self):
        if not dist.is_initialized():
            return
        world = dist.get_world_size(self.group)
        params = list(self._params_with_grads())
        # fuse into buffers of <= fusion_bytes (horovod tensor-fusion equivalent)
        i = 0
        while i < len(params):
            chunk, size = [], 0
            while i < len(params) and (not chunk or size <
                                       self.fusion_bytes):
                p = params[i]
                chunk.append(p)
                size += p.grad.numel() * p.grad.element_size()
                i += 1
            dtype = self.compression or chunk[0].grad.dtype
            flat = torch.cat([p.grad.reshape(-1).to(dtype) for p in chunk])
            dist.all_reduce(flat, op=dist.ReduceOp.SUM, group=self.group)
            if self.average:
                flat = flat / world
            off = 0
            for p in chunk:
                n = p.grad.numel()
                p.grad.copy_(flat[off:off + n].view_as(p.grad).to(p.grad.dtype))
                off += n

    def step(self, closure=None):
        self._allreduce_grads()
        return self.optimizer.step(closure)

    def zero_grad(self, set_to_none: bool = True):
        self.optimizer.zero_grad(set_to_none=set_to_none)

    def state_dict(self):
        return self.optimizer.state_dict()

    def load_state_dict(self, sd):
        self.optimizer.load_state_dict(sd)
