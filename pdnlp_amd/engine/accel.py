"""Accelerate-style high-level API (SURVEY.md §2.1 Accelerate row).

Reference ergonomics (multi-gpu-accelerate-cls.py:289-294): ``Accelerator()``
auto-detects topology from the launcher env, ``prepare()`` wraps
model/optimizer/dataloaders (injecting a DistributedSampler), and
``accelerator.backward(loss)`` hides AMP/scaling. Thin layer over our engine.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist
from torch.utils.data import DataLoader

from ..amp import GradScaler, cast_model_to
from ..data.sampler import DistributedSampler
from ..parallel.bootstrap import init_distributed
from ..parallel.ddp import DistributedDataParallel


class Accelerator:
    def __init__(self, mixed_precision: Optional[str] = None,
                 gradient_accumulation_steps: int = 1,
                 bucket_cap_mb: float = 50.0):
        self.local_rank = init_distributed()
        self.mixed_precision = mixed_precision  # None|"bf16"|"fp16"
        self.grad_accum = gradient_accumulation_steps
        self.bucket_cap_mb = bucket_cap_mb
        self.device = torch.device(f"cuda:{self.local_rank}"
                                   if torch.cuda.is_available() else "cpu")
        self.scaler = GradScaler() if mixed_precision == "fp16" else None
        self._micro = 0

    @property
    def is_main_process(self) -> bool:
        return (not dist.is_initialized()) or dist.get_rank() == 0

    @property
    def num_processes(self) -> int:
        return dist.get_world_size() if dist.is_initialized() else 1

    def prepare(self, *objs):
        out = []
        for obj in objs:
            if isinstance(obj, DataLoader):
                out.append(self._prepare_loader(obj))
            elif isinstance(obj, torch.nn.Module):
                out.append(self._prepare_model(obj))
            else:
                out.append(obj)  # optimizers/schedulers pass through
        return out[0] if len(out) == 1 else tuple(out)

    def _prepare_model(self, model):
        if self.mixed_precision in ("bf16", "fp16"):
            model = cast_model_to(model, self.mixed_precision)
        model = model.to(self.device)
        if dist.is_initialized():
            model = DistributedDataParallel(model,
                                            bucket_cap_mb=self.bucket_cap_mb)
        return model

    def _prepare_loader(self, loader: DataLoader) -> DataLoader:
        if not dist.is_initialized():
            return loader
        sampler = DistributedSampler(loader.dataset,
                                     shuffle=not isinstance(
                                         loader.sampler,
                                         torch.utils.data.SequentialSampler))
        return DataLoader(loader.dataset, batch_size=loader.batch_size,
                          sampler=sampler, num_workers=loader.num_workers,
                          collate_fn=loader.collate_fn,
                          pin_memory=loader.pin_memory,
                          drop_last=loader.drop_last)

    def backward(self, loss):
        self._micro += 1
        loss = loss / max(self.grad_accum, 1)
        if self.scaler is not None:
            loss = self.scaler.scale(loss)
        loss.backward()

    def sync_gradients_ready(self, model) -> bool:
        return self._micro % max(self.grad_accum, 1) == 0

    def step(self, optimizer, model=None):
        if isinstance(model, DistributedDataParallel):
            model.finalize_backward()
        if self.scaler is not None:
            self.scaler.step(optimizer)
            self.scaler.update()
        else:
            optimizer.step()

    def wait_for_everyone(self):
        if dist.is_initialized():
            dist.barrier()

    def gather(self, tensor: torch.Tensor) -> torch.Tensor:
        if not dist.is_initialized():
            return tensor
        outs = [torch.zeros_like(tensor) for _ in range(self.num_processes)]
        dist.all_gather(outs, tensor.contiguous())
        return torch.cat(outs, dim=0)

    def unwrap_model(self, model):
        return model.module if hasattr(model, "module") else model
