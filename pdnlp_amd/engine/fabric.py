"""Fabric-style API (SURVEY.md §2.1 Lightning Fabric row).

Reference ergonomics (fabric/fabric-cls.py:266-292): precision plugins
("16-mixed" / "bf16-mixed" / "32-true"), ``init_module()`` on-device
construction, ``setup(model, optimizer)``, ``setup_dataloaders``,
``fabric.backward(loss)``, gradient accumulation — the single-GPU
memory-optimization study surface.
"""

from __future__ import annotations

import contextlib


import torch
import torch.distributed as dist
from torch.utils.data import DataLoader

from ..amp import GradScaler, cast_model_to
from ..data.sampler import DistributedSampler
from ..parallel.bootstrap import init_distributed
from ..parallel.ddp import DistributedDataParallel

_PRECISION_DTYPE = {"32-true": torch.float32, "16-mixed": torch.float16,
                    "bf16-mixed": torch.bfloat16, "bf16-true": torch.bfloat16}


class Fabric:
    def __init__(self, accelerator: str = "auto", devices: int = 1,
                 precision: str = "32-true"):
        if precision not in _PRECISION_DTYPE:
            raise ValueError(f"unknown precision {precision}")
        self.precision = precision
        self.devices = devices
        self.local_rank = 0
        self.device = torch.device("cpu")
        self.scaler = GradScaler() if precision == "16-mixed" else None
        self._launched = False

    def launch(self):
        self.local_rank = init_distributed(
            world_size=self.devices if self.devices > 1 else None)
        use_cuda = torch.cuda.is_available()
        self.device = torch.device(f"cuda:{self.local_rank}" if use_cuda else "cpu")
        self._launched = True

    @contextlib.contextmanager
    def init_module(self):
        """Construct the model directly on the target device/dtype (the
        memory-saving on-device init of fabric/fabric-cls.py:273-275)."""
        dtype = _PRECISION_DTYPE[self.precision]
        old = torch.get_default_dtype()
        if dtype in (torch.float32, torch.bfloat16):
            torch.set_default_dtype(dtype if dtype != torch.float16 else old)
        try:
            with torch.device(self.device):
                yield
        finally:
            torch.set_default_dtype(old)

    def setup(self, model: torch.nn.Module, optimizer=None):
        dtype = _PRECISION_DTYPE[self.precision]
        if dtype != torch.float32:
            model = cast_model_to(
                model, "bf16" if dtype == torch.bfloat16 else "fp16")
        model = model.to(self.device)
        if dist.is_initialized():
            model = DistributedDataParallel(model)
        self._model = model
        if optimizer is None:
            return model
        return model, optimizer

    def setup_dataloaders(self, *loaders):
        outs = []
        for loader in loaders:
            if dist.is_initialized():
                sampler = DistributedSampler(loader.dataset)
                loader = DataLoader(loader.dataset, batch_size=loader.batch_size,
                                    sampler=sampler,
                                    collate_fn=loader.collate_fn,
                                    num_workers=loader.num_workers)
            outs.append(loader)
        return outs[0] if len(outs) == 1 else tuple(outs)

    def backward(self, loss):
        if self.scaler is not None:
            loss = self.scaler.scale(loss)
        loss.backward()

    def optimizer_step(self, optimizer, model=None):
        if isinstance(model or getattr(self, "_model", None),
                      DistributedDataParallel):
            (model or self._model).finalize_backward()
        if self.scaler is not None:
            self.scaler.step(optimizer)
            self.scaler.update()
        else:
            optimizer.step()

    def print(self, *args, **kwargs):
        if (not dist.is_initialized()) or dist.get_rank() == 0:
            print(*args, **kwargs)
