"""Dynamic-loss-scaling GradScaler for the fp16 AMP path (K13).

Reference: ``torch.cuda.amp.GradScaler`` — ``scaler.scale(loss).backward();
scaler.step(optimizer); scaler.update()``
(multi-gpu-distributed-mp-amp-cls.py:161,173-175). Same API here. The
unscale + non-finite check runs on the HIP multi-tensor kernel when built
(device-side found-inf flag, one launch per chunk), torch ops otherwise.
Scale bookkeeping (growth 2× per ``growth_interval`` good steps, 0.5 backoff
on overflow) is host-side.
"""

from __future__ import annotations

from typing import Iterable

import torch

from ..ops import ext, hip_enabled


class GradScaler:
    def __init__(self, init_scale: float = 2.0 ** 16, growth_factor: float = 2.0,
                 backoff_factor: float = 0.5, growth_interval: int = 2000,
                 enabled: bool = True):
        self._scale = float(init_scale)
        self.growth_factor = growth_factor
        self.backoff_factor = backoff_factor
        self.growth_interval = growth_interval
        self.enabled = enabled
        self._good_steps = 0
        self._found_inf = False
        self._unscaled = False

    def get_scale(self) -> float:
        return self._scale if self.enabled else 1.0

    def scale(self, loss: torch.Tensor) -> torch.Tensor:
        if not self.enabled:
            return loss
        return loss * self._scale

    def _grads(self, optimizer) -> Iterable[torch.Tensor]:
        for group in optimizer.param_groups:
            for p in group["params"]:
                if p.grad is not None:
                    yield p.grad

    def unscale_(self, optimizer) -> None:
        if not self.enabled or self._unscaled:
            return
        inv = 1.0 / self._scale
        grads = list(self._grads(optimizer))
        if grads and hip_enabled(grads[0]) and \
                getattr(ext(), "multi_tensor_unscale", None) is not None:
            found = torch.zeros(1, dtype=torch.float32, device=grads[0].device)
            for i in range(0, len(grads), 512):
                ext().multi_tensor_unscale(grads[i:i + 512], found, inv)
            self._found_inf = bool(found.item() != 0)
        else:
            found = False
            for g in grads:
                g.mul_(inv)
                if not found and not torch.isfinite(g).all():
                    found = True
            self._found_inf = bool(found)
        self._unscaled = True

    def step(self, optimizer, *args, **kwargs):
        if not self.enabled:
            return optimizer.step(*args, **kwargs)
        self.unscale_(optimizer)
        if self._found_inf:
            return None  # skip the step on overflow
        return optimizer.step(*args, **kwargs)

    def update(self) -> None:
        if not self.enabled:
            return
        if self._found_inf:
            self._scale = max(self._scale * self.backoff_factor, 1.0)
            self._good_steps = 0
        else:
            self._good_steps += 1
            if self._good_steps >= self.growth_interval:
                self._scale *= self.growth_factor
                self._good_steps = 0
        self._found_inf = False
        self._unscaled = False
