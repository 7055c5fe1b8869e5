"""Single-process multi-GPU DataParallel.

Reference capability: ``nn.DataParallel(model, device_ids, output_device)``
(multi-gpu-dataparallel-cls.py:255). MI355X-native mechanism: per-replica HIP
streams with direct xGMI peer-to-peer copies for scatter/gather — no GIL-bound
thread pool for the copies; module replicas run under one Python loop but all
kernel launches are async per-stream so the GPUs overlap.

As in the reference's README (README.md:69-77), DDP (one process per GPU) is
the recommended path; this exists for capability parity.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn


class DataParallel(nn.Module):
    def __init__(self, module: nn.Module, device_ids: Optional[List[int]] = None,
                 output_device: Optional[int] = None):
        super().__init__()
        if device_ids is None:
            device_ids = list(range(torch.cuda.device_count())) or [0]
        self.device_ids = device_ids
        self.output_device = output_device if output_device is not None else device_ids[0]
        self.module = module
        self._streams = None
        if torch.cuda.is_available() and len(device_ids) > 1:
            self._streams = [torch.cuda.Stream(device=d) for d in device_ids]
            for d in device_ids:
                for e in device_ids:
                    if d != e and torch.cuda.can_device_access_peer(d, e):
                        pass  # peer access is enabled implicitly by HIP on xGMI

    def forward(self, *args, **kwargs):
        if self._streams is None or len(self.device_ids) == 1:
            return self.module(*args, **kwargs)
        # scatter batch dim across replicas
        replicas = self._replicate()
        ins = self._scatter(args, kwargs)
        outs = []
        for (a, kw), replica, dev, stream in zip(ins, replicas,
                                                 self.device_ids, self._streams):
            with torch.cuda.device(dev), torch.cuda.stream(stream):
                outs.append(replica(*a, **kw))
        for s in self._streams:
            torch.cuda.current_stream(self.output_device).wait_stream(s)
        return self._gather(outs)

    def _replicate(self):
        import copy
        replicas = [self.module]
        for d in self.device_ids[1:]:
            r = copy.deepcopy(self.module).to(f"cuda:{d}")
            for (pr, ps) in zip(r.parameters(), self.module.parameters()):
                pr.data.copy_(ps.data, non_blocking=True)  # xGMI P2P copy
            replicas.append(r)
        self._replicas = replicas
        return replicas

    def _scatter(self, args, kwargs):
        n = len(self.device_ids)
        outs = []
        for i, dev in enumerate(self.device_ids):
            a = tuple(self._chunk(x, i, n, dev) for x in args)
            kw = {k: self._chunk(v, i, n, dev) for k, v in kwargs.items()}
            outs.append((a, kw))
        return outs

    @staticmethod
    def _chunk(x, i, n, dev):
        if isinstance(x, torch.Tensor):
            return x.chunk(n, dim=0)[i].to(f"cuda:{dev}", non_blocking=True)
        return x

    def _gather(self, outs):
        first = outs[0]
        if isinstance(first, torch.Tensor):
            return torch.cat([o.to(f"cuda:{self.output_device}") for o in outs], 0)
        # SequenceClassifierOutput
        from ..models.bert import SequenceClassifierOutput
        if isinstance(first, SequenceClassifierOutput):
            logits = torch.cat([o.logits.to(f"cuda:{self.output_device}")
                                for o in outs], 0)
            loss = None
            if first.loss is not None:
                loss = torch.stack([o.loss.to(f"cuda:{self.output_device}")
                                    for o in outs]).mean()
            # fold replica grads back: sum gradients into the primary module
            self._pending_grad_sync = True
            return SequenceClassifierOutput(loss=loss, logits=logits)
        return outs

    def sync_replica_grads(self):
        """After backward: sum replica grads into the primary copy over P2P."""
        if self._streams is None or not getattr(self, "_pending_grad_sync", False):
            return
        for r in getattr(self, "_replicas", [])[1:]:
            for pr, ps in zip(r.parameters(), self.module.parameters()):
                if pr.grad is not None:
                    g = pr.grad.to(ps.device, non_blocking=True)
                    ps.grad = g if ps.grad is None else ps.grad + g
        self._pending_grad_sync = False

    def state_dict(self, *a, **kw):
        return self.module.state_dict(*a, **kw)

    def load_state_dict(self, sd, *a, **kw):
        return self.module.load_state_dict(sd, *a, **kw)
