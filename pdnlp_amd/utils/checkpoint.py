"""Checkpoint save/load.

Contract: save *unwrapped* HF-layout state dicts so a bare
``BertForSequenceClassification`` can load them without the ``module.`` prefix
strip the reference needs (reference: test.py:96-101, README.md:663-670).
``strip_module_prefix`` is still provided so reference-produced checkpoints
load too.
"""

from __future__ import annotations

import os
from typing import Optional

import torch


def strip_module_prefix(state_dict: dict) -> dict:
    """Strip DDP/DP ``module.`` prefixes (reference: test.py:96-101)."""
    out = {}
    for k, v in state_dict.items():
        out[k[len("module."):] if k.startswith("module.") else k] = v
    return out


def unwrap_model(model):
    while hasattr(model, "module"):
        model = model.module
    return model


def save_checkpoint(model, path: str, optimizer=None, extra: Optional[dict] = None,
                    rank: int = 0) -> None:
    """Rank-0 save of the unwrapped model state dict (+ optional optimizer/
    step state for resume — a capability the reference lacks)."""
    if rank != 0:
        return
    os.makedirs(os.path.dirname(os.path.abspath(path)) or ".", exist_ok=True)
    m = unwrap_model(model)
    if optimizer is None and not extra:
        torch.save(m.state_dict(), path)
    else:
        payload = {"model": m.state_dict()}
        if optimizer is not None:
            payload["optimizer"] = optimizer.state_dict()
        if extra:
            payload.update(extra)
        torch.save(payload, path)


def load_checkpoint(model, path: str, map_location="cpu", optimizer=None) -> dict:
    """Load either a bare state dict or a {model, optimizer, ...} payload into
    an unwrapped (or wrapped) model; tolerates ``module.`` prefixes."""
    payload = torch.load(path, map_location=map_location, weights_only=False)
    extra = {}
    if isinstance(payload, dict) and "model" in payload and any(
            isinstance(v, torch.Tensor) for v in payload["model"].values()):
        sd = payload["model"]
        if optimizer is not None and "optimizer" in payload:
            optimizer.load_state_dict(payload["optimizer"])
        extra = {k: v for k, v in payload.items() if k not in ("model", "optimizer")}
    else:
        sd = payload
    sd = strip_module_prefix(sd)
    unwrap_model(model).load_state_dict(sd)
    return extra
