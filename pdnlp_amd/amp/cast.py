"""AMP policy (K13/K14).

bf16-first on CDNA4: no loss scaling needed, MFMA bf16 rate is the chip's
dense peak. The fp16+GradScaler path is kept for parity with the reference's
fp16 AMP (multi-gpu-distributed-mp-amp-cls.py:160-175).

Mechanism: the model's parameters are cast to the compute dtype and the
optimizer keeps fp32 master weights (FusedAdamW ``master_weights=True``) —
the production-style fixed-precision scheme rather than autocast's per-op
re-casting (which would re-cast every weight every step).
"""

from __future__ import annotations

import torch

_DTYPES = {"bf16": torch.bfloat16, "fp16": torch.float16, "fp32": torch.float32}


def amp_dtype_of(name: str) -> torch.dtype:
    return _DTYPES[name]


def cast_model_to(model: torch.nn.Module, dtype_name: str) -> torch.nn.Module:
    dtype = amp_dtype_of(dtype_name)
    if dtype == torch.float32:
        return model.float()
    return model.to(dtype)
