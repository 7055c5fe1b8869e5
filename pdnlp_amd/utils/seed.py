"""Seeding identical in effect to the reference's ``set_seed``
(reference: single-gpu-cls.py:14-23): random, numpy, torch CPU + all GPUs."""

import os
import random

import numpy as np
import torch


def set_seed(seed: int = 123, deterministic: bool = False) -> None:
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)
    os.environ.setdefault("PYTHONHASHSEED", str(seed))
    if deterministic:
        torch.use_deterministic_algorithms(True, warn_only=True)
