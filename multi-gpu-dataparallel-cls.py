#!/usr/bin/env python
"""Single-process multi-GPU DataParallel.

Reference-equivalent of multi-gpu-dataparallel-cls.py: replicate the module,
scatter the batch, gather outputs — here over per-replica HIP streams and
xGMI peer copies. DDP remains the recommended path (README).

    python multi-gpu-dataparallel-cls.py
"""
from pdnlp_amd.cli import main

if __name__ == "__main__":
    main(strategy="dp")
