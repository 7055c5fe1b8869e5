#!/usr/bin/env python
"""Optimizer-hooked allreduce-DP — the Horovod capability.

Reference-equivalent of multi-gpu-horovod-cls.py without the Horovod
dependency: rank-0 parameter + optimizer-state broadcast, gradient all-reduce
fused inside ``optimizer.step`` with bf16/fp16 gradient compression.

    python -m torch.distributed.run --nproc-per-node 2 \
        --master-addr 127.0.0.1 multi-gpu-hooks-cls.py --grad-compression fp16
"""
from pdnlp_amd.cli import main

if __name__ == "__main__":
    main(strategy="hooks", extra_defaults={"grad_compression": "fp16"})
