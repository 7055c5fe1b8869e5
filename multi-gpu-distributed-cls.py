#!/usr/bin/env python
"""DDP via external launcher (the flagship path).

Reference-equivalent of multi-gpu-distributed-cls.py: one process per GPU,
env-var rendezvous, our bucketed-RCCL DDP reducer overlapped with backward.

    python -m torch.distributed.run --nproc-per-node 2 \
        --master-addr 127.0.0.1 multi-gpu-distributed-cls.py
"""
from pdnlp_amd.cli import main

if __name__ == "__main__":
    main(strategy="ddp")
