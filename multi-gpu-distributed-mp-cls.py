#!/usr/bin/env python
"""DDP via mp.spawn self-launch (no AMP).

Reference-equivalent of multi-gpu-distributed-mp-cls.py: the script forks one
worker per GPU itself (TCP rendezvous on 127.0.0.1).

    python multi-gpu-distributed-mp-cls.py [--world-size N]
"""
from pdnlp_amd.cli import main

if __name__ == "__main__":
    main(strategy="ddp", use_spawn=True)
