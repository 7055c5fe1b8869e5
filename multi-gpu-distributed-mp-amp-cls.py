#!/usr/bin/env python
"""DDP + AMP via mp.spawn.

Reference-equivalent of multi-gpu-distributed-mp-amp-cls.py, bf16-first on
CDNA4 (no loss scaling needed); ``--amp-dtype fp16`` selects the fp16 +
GradScaler parity path. The reference's missing-zero_grad AMP bug
(SURVEY.md §3.2 note) is fixed, not replicated.

    python multi-gpu-distributed-mp-amp-cls.py [--world-size N --amp-dtype bf16]
"""
from pdnlp_amd.cli import main

if __name__ == "__main__":
    main(strategy="ddp", use_spawn=True, amp=True, amp_dtype="bf16")
