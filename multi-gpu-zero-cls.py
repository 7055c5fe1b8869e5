#!/usr/bin/env python
"""ZeRO-sharded data parallelism — the DeepSpeed capability.

Reference-equivalent of multi-gpu-deepspeed-cls.py: grad reduce-scatter over
xGMI, sharded fused AdamW with fp32 master weights, param all-gather,
activation checkpointing with optional CPU offload, sharded checkpoints with
a ``tools/zero_to_fp32.py`` consolidation tool.

    python -m torch.distributed.run --nproc-per-node 2 \
        --master-addr 127.0.0.1 multi-gpu-zero-cls.py --amp true
"""
from pdnlp_amd.cli import main

if __name__ == "__main__":
    main(strategy="zero", amp=True, amp_dtype="bf16",
         extra_defaults={"activation_checkpointing": True})
