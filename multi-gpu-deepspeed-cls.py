#!/usr/bin/env python
"""Drop-in for the reference's multi-gpu-deepspeed-cls.py,
with no DeepSpeed dependency: the ZeRO capability — gradient
reduce-scatter, sharded fused-AdamW state, parameter allgather, fp16/bf16
training, activation checkpointing with optional CPU offload, sharded
checkpoints + tools/zero_to_fp32.py consolidation — implemented natively
on RCCL over xGMI (pdnlp_amd.parallel.zero).

    python -m torch.distributed.run --nproc-per-node 2 \
        --master-addr 127.0.0.1 multi-gpu-deepspeed-cls.py
"""
from pdnlp_amd.cli import main

if __name__ == "__main__":
    main(strategy="zero")
