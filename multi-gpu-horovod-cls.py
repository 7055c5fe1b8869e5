#!/usr/bin/env python
"""Drop-in for the reference's multi-gpu-horovod-cls.py,
with no Horovod dependency: the same capability — rank-0 parameter +
optimizer-state broadcast, allreduce fused into ``optimizer.step`` via
gradient hooks, fp16/bf16 gradient compression — implemented natively on
RCCL over xGMI (pdnlp_amd.parallel.hooks). Same CLI as our other
entrypoints; launch with torchrun instead of horovodrun:

    python -m torch.distributed.run --nproc-per-node 2 \
        --master-addr 127.0.0.1 multi-gpu-horovod-cls.py
"""
from pdnlp_amd.cli import main

if __name__ == "__main__":
    main(strategy="hooks", extra_defaults={"grad_compression": "fp16"})
