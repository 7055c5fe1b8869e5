#!/usr/bin/env python
"""Accelerate-style wrapped training (SURVEY.md §2.1 Accelerate row).

Reference-equivalent of multi-gpu-accelerate-cls.py using our ``Accelerator``
(prepare()-wraps model/optimizer/loaders, auto topology, accelerator.backward).

    python -m torch.distributed.run --nproc-per-node 2 \
        --master-addr 127.0.0.1 multi-gpu-accelerate-cls.py
"""
import os

import torch
from torch.utils.data import DataLoader

from pdnlp_amd.config import Args
from pdnlp_amd.cli import build_dataloaders
from pdnlp_amd.data import LABELS
from pdnlp_amd.engine import Accelerator
from pdnlp_amd.models import build_model
from pdnlp_amd.ops.adamw import build_optimizer
from pdnlp_amd.utils import set_seed, rank0_print, save_checkpoint
from pdnlp_amd.engine.trainer import Trainer


def main():
    args = Args().apply_cli()
    args.strategy = "ddp"
    set_seed(args.seed)
    accelerator = Accelerator(
        mixed_precision=args.amp_dtype if args.amp else None,
        gradient_accumulation_steps=args.grad_accum_steps)
    args.local_rank = accelerator.local_rank
    args.rank = 0 if accelerator.is_main_process else 1
    import torch.distributed as dist
    if dist.is_initialized():
        args.rank = dist.get_rank()
        args.world_size = dist.get_world_size()
    train_loader, dev_loader, _ = build_dataloaders(args, 1, 0)  # unsharded
    model = build_model(args.model, model_path=args.model_path)
    optimizer = build_optimizer(model, lr=args.learning_rate,
                                weight_decay=args.weight_decay)
    model, optimizer, train_loader, dev_loader = accelerator.prepare(
        model, optimizer, train_loader, dev_loader)
    trainer = Trainer(args, model, optimizer, accelerator.device,
                      scaler=accelerator.scaler)
    sampler = getattr(train_loader, "sampler", None)
    minutes = trainer.train(train_loader, dev_loader, sampler)
    rank0_print(f"accelerate-mode done in {minutes:.4f} min")
    trainer.test(dev_loader, label_names=LABELS)


if __name__ == "__main__":
    main()
