#!/usr/bin/env python
"""Fabric-style single-GPU memory-optimization study (SURVEY.md §2.1 Fabric
row; reference: fabric/fabric-cls.py).

Feature matrix via flags like the reference (fabric/fabric-cls.py:200-218):
precision plugin (``--precision 16-mixed|bf16-mixed|32-true``), gradient
accumulation, SGD+cosine vs AdamW, on-device ``init_module``.

    python fabric/fabric-cls.py --precision bf16-mixed --grad-accum-steps 4
"""
import argparse
import os
import sys
import time

import torch
from torch.utils.data import DataLoader

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from pdnlp_amd.config import Args, BertConfig  # noqa: E402
from pdnlp_amd.cli import build_dataloaders  # noqa: E402
from pdnlp_amd.engine import Fabric  # noqa: E402
from pdnlp_amd.models import BertForSequenceClassification  # noqa: E402
from pdnlp_amd.ops.adamw import build_optimizer  # noqa: E402
from pdnlp_amd.utils import set_seed, rank0_print  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--precision", default="32-true",
                   choices=["32-true", "16-mixed", "bf16-mixed"])
    p.add_argument("--grad-accum-steps", type=int, default=1)
    p.add_argument("--optimizer", default="adamw", choices=["adamw", "sgd"])
    p.add_argument("--init-module-on-device", action="store_true")
    p.add_argument("--epochs", type=int, default=1)
    ns, rest = p.parse_known_args()

    args = Args()
    args.data_path = os.path.join(os.path.dirname(__file__), "..", "data",
                                  "train.json")
    args.epochs = ns.epochs
    args.grad_accum_steps = ns.grad_accum_steps
    args.optimizer = ns.optimizer
    args.lr_scheduler = "cosine" if ns.optimizer == "sgd" else "none"
    args.apply_cli(rest)
    set_seed(args.seed)

    fabric = Fabric(accelerator="auto", devices=1, precision=ns.precision)
    fabric.launch()
    if ns.init_module_on_device:
        with fabric.init_module():
            model = BertForSequenceClassification(BertConfig.bert_base_chinese())
    else:
        model = BertForSequenceClassification(BertConfig.bert_base_chinese())
    optimizer = build_optimizer(model, lr=args.learning_rate,
                                weight_decay=args.weight_decay,
                                optimizer=ns.optimizer)
    model, optimizer = fabric.setup(model, optimizer)
    train_loader, dev_loader, _ = build_dataloaders(args, 1, 0)
    train_loader, dev_loader = fabric.setup_dataloaders(train_loader, dev_loader)

    sched = None
    if ns.optimizer == "sgd":
        sched = torch.optim.lr_scheduler.CosineAnnealingLR(
            optimizer, T_max=len(train_loader) * args.epochs)

    t0 = time.time()
    model.train()
    step = 0
    for epoch in range(1, args.epochs + 1):
        for batch in train_loader:
            step += 1
            batch = {k: v.to(fabric.device) for k, v in batch.items()}
            out = model(input_ids=batch["input_ids"],
                        attention_mask=batch["attention_mask"],
                        token_type_ids=batch["token_type_ids"],
                        labels=batch["label"])
            fabric.backward(out.loss / ns.grad_accum_steps)
            if step % ns.grad_accum_steps == 0:
                fabric.optimizer_step(optimizer, model)
                optimizer.zero_grad(set_to_none=False)
                if sched is not None:
                    sched.step()
            if step % 10 == 0:
                fabric.print(f"【train】 epoch：{epoch}/{args.epochs} "
                             f"step：{step} loss：{out.loss.item():.6f}")
    mins = (time.time() - t0) / 60.0
    mem = (torch.cuda.max_memory_allocated() / 2**30
           if torch.cuda.is_available() else 0.0)
    fabric.print(f"耗时：{mins:.4f}分钟  peak-memory：{mem:.2f} GB")


if __name__ == "__main__":
    main()
