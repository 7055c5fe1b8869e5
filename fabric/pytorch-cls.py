#!/usr/bin/env python
"""Plain single-GPU control twin of fabric-cls.py for the memory/time
comparison table (reference: fabric/pytorch-cls.py, fabric/README.md:31-39).

    python fabric/pytorch-cls.py
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from pdnlp_amd.config import Args, BertConfig  # noqa: E402
from pdnlp_amd.cli import build_dataloaders  # noqa: E402
from pdnlp_amd.models import BertForSequenceClassification  # noqa: E402
from pdnlp_amd.ops.adamw import build_optimizer  # noqa: E402
from pdnlp_amd.utils import set_seed, rank0_print  # noqa: E402


def main():
    args = Args()
    args.data_path = os.path.join(os.path.dirname(__file__), "..", "data",
                                  "train.json")
    args.apply_cli()
    set_seed(args.seed)
    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    model = BertForSequenceClassification(
        BertConfig.bert_base_chinese()).to(device)
    optimizer = build_optimizer(model, lr=args.learning_rate,
                                weight_decay=args.weight_decay)
    train_loader, dev_loader, _ = build_dataloaders(args, 1, 0)

    t0 = time.time()
    model.train()
    step = 0
    for epoch in range(1, args.epochs + 1):
        for batch in train_loader:
            step += 1
            batch = {k: v.to(device) for k, v in batch.items()}
            out = model(input_ids=batch["input_ids"],
                        attention_mask=batch["attention_mask"],
                        token_type_ids=batch["token_type_ids"],
                        labels=batch["label"])
            optimizer.zero_grad(set_to_none=False)
            out.loss.backward()
            optimizer.step()
            if step % 10 == 0:
                rank0_print(f"【train】 epoch：{epoch}/{args.epochs} "
                            f"step：{step} loss：{out.loss.item():.6f}")
    mins = (time.time() - t0) / 60.0
    mem = (torch.cuda.max_memory_allocated() / 2**30
           if torch.cuda.is_available() else 0.0)
    rank0_print(f"耗时：{mins:.4f}分钟  peak-memory：{mem:.2f} GB")


if __name__ == "__main__":
    main()
