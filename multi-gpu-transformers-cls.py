#!/usr/bin/env python
"""HF-Trainer-style fully delegated training (SURVEY.md §2.1 HF Trainer row).

Reference-equivalent of multi-gpu-transformers-cls.py: TrainingArguments +
Trainer(.train/.evaluate), steps-based eval/save, load-best-at-end, the
``labels`` collator key.

    python -m torch.distributed.run --nproc-per-node 2 \
        --master-addr 127.0.0.1 multi-gpu-transformers-cls.py
"""
import os

import numpy as np

from pdnlp_amd.config import Args
from pdnlp_amd.data import (ClsDataset, Collate, SyntheticClsDataset,
                            build_tokenizer, load_data, train_dev_split)
from pdnlp_amd.engine import HFStyleTrainer, TrainingArguments
from pdnlp_amd.models import build_model
from pdnlp_amd.utils import set_seed


def compute_metrics(logits: np.ndarray, labels: np.ndarray):
    return {"accuracy": float((logits.argmax(-1) == labels).mean())}


def main():
    base = Args().apply_cli()
    set_seed(base.seed)
    hf_args = TrainingArguments(
        output_dir=base.output_dir,
        per_device_train_batch_size=base.train_batch_size,
        per_device_eval_batch_size=base.dev_batch_size,
        num_train_epochs=base.epochs,
        learning_rate=base.learning_rate,
        weight_decay=base.weight_decay,
        bf16=True,                      # fp16 in the reference; bf16-first here
        eval_steps=base.eval_step, save_steps=base.eval_step,
        seed=base.seed,
    )
    if os.path.isfile(base.data_path):
        data = load_data(base.data_path, limit=base.data_limit)
        train_data, dev_data = train_dev_split(data, base.ratio, base.seed)
        tok = build_tokenizer(base.model_path)
        collate = Collate(tok, base.max_seq_len, label_key="labels")
        train_ds, dev_ds = ClsDataset(train_data), ClsDataset(dev_data)
    else:
        n_train = int(base.data_limit * base.ratio)
        train_ds = SyntheticClsDataset(n_train, base.max_seq_len, seed=base.seed)
        dev_ds = SyntheticClsDataset(base.data_limit - n_train,
                                     base.max_seq_len, seed=base.seed + 1)
        collate = Collate(None, base.max_seq_len, label_key="labels")
    model = build_model(base.model, model_path=base.model_path)
    trainer = HFStyleTrainer(model, hf_args, train_dataset=train_ds,
                             eval_dataset=dev_ds, data_collator=collate,
                             compute_metrics=compute_metrics)
    trainer.train()
    trainer.evaluate()
    trainer.save_model()


if __name__ == "__main__":
    main()
