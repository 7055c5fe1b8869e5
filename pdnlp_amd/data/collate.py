"""Batch collation: text → fixed-length int64 tensors.

Reference equivalent: ``Collate.collate_fn`` (single-gpu-cls.py:44-84) runs
``tokenizer.encode_plus`` per sample in a Python loop — the CPU-side hot spot
SURVEY.md §2.2 flags. Here encoding happens per sample too (tokenization is
inherently per-text) but tensorization is one ``torch.tensor`` call per batch,
and pre-tokenized dict samples (SyntheticClsDataset) pass through with a
single stack.
"""

from __future__ import annotations

import torch


class Collate:
    def __init__(self, tokenizer, max_seq_len: int = 128, label_key: str = "label"):
        self.tokenizer = tokenizer
        self.max_seq_len = max_seq_len
        self.label_key = label_key  # HF-Trainer mode uses "labels"

    def __call__(self, batch):
        return self.collate_fn(batch)

    def collate_fn(self, batch):
        if isinstance(batch[0], dict):  # pre-tokenized
            out = {
                "input_ids": torch.stack([b["input_ids"] for b in batch]),
                "attention_mask": torch.stack([b["attention_mask"] for b in batch]),
                "token_type_ids": torch.stack([b["token_type_ids"] for b in batch]),
                self.label_key: torch.stack([b.get("label", b.get("labels")) for b in batch]),
            }
            return out
        ids_l, mask_l, type_l, labels = [], [], [], []
        for text, label in batch:
            ids, mask, type_ids = self.tokenizer.encode(text, self.max_seq_len)
            ids_l.append(ids)
            mask_l.append(mask)
            type_l.append(type_ids)
            labels.append(int(label))
        return {
            "input_ids": torch.tensor(ids_l, dtype=torch.long),
            "attention_mask": torch.tensor(mask_l, dtype=torch.long),
            "token_type_ids": torch.tensor(type_l, dtype=torch.long),
            self.label_key: torch.tensor(labels, dtype=torch.long),
        }
