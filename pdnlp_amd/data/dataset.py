"""Data loading.

Same on-disk contract as the reference: ``data/train.json`` is a JSON list of
``["space-tokenized text", label_id]`` pairs with 6 emotion classes
(reference: single-gpu-cls.py:26-41, label map :212-219). The loader strips
the spaces, takes the first ``limit`` samples, shuffles with the run's seed
and splits 92/8 (reference: single-gpu-cls.py:227-232).

Also provides a synthetic dataset (random token ids of the same shape) for
the no-network benchmark path (BASELINE.json: synthetic data, random-init
weights).
"""

from __future__ import annotations

import json
import random
from typing import List, Optional, Tuple

import torch
from torch.utils.data import Dataset

LABELS = ["其他", "喜好", "悲伤", "厌恶", "愤怒", "高兴"]
label2id = {l: i for i, l in enumerate(LABELS)}
id2label = {i: l for i, l in enumerate(LABELS)}


def load_data(path: str, limit: Optional[int] = None) -> List[Tuple[str, int]]:
    with open(path, encoding="utf-8") as f:
        raw = json.load(f)
    out = []
    for item in raw:
        text, label = item[0], int(item[1])
        out.append(("".join(str(text).split(" ")), label))
        if limit is not None and len(out) >= limit:
            break
    return out


def train_dev_split(data: List[Tuple[str, int]], ratio: float = 0.92,
                    seed: int = 123, shuffle: bool = True):
    data = list(data)
    if shuffle:
        random.Random(seed).shuffle(data)
    n_train = int(len(data) * ratio)
    return data[:n_train], data[n_train:]


class ClsDataset(Dataset):
    """Trivial dataset over (text, label) tuples
    (reference: multi-gpu-distributed-cls.py:47-55)."""

    def __init__(self, data: List[Tuple[str, int]]):
        self.data = list(data)

    def __len__(self):
        return len(self.data)

    def __getitem__(self, idx):
        return self.data[idx]


class SyntheticClsDataset(Dataset):
    """Pre-tokenized synthetic classification samples of the reference shape.

    Emits dict batches directly (no tokenizer needed): random token ids in
    [CLS] ... [SEP] form, full attention mask, zero token-type ids.
    Deterministic per (seed, index).

    ``learnable=True`` (default) makes the label a function of the token
    content (the bucketed second token id): a correctly-wired model LEARNS
    it, so epoch runs on synthetic data show real loss curves and dev
    accuracy — an end-to-end numerics check of every kernel in the stack
    (random labels would hide a broken gradient anywhere). ``learnable=
    False`` gives i.i.d. random labels (pure-throughput shape)."""

    def __init__(self, n: int, seq_len: int = 128, vocab_size: int = 21128,
                 num_labels: int = 6, seed: int = 123, var_len: bool = False,
                 learnable: bool = True):
        self.learnable = learnable
        self.n = n
        self.seq_len = seq_len
        self.vocab_size = vocab_size
        self.num_labels = num_labels
        self.seed = seed
        self.var_len = var_len

    def __len__(self):
        return self.n

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self.seed * 1000003 + idx)
        L = self.seq_len
        if self.var_len:
            L = int(torch.randint(8, self.seq_len + 1, (1,), generator=g))
        ids = torch.randint(106, self.vocab_size, (self.seq_len,), generator=g)
        ids[0] = 101  # [CLS]
        ids[L - 1] = 102  # [SEP]
        mask = torch.zeros(self.seq_len, dtype=torch.long)
        mask[:L] = 1
        ids = ids * mask  # pad with 0 past L
        ids[0] = 101
        if self.learnable:
            # label encoded in the token content: the second token is drawn
            # from a SMALL repeated set (4 tokens per class) so its embedding
            # actually trains within an epoch — a bucket of the full vocab
            # would show each token <1x per epoch and never learn
            tok = 106 + int(torch.randint(0, self.num_labels * 4, (1,),
                                          generator=g))
            ids[1] = tok
            label = (tok - 106) % self.num_labels
        else:
            label = int(torch.randint(0, self.num_labels, (1,), generator=g))
        return {
            "input_ids": ids.long(),
            "attention_mask": mask,
            "token_type_ids": torch.zeros(self.seq_len, dtype=torch.long),
            "label": torch.tensor(label, dtype=torch.long),
        }
