#!/usr/bin/env python
"""Single-text inference demo across saved checkpoints (reference: predict.py).

Picks a sample (by label or random) from the dataset, runs batch-1 inference
with every checkpoint, prints true vs predicted label.

    python predict.py [--text "..."] [--label 3]
"""
import argparse
import glob
import os
import random

import torch

from pdnlp_amd.config import Args
from pdnlp_amd.data import LABELS, build_tokenizer, load_data
from pdnlp_amd.models import build_model
from pdnlp_amd.utils import set_seed, load_checkpoint


def tokenize_input(text: str, tokenizer, max_seq_len: int, device):
    ids, mask, type_ids = tokenizer.encode(text, max_seq_len)
    to = lambda x: torch.tensor([x], dtype=torch.long, device=device)  # noqa: E731
    return to(ids), to(mask), to(type_ids)


@torch.no_grad()
def predict(model, text: str, tokenizer, max_seq_len: int, device) -> int:
    model.eval()
    input_ids, mask, type_ids = tokenize_input(text, tokenizer, max_seq_len,
                                               device)
    out = model(input_ids=input_ids, attention_mask=mask,
                token_type_ids=type_ids)
    return int(out.logits.float().argmax(-1).item())


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--text", default=None)
    p.add_argument("--label", type=int, default=3,
                   help="pick a random dataset sample with this label")
    p.add_argument("--ckpt", action="append", default=None)
    p.add_argument("--engine", choices=["eager", "graphed"],
                   default="eager",
                   help="graphed = hipGraph-captured serving forward "
                        "(pdnlp_amd.engine.InferenceEngine)")
    ns, rest = p.parse_known_args()
    args = Args().apply_cli(rest)
    set_seed(args.seed)
    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    tokenizer = build_tokenizer(args.model_path)

    if ns.text is not None:
        text, true_label = ns.text, None
    elif os.path.isfile(args.data_path):
        data = load_data(args.data_path, limit=args.data_limit)
        cands = [d for d in data if d[1] == ns.label] or data
        text, true_label = random.choice(cands)
    else:
        text, true_label = "今天天气真好，非常开心", None
    print(f"text: {text}")
    if true_label is not None:
        print(f"true label: {LABELS[true_label]}")

    ckpts = ns.ckpt or sorted(glob.glob(os.path.join(args.output_dir, "*.pt")))
    if not ckpts:
        print(f"no checkpoints under {args.output_dir}; using random init")
        ckpts = [None]
    for c in ckpts:
        model = build_model(args.model, model_path=args.model_path)
        if c is not None:
            load_checkpoint(model, c)
        if ns.engine == "graphed":
            from pdnlp_amd.engine import InferenceEngine
            if torch.cuda.is_available():
                model = model.to(torch.bfloat16)
            eng = InferenceEngine(model, tokenizer, device=str(device),
                                  max_seq_len=args.max_seq_len)
            pred = eng.predict([text])[0]
            stats = eng.latency_bench(batch=1, seq=args.max_seq_len,
                                      iters=30, warmup=10)
            print(f"{c or '<random-init>'} → predicted: {LABELS[pred]} "
                  f"(p50 {stats['p50_ms']} ms, graphed={stats['graph']})")
        else:
            model = model.to(device)
            pred = predict(model, text, tokenizer, args.max_seq_len, device)
            print(f"{c or '<random-init>'} → predicted: {LABELS[pred]}")


if __name__ == "__main__":
    main()
