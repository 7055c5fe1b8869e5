#!/usr/bin/env python
"""Run bench.py across the BASELINE configs and print a measured table."""
import json
import subprocess
import sys

CONFIGS = [
    ["--model", "bert-base", "--batch-size", "32", "--seq-len", "128"],
    ["--model", "bert-base", "--batch-size", "64", "--seq-len", "128"],
    ["--model", "bert-base", "--batch-size", "128", "--seq-len", "128"],
    ["--model", "bert-base", "--batch-size", "32", "--seq-len", "128",
     "--dtype", "fp16"],
    ["--model", "bert-large", "--batch-size", "16", "--seq-len", "512"],
    ["--model", "roberta-base", "--batch-size", "64", "--seq-len", "128"],
]

for cfg in CONFIGS:
    out = subprocess.run(
        [sys.executable, "bench.py", *cfg, "--steps", "20", "--warmup", "5"],
        capture_output=True, text=True)
    line = [l for l in out.stdout.splitlines() if l.startswith("{")]
    if not line:
        print(" ".join(cfg), "FAILED", out.stderr[-200:])
        continue
    d = json.loads(line[0])
    c = d["config"]
    print(f"{c['model']:13s} bs{c['global_batch']:<4d} seq{c['seq_len']:<4d} "
          f"{d['dtype']}: {d['value']:8.1f} samples/s  "
          f"{d['ms_per_step']:7.2f} ms/step")
