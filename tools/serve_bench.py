#!/usr/bin/env python
"""Serving latency/throughput across batch sizes (hipGraph engine)."""
import sys

import torch

sys.path.insert(0, ".")
from pdnlp_amd.engine import InferenceEngine  # noqa: E402
from pdnlp_amd.models import build_model  # noqa: E402
from pdnlp_amd.utils import set_seed  # noqa: E402

set_seed(5)
model = build_model("bert-base").to(torch.bfloat16)
eng = InferenceEngine(model, device="cuda", use_graph=True)
for b in (1, 8, 64):
    st = eng.latency_bench(batch=b, seq=128, iters=50, warmup=15)
    p50, p99 = st["p50_ms"], st["p99_ms"]
    print(f"serve b{b:<3d}: p50 {p50:6.3f} ms  p99 {p99:6.3f} ms  "
          f"-> {b / p50 * 1000:8.0f} seq/s")
