#!/usr/bin/env python
"""Single-GPU (or CPU) BERT classification fine-tuning.

Reference-equivalent entrypoint (reference: single-gpu-cls.py — the minimal
slice of SURVEY.md §3.4): full pipeline on one device, no collectives.

    python single-gpu-cls.py [--max-seq-len 128 --train-batch-size 32 ...]
"""
from pdnlp_amd.cli import main

if __name__ == "__main__":
    main(strategy="single")
