#!/usr/bin/env python
"""Consolidate a ZeRO sharded checkpoint directory into a single fp32 model
state dict (the deepspeed ``zero_to_fp32.py`` capability, reference:
README.md:484-488).

    python tools/zero_to_fp32.py <ckpt_dir> <out.pt>
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from pdnlp_amd.parallel.zero import consolidate_zero_checkpoint  # noqa: E402


def main():
    if len(sys.argv) != 3:
        print(__doc__)
        sys.exit(1)
    ckpt_dir, out = sys.argv[1], sys.argv[2]
    sd = consolidate_zero_checkpoint(ckpt_dir, out)
    print(f"consolidated {len(sd)} tensors -> {out}")


if __name__ == "__main__":
    main()
