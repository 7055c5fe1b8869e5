"""Debug / failure-handling utilities (SURVEY.md §5.2-5.3 — absent in the
reference; minimum-viable here).

- ``enable_debug_sync()``: serialized kernel launches (AMD_SERIALIZE_KERNEL)
  + blocking HIP launches, for race hunting. Set BEFORE torch/HIP init.
- ``watchdog()``: wraps a training run; on an exception it aborts the
  process group (so peer ranks' RCCL collectives fail fast instead of
  hanging) and exits nonzero.
"""

from __future__ import annotations

import contextlib
import os
import sys
import traceback

import torch.distributed as dist


def enable_debug_sync() -> None:
    os.environ["AMD_SERIALIZE_KERNEL"] = "3"
    os.environ["AMD_SERIALIZE_COPY"] = "3"
    os.environ["HIP_LAUNCH_BLOCKING"] = "1"


@contextlib.contextmanager
def watchdog(exit_on_error: bool = True):
    try:
        yield
    except Exception:
        traceback.print_exc()
        if dist.is_available() and dist.is_initialized():
            try:
                # abort collectives so other ranks don't hang on a dead peer
                pg = dist.distributed_c10d._get_default_group()
                if hasattr(pg, "_shutdown"):
                    pg._shutdown()
                dist.destroy_process_group()
            except Exception:
                pass
        if exit_on_error:
            sys.exit(13)
        raise
