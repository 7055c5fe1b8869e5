"""Rank-0 console logging with the reference's de-facto UX contract
(reference: multi-gpu-distributed-cls.py:178-181 prints
``【train】 epoch：e/E step：s/S loss：x`` gated on local_rank == 0)."""

import logging
import sys

import torch.distributed as dist


def _rank() -> int:
    if dist.is_available() and dist.is_initialized():
        return dist.get_rank()
    return 0


def rank0_print(*args, **kwargs) -> None:
    if _rank() == 0:
        print(*args, **kwargs)
        sys.stdout.flush()


def get_logger(name: str = "pdnlp") -> logging.Logger:
    logger = logging.getLogger(name)
    if not logger.handlers:
        h = logging.StreamHandler(sys.stderr)
        h.setFormatter(logging.Formatter(
            "[%(asctime)s][%(name)s][rank{}] %(message)s".format(_rank())))
        logger.addHandler(h)
        logger.setLevel(logging.INFO)
    return logger
