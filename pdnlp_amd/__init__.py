"""pdnlp_amd — MI355X-native single-node distributed NLP fine-tuning framework.

A from-scratch rebuild of the capabilities of ``taishan1994/pytorch-distributed-NLP``
(reference layer map in SURVEY.md) designed CDNA4-first:

- compute: PyTorch-ROCm + hand-written HIP/gfx950 kernels (``pdnlp_amd.ops``)
- communication: RCCL over xGMI via ``torch.distributed`` (``pdnlp_amd.parallel``)
- engine: one Trainer with strategy modes instead of N copied scripts
  (reference duplicates the pipeline per strategy)
"""

__version__ = "0.1.0"

from .config import Args, BertConfig  # noqa: F401
