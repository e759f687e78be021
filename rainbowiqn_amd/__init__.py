"""MI355X-native Rainbow-IQN Ape-X framework.

A from-scratch rebuild of valeoai/rainbow-iqn-apex for AMD Instinct MI355X
(gfx950 / CDNA4): PyTorch-ROCm framework layer, hand-written HIP kernels for
the hot ops (rainbowiqn_amd.ops), on-GPU prioritized sum-tree replay, and the
Ape-X actor/replay/learner split over RCCL/xGMI instead of Redis
(rainbowiqn_amd.parallel).  See SURVEY.md for the reference blueprint.
"""
from .config import Config, parse_args  # noqa: F401

__version__ = "0.1.0"
