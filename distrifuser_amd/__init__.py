"""distrifuser_amd — MI355X-native displaced-patch-parallel diffusion inference.

A from-scratch framework with the capabilities of mit-han-lab/distrifuser
(see SURVEY.md), designed for AMD Instinct MI355X (gfx950 / CDNA4):
PyTorch-ROCm host, hand-written HIP/MFMA kernels for the hot ops, and
RCCL collectives over xGMI for the displaced-patch stale-activation
exchange.

Public API (parity with the reference's surface,
/root/reference/distrifuser/__init__.py):
    DistriConfig, DistriSDXLPipeline, DistriSDPipeline
"""

from .__version__ import __version__
from .utils.config import DistriConfig
from .utils.comm import PatchParallelismCommManager
from .pipelines import DistriSDXLPipeline, DistriSDPipeline

__all__ = [
    "__version__",
    "DistriConfig",
    "PatchParallelismCommManager",
    "DistriSDXLPipeline",
    "DistriSDPipeline",
]
