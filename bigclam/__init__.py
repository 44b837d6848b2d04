"""bigclam — an MI355X-native BigCLAM overlapping-community-detection engine.

A from-scratch GPU framework with the capabilities of the reference
Spark/Scala scripts (thangdnsf/BigCLAM-ApacheSpark): PyTorch-ROCm
orchestration, hand-written CDNA4 HIP kernels for the gradient/line-search
hot path, RCCL-over-xGMI halo exchange + collectives for multi-GPU row
sharding.  See SURVEY.md for the layer map and BASELINE.md for metrics.
"""
from .config import BigClamConfig, k_grid

__version__ = "0.1.0"

__all__ = ["BigClamConfig", "k_grid", "__version__"]
