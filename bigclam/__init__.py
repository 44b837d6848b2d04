"""bigclam — an MI355X-native BigCLAM overlapping-community-detection engine.

A from-scratch GPU framework with the capabilities of the reference
Spark/Scala scripts (thangdnsf/BigCLAM-ApacheSpark): PyTorch-ROCm
orchestration, hand-written CDNA4 HIP kernels for the gradient/line-search
hot path, RCCL-over-xGMI halo exchange + collectives for multi-GPU row
sharding.  See SURVEY.md for the layer map and BASELINE.md for metrics.
"""
from .config import BigClamConfig, k_grid

__version__ = "0.1.0"

__all__ = [
    "BigClamConfig",
    "k_grid",
    "Trainer",
    "load_graph",
    "select_k",
    "extract_communities",
    "__version__",
]


def __getattr__(name):  # lazy: keep `import bigclam` light (no torch)
    if name == "Trainer":
        from .engine.trainer import Trainer

        return Trainer
    if name == "load_graph":
        from .io import load_graph

        return load_graph
    if name == "select_k":
        from .engine.model_select import select_k

        return select_k
    if name == "extract_communities":
        from .engine.extract import extract_communities

        return extract_communities
    raise AttributeError(name)
