"""BigCLAM compute ops.

``reference``: vectorized torch implementations (CPU path + numerics
reference).  ``hip``: hand-written CDNA4 HIP kernels (GPU path; requires the
in-tree ``bigclam._C`` extension built for gfx950).
"""
from . import reference  # noqa: F401
