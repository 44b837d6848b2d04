"""Community extraction & output (K7; codes/Bigclamv2.scala:223-230).

Threshold: eps = 2E/(N(N-1)) (background edge density), delta =
sqrt(-log(1-eps)); node u belongs to community c iff F[u,c] >= delta; a node
whose max affiliation is below delta joins its argmax communities (ties
included, matching the reference).  Deviation (documented): all-zero rows
get NO membership — the reference's sparse-round-trip would put them in
every community.

Output format (ours to define — the reference's is Spark's
``(c,CompactBuffer(...))`` toString): one line per non-empty community,
``<community-id>: <raw node ids space-separated>``.
"""
from __future__ import annotations

from typing import List, Optional

import numpy as np
import torch


def membership_threshold(num_nodes: int, num_edges: int) -> float:
    eps = 2.0 * num_edges / (num_nodes * (num_nodes - 1))
    return float(np.sqrt(-np.log1p(-eps)))


def extract_communities(
    F: torch.Tensor, num_edges: int, delta: Optional[float] = None
) -> List[np.ndarray]:
    """Returns per-community member lists (internal ids), length K."""
    n, k = F.shape
    if delta is None:
        delta = membership_threshold(n, num_edges)
    Ff = F.float()
    fmax = Ff.max(dim=1).values
    above = Ff >= delta
    # fallback rows: max below delta but nonzero -> argmax columns (ties)
    fallback = (fmax < delta) & (fmax > 0)
    ties = Ff == fmax.unsqueeze(1)
    mem = torch.where(fallback.unsqueeze(1), ties, above)
    mem &= fmax.unsqueeze(1) > 0  # all-zero rows -> no membership
    out = []
    for c in range(k):
        out.append(torch.nonzero(mem[:, c], as_tuple=False).flatten().numpy())
    return out


def write_communities(
    path: str, members: List[np.ndarray], raw_ids: Optional[np.ndarray] = None
):
    with open(path, "w") as f:
        for c, nodes in enumerate(members):
            if len(nodes) == 0:
                continue
            ids = raw_ids[nodes] if raw_ids is not None else nodes
            f.write(f"{c}: " + " ".join(str(int(i)) for i in ids) + "\n")
