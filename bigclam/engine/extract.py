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


def local_memberships(
    F_local: torch.Tensor, k_true: int, delta: float, use_hip: bool = False
):
    """Per-row memberships of a local F slice: ``(counts int64 [n],
    comms int32 [total])``, row-major, ascending community ids per row.

    ``use_hip``: run the K7 device kernels on ``F_local`` (padded rows ok —
    ``k_true`` bounds the columns); else a vectorized torch pass.
    """
    if use_hip:
        from ..ops import hip as hip_ops

        counts, comms = hip_ops.extract_membership(F_local, k_true, delta)
        return counts.long().cpu().numpy(), comms.cpu().numpy()
    Ff = F_local[:, :k_true].float().cpu()
    fmax = Ff.max(dim=1).values
    above = Ff >= delta
    fallback = (fmax < delta) & (fmax > 0)
    ties = Ff == fmax.unsqueeze(1)
    mem = torch.where(fallback.unsqueeze(1), ties, above)
    mem &= fmax.unsqueeze(1) > 0
    counts = mem.sum(dim=1).numpy().astype(np.int64)
    comms = torch.nonzero(mem, as_tuple=False)[:, 1].numpy().astype(np.int32)
    return counts, comms


def extract_communities_sharded(trainer, delta: float = None):
    """Sharded K7 extraction: each rank thresholds its own F rows on
    device (HIP kernel on GPU), D2H only the compacted (count, community)
    stream, gather the compacted streams to rank 0 — never materializes
    N×K anywhere (codes/Bigclamv2.scala:223-230).

    Returns on rank 0: ``(comms int32 [M], nodes int64 [M])`` membership
    pairs sorted by (community, node); ``None`` elsewhere.
    """
    g = trainer.graph
    st = trainer.state
    if delta is None:
        delta = membership_threshold(g.num_nodes, g.num_edges)
    counts, comms = local_memberships(
        st.F_local if st.use_hip else st.F_local_k,
        trainer.cfg.k,
        delta,
        use_hip=st.use_hip,
    )
    start = trainer.shard.start
    nodes = np.repeat(
        np.arange(start, trainer.shard.stop, dtype=np.int64), counts
    )
    if trainer.world_size > 1:
        import torch.distributed as dist

        parts = [None] * trainer.world_size if trainer.rank == 0 else None
        dist.gather_object((comms, nodes), parts, dst=0)
        if trainer.rank != 0:
            return None
        comms = np.concatenate([p[0] for p in parts])
        nodes = np.concatenate([p[1] for p in parts])
    order = np.lexsort((nodes, comms))
    return comms[order], nodes[order]


def write_membership_pairs(
    path: str, comms: np.ndarray, nodes: np.ndarray,
    raw_ids: Optional[np.ndarray] = None,
):
    """Single-writer output from sorted (community, node) pairs — same
    format as ``write_communities``: one line per non-empty community."""
    with open(path, "w") as f:
        i = 0
        m = len(comms)
        while i < m:
            c = comms[i]
            j = i
            while j < m and comms[j] == c:
                j += 1
            ids = nodes[i:j] if raw_ids is None else raw_ids[nodes[i:j]]
            f.write(f"{int(c)}: " + " ".join(str(int(x)) for x in ids) + "\n")
            i = j


def write_communities(
    path: str, members: List[np.ndarray], raw_ids: Optional[np.ndarray] = None
):
    with open(path, "w") as f:
        for c, nodes in enumerate(members):
            if len(nodes) == 0:
                continue
            ids = raw_ids[nodes] if raw_ids is not None else nodes
            f.write(f"{c}: " + " ".join(str(int(i)) for i in ids) + "\n")
