"""Model initialization: conductance seed ranking + neighborhood-indicator F.

Implements SURVEY.md §2.3-§2.5 (reference codes/bigclamv3-7.scala:35-87).

Conductance of node u's closed ego-net y = {u} ∪ N(u):
  z      = multiset of neighbors of all members of y  (|z| = Σ_{m∈y} deg(m))
  cut    = number of entries of z outside y
  vol_S  = |z| - cut
  vol_T  = Σdeg - vol_S - 2·cut
  cond   = cut / min(vol_S, vol_T), with guards vol_S==0 → 0, vol_T==0 → 1.

Ranking: the *intended* Yang-Leskovec semantics (default) pick, for every
node x, the minimum-conductance member of its closed ego-net (ties → lowest
id), dedupe, and sort ascending by conductance.  ``compat=True`` reproduces
the reference's latent bug (codes/bigclamv3-7.scala:51): Scala tuple ``min``
orders by node id first, so it actually selects each node's lowest-id
*neighbor* regardless of conductance.

Host-side NumPy: this is a one-time cost per fit (the reference also ranks
once, codes/bigclam4-7.scala:75); a HIP version is only worth it for graphs
whose Σdeg² makes the 2-hop pass heavy.
"""
from __future__ import annotations

from typing import Optional

import numpy as np

from ..io.edgelist import Graph


def conductances(graph: Graph) -> np.ndarray:
    """Ego-net conductance per node (fp64 [N])."""
    indptr, indices = graph.indptr, graph.indices
    n = graph.num_nodes
    deg = np.diff(indptr).astype(np.int64)
    total_degree = int(deg.sum())
    cond = np.zeros(n, dtype=np.float64)
    for u in range(n):
        nbrs = indices[indptr[u] : indptr[u + 1]]
        ego = np.append(nbrs.astype(np.int64), u)
        ego_sorted = np.sort(ego)
        z_size = int(deg[ego].sum())
        # count z entries inside ego
        inside = 0
        for m in ego:
            nm = indices[indptr[m] : indptr[m + 1]]
            pos = np.searchsorted(ego_sorted, nm)
            pos[pos >= len(ego_sorted)] = len(ego_sorted) - 1
            inside += int((ego_sorted[pos] == nm).sum())
        cut = z_size - inside
        vol_s = z_size - cut
        vol_t = total_degree - vol_s - 2 * cut
        if vol_s == 0:
            cond[u] = 0.0
        elif vol_t == 0:
            cond[u] = 1.0
        else:
            cond[u] = cut / min(vol_s, vol_t)
    return cond


def conductance_ranking(
    graph: Graph, compat: bool = False, cond: Optional[np.ndarray] = None
) -> np.ndarray:
    """Seed candidates ranked ascending by conductance (int64 node ids)."""
    if cond is None:
        cond = conductances(graph)
    indptr, indices = graph.indptr, graph.indices
    n = graph.num_nodes
    picked = {}
    for x in range(n):
        nbrs = indices[indptr[x] : indptr[x + 1]].astype(np.int64)
        if compat:
            # reference behavior: lowest-id neighbor; isolated -> (x, 10.0)
            if len(nbrs) == 0:
                m, c = x, 10.0
            else:
                m = int(nbrs.min())
                c = float(cond[m])
        else:
            members = np.append(nbrs, x)
            cvals = cond[members]
            # argmin by (conductance, id)
            best = np.lexsort((members, cvals))[0]
            m, c = int(members[best]), float(cvals[best])
        if m not in picked or c < picked[m]:
            picked[m] = c
    items = sorted(picked.items(), key=lambda kv: (kv[1], kv[0]))
    return np.array([m for m, _ in items], dtype=np.int64)


def seed_init_local_F(
    graph: Graph,
    k: int,
    start: int,
    stop: int,
    seeds: Optional[np.ndarray] = None,
    include_seed: bool = False,
    rng_seed: int = 0,
    compat: bool = False,
) -> np.ndarray:
    """Build the [stop-start, k] fp32 slice of the seed-initialized F.

    Community c's initial members = neighborhood of the c-th ranked seed
    (codes/bigclamv3-7.scala:64-65; ``include_seed`` adds the seed itself —
    the v2 variant, codes/Bigclamv2.scala:70).  Columns past the seed count
    are Bernoulli(0.5) pad rows (codes/bigclamv3-7.scala:56-58, 69-81) —
    generated over the full node range with a fixed seed so every rank's
    slice is consistent.
    """
    if seeds is None:
        seeds = conductance_ranking(graph, compat=compat)
    seeds = seeds[:k]
    n_local = stop - start
    F = np.zeros((n_local, k), dtype=np.float32)
    for c, s in enumerate(seeds):
        nbrs = graph.indices[graph.indptr[s] : graph.indptr[s + 1]].astype(np.int64)
        sel = nbrs[(nbrs >= start) & (nbrs < stop)] - start
        F[sel, c] = 1.0
        if include_seed and start <= s < stop:
            F[s - start, c] = 1.0
    n_pad = k - len(seeds)
    if n_pad > 0:
        rng = np.random.default_rng(rng_seed)
        pad = rng.integers(
            0, 2, size=(graph.num_nodes, n_pad), dtype=np.int8
        ).astype(np.float32)
        F[:, len(seeds) :] = pad[start:stop]
    return F


def random_init_local_F(
    n_total: int,
    k: int,
    start: int,
    stop: int,
    rng_seed: int = 0,
    scale: Optional[float] = None,
) -> np.ndarray:
    """Uniform(0, scale) random init (the BASELINE 'random-init F' configs).

    ``scale`` defaults to 1/sqrt(K) so initial edge dots x = Fu.Fv are O(1)
    regardless of K (keeps exp(-x) off its clamps at t=0).  Rows are
    generated globally deterministic (seeded by row-block) so a sharded run
    initializes identically to a single-shard run.
    """
    if scale is None:
        scale = 1.0 / float(np.sqrt(k))
    out = np.empty((stop - start, k), dtype=np.float32)
    block = 65536
    b0 = start // block
    b1 = (stop - 1) // block if stop > start else b0
    for b in range(b0, b1 + 1):
        rng = np.random.default_rng((rng_seed << 20) + b)
        rows = rng.random((min(block, n_total - b * block), k), dtype=np.float32)
        lo = max(start, b * block)
        hi = min(stop, b * block + rows.shape[0])
        out[lo - start : hi - start] = rows[lo - b * block : hi - b * block]
    out *= np.float32(scale)
    return out
