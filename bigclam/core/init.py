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
    """Ego-net conductance per node (fp64 [N]).

    Vectorized closed form (for the simple undirected canonical graph):
      z_size(u)  = deg(u) + sum_{v in N(u)} deg(v)
      inside(u)  = 2*deg(u) + sum_{v in N(u)} tri(u,v)    (tri = common nbrs)
      cut        = z_size - inside;  vol_S = inside
      vol_T      = total_degree - vol_S - 2*cut
    with tri row-sums from (A@A) ∘ A (sparse).  Equivalent to the brute-force
    2-hop pass (tests/test_init.py checks it against tests/oracle.py).
    """
    import scipy.sparse as sp

    n = graph.num_nodes
    deg = np.diff(graph.indptr).astype(np.int64)
    total_degree = int(deg.sum())
    A = sp.csr_matrix(
        (
            np.ones(len(graph.indices), dtype=np.float64),
            graph.indices.astype(np.int64),
            graph.indptr,
        ),
        shape=(n, n),
    )
    z_size = deg + (A @ deg.astype(np.float64)).astype(np.int64)
    tri_row = np.asarray(((A @ A).multiply(A)).sum(axis=1)).ravel()
    inside = 2 * deg + tri_row.astype(np.int64)
    cut = z_size - inside
    vol_s = inside
    vol_t = total_degree - vol_s - 2 * cut
    cond = np.where(
        vol_s == 0,
        0.0,
        np.where(
            vol_t == 0,
            1.0,
            cut / np.maximum(np.minimum(vol_s, vol_t), 1).astype(np.float64),
        ),
    )
    return cond


def conductance_ranking(
    graph: Graph, compat: bool = False, cond: Optional[np.ndarray] = None
) -> np.ndarray:
    """Seed candidates ranked ascending by conductance (int64 node ids)."""
    if cond is None:
        cond = conductances(graph)
    indptr, indices = graph.indptr, graph.indices
    n = graph.num_nodes
    deg = np.diff(indptr)
    dst = indices.astype(np.int64)
    has_nbrs = deg > 0
    if compat:
        # reference behavior (codes/bigclamv3-7.scala:51): Scala tuple min
        # orders by node id first -> each node picks its lowest-id neighbor.
        # (The canonical graph has no isolated nodes — build_graph drops
        # them — so the reference's (x, 10.0) sentinel branch cannot fire.)
        picked = np.arange(n, dtype=np.int64)
        starts = indptr[:-1].clip(max=max(len(dst) - 1, 0))
        mins = np.minimum.reduceat(dst, starts) if len(dst) else picked
        picked[has_nbrs] = mins[has_nbrs]
    else:
        # intended semantics: min-conductance member of the closed ego-net
        # (ties -> lowest id).  Per row, the neighbor minimizing (cond, id)
        # falls first after a lexsort keyed (src-major, cond, id).
        src = np.repeat(np.arange(n, dtype=np.int64), deg)
        o = np.lexsort((dst, cond[dst], src))
        best_nbr = np.full(n, -1, dtype=np.int64)
        best_nbr[has_nbrs] = dst[o][indptr[:-1][has_nbrs]]
        picked = np.arange(n, dtype=np.int64)
        take_nbr = has_nbrs & (
            (cond[best_nbr.clip(min=0)] < cond[picked])
            | (
                (cond[best_nbr.clip(min=0)] == cond[picked])
                & (best_nbr < picked)
            )
        )
        picked[take_nbr] = best_nbr[take_nbr]
    cands = np.unique(picked)
    order = np.lexsort((cands, cond[cands]))
    return cands[order]


def conductance_ranking_device(graph: Graph, cond_t) -> np.ndarray:
    """Seed ranking on the DEVICE (same result as ``conductance_ranking``
    intended semantics, tested): total-order ranks consistent with
    (cond, id), then a segment-min over each closed ego-net.

    The host version's lexsort over the edge list is ~0.4 s of the
    headline fit wall; this is a handful of tensor ops on the K5 output.
    """
    import torch

    dev = cond_t.device
    n = graph.num_nodes
    # rank[node]: position in the (cond, id) total order
    order = torch.argsort(cond_t, stable=True)  # ties keep ascending id
    rank = torch.empty(n, device=dev, dtype=torch.int64)
    rank[order] = torch.arange(n, device=dev, dtype=torch.int64)
    deg = torch.from_numpy(
        (graph.indptr[1:] - graph.indptr[:-1]).astype(np.int64)
    ).to(dev)
    src = torch.repeat_interleave(
        torch.arange(n, device=dev, dtype=torch.int64), deg
    )
    dst = torch.from_numpy(graph.indices.astype(np.int64)).to(dev)
    best = rank.clone()  # closed ego-net: self included
    best.scatter_reduce_(0, src, rank[dst], reduce="amin")
    picked = order[best]  # unrank -> node id of the ego-net minimizer
    cands = torch.unique(picked)
    cands = cands[torch.argsort(rank[cands], stable=True)]
    return cands.cpu().numpy()


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
    if len(seeds):
        # vectorized scatter: one fancy assignment instead of a per-seed
        # python loop (the loop was ~80% of the fit wall at K=5000)
        seeds = np.asarray(seeds, dtype=np.int64)
        deg = (graph.indptr[seeds + 1] - graph.indptr[seeds]).astype(np.int64)
        cols = np.repeat(np.arange(len(seeds), dtype=np.int64), deg)
        # gather each seed's neighbor slice: ranges via offset arithmetic
        starts = graph.indptr[seeds]
        offs = np.arange(int(deg.sum()), dtype=np.int64) - np.repeat(
            np.cumsum(deg) - deg, deg
        )
        rows = graph.indices[np.repeat(starts, deg) + offs].astype(np.int64)
        m = (rows >= start) & (rows < stop)
        F[rows[m] - start, cols[m]] = 1.0
        if include_seed:
            sm = (seeds >= start) & (seeds < stop)
            F[seeds[sm] - start, np.flatnonzero(sm)] = 1.0
    n_pad = k - len(seeds)
    if n_pad > 0:
        # Bernoulli(0.5) pad columns, generated per global row-block with a
        # block-keyed RNG so every rank's slice is identical regardless of
        # sharding — and only the local rows are ever materialized (the
        # full-[N, n_pad] array this replaces is ~150 GB/rank at the
        # N=15M, K=10000 config).
        block = 8192
        b0 = start // block
        b1 = (stop - 1) // block if stop > start else b0
        for b in range(b0, b1 + 1):
            rng = np.random.default_rng(((rng_seed + 1) << 20) + b)
            rows = rng.integers(
                0,
                2,
                size=(min(block, graph.num_nodes - b * block), n_pad),
                dtype=np.int8,
            )
            lo = max(start, b * block)
            hi = min(stop, b * block + rows.shape[0])
            F[lo - start : hi - start, len(seeds) :] = rows[
                lo - b * block : hi - b * block
            ]
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
