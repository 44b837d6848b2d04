"""Synthetic graph generators.

There is no network access and the reference's large datasets (com-Youtube,
com-Amazon) are missing blobs, so benchmarks and tests run on synthetic
graphs of the same shape (BASELINE.json: "synthetic graphs / random-init F").

- :func:`rmat`: R-MAT power-law graphs (Chakrabarti et al. 2004) — the
  100M-edge 8-GPU config and the com-Amazon-shaped bench graph.
- :func:`planted_partition`: graphs with known ground-truth communities for
  recovery tests (SURVEY.md §4 "planted-partition graphs").

R-MAT has no counterpart in the reference (SURVEY.md §2-B note).
"""
from __future__ import annotations

import numpy as np

from .edgelist import Graph, build_graph


def rmat_edges(
    scale: int,
    edge_factor: float,
    a: float = 0.57,
    b: float = 0.19,
    c: float = 0.19,
    seed: int = 0,
) -> np.ndarray:
    """Generate ~``edge_factor * 2**scale`` R-MAT edge pairs (with dups/loops).

    Vectorized bit-by-bit quadrant sampling; IDs are scrambled so hubs are not
    clustered at low indices.
    """
    n = 1 << scale
    m = int(edge_factor * n)
    if m >= 2_000_000:
        # counter-based parallel native generator (io_native.cpp):
        # deterministic per (seed, edge index), thread-count independent.
        # Different stream than the numpy path below — gated by size so
        # small test fixtures keep their numpy streams.
        try:
            from .. import _io_native

            return np.asarray(
                _io_native.rmat_edges(scale, edge_factor, a, b, c, seed)
            )
        except ImportError:
            pass
    rng = np.random.default_rng(seed)
    src = np.zeros(m, dtype=np.int64)
    dst = np.zeros(m, dtype=np.int64)
    ab = a + b
    a_norm = a / ab
    c_norm = c / (1.0 - ab)
    for _ in range(scale):
        r1 = rng.random(m, dtype=np.float32)
        r2 = rng.random(m, dtype=np.float32)
        src_bit = r1 > np.float32(ab)
        dst_bit = np.where(src_bit, r2 > np.float32(c_norm), r2 > np.float32(a_norm))
        src = (src << 1) | src_bit
        dst = (dst << 1) | dst_bit
    # scramble ids (fixed permutation derived from seed)
    perm = rng.permutation(n)
    return np.stack([perm[src], perm[dst]], axis=1)


def rmat_graph(scale: int, edge_factor: float = 16.0, seed: int = 0) -> Graph:
    return build_graph(rmat_edges(scale, edge_factor, seed=seed))


def rmat_graph_with_edges(
    target_nodes: int, target_edges: int, seed: int = 0
) -> Graph:
    """R-MAT graph shaped like (target_nodes, target_edges undirected).

    Used by bench.py to mimic named SNAP datasets (e.g. com-Amazon: 335k
    nodes / 926k edges) without the actual files.  Oversamples to compensate
    for dedup losses, then trims.
    """
    scale = max(1, int(np.ceil(np.log2(target_nodes))))
    # oversample: R-MAT dedup + self-loop losses are modest at low density
    factor = 1.3 * target_edges / (1 << scale)
    edges = rmat_edges(scale, factor, seed=seed)
    g = build_graph(edges)
    if g.num_edges > target_edges:
        # trim by dropping a random subset of undirected edges
        rng = np.random.default_rng(seed + 1)
        lo_mask = g.indices.astype(np.int64) > np.repeat(
            np.arange(g.num_nodes, dtype=np.int64), g.degrees()
        )
        src = np.repeat(np.arange(g.num_nodes, dtype=np.int64), g.degrees())
        und_src = src[lo_mask]
        und_dst = g.indices[lo_mask].astype(np.int64)
        keep = rng.choice(len(und_src), size=target_edges, replace=False)
        g = build_graph(np.stack([und_src[keep], und_dst[keep]], axis=1))
    return g


def shaped_graph(
    n: int,
    e: int,
    alpha: float = 0.4,
    locality: float = 0.7,
    seed: int = 0,
) -> Graph:
    """Power-law graph with EXACTLY n nodes and e undirected edges.

    Chung-Lu-style: node weights w_i ∝ (i+1)^-alpha (shuffled), a base pass
    gives every node one incident edge (no isolated nodes, so the node count
    is exact — plain R-MAT at low edge factor touches far fewer nodes than
    2**scale), then endpoint sampling fills in the remaining edges;
    oversampled and trimmed to exactly e after dedup.  alpha=0.4 at
    com-Amazon's density gives a ~350 max degree, close to the real 549.

    ``locality``: fraction of extra edges whose partner is drawn from a
    geometric window around the source instead of globally by weight.
    Real SNAP community graphs (com-Amazon is a product co-purchase
    network) have strong neighborhood locality under their natural node
    ordering; a locality-free Chung-Lu sample would make every
    contiguous-range partition cut ~(R-1)/R of all edges — unrepresentative
    for the halo-exchange benchmarks.
    """
    rng = np.random.default_rng(seed)
    w = (np.arange(1, n + 1, dtype=np.float64)) ** (-alpha)
    rng.shuffle(w)
    p = w / w.sum()

    def local_partner(src_ids):
        off = rng.geometric(1.0 / 64.0, size=len(src_ids)).astype(np.int64)
        sign = rng.integers(0, 2, size=len(src_ids)) * 2 - 1
        d = np.clip(src_ids + sign * off, 0, n - 1)
        bad = d == src_ids
        d[bad] = (src_ids[bad] + 1) % n
        return d

    # base: every node gets one (local) partner so node count is exact
    base_src = np.arange(n, dtype=np.int64)
    base_dst = local_partner(base_src)
    m_extra = int(1.35 * max(0, e - n))
    src = rng.choice(n, size=m_extra, p=p)
    loc = rng.random(m_extra) < locality
    dst = np.empty(m_extra, dtype=np.int64)
    dst[loc] = local_partner(src[loc])
    dst[~loc] = rng.choice(n, size=int((~loc).sum()), p=p)
    edges = np.concatenate(
        [
            np.stack([base_src, base_dst], axis=1),
            np.stack([src, dst], axis=1),
        ]
    )
    g = build_graph(edges)
    if g.num_nodes != n:
        raise RuntimeError("shaped_graph lost nodes unexpectedly")
    if g.num_edges < e:
        raise RuntimeError(
            f"shaped_graph undersampled: {g.num_edges} < {e}; raise oversample"
        )
    # trim extra undirected edges at random, never touching the base pass
    # (trim only edges whose both endpoints keep degree >= 2)
    srcs = np.repeat(np.arange(n, dtype=np.int64), g.degrees())
    lo_mask = srcs < g.indices
    und_src = srcs[lo_mask]
    und_dst = g.indices[lo_mask].astype(np.int64)
    n_trim = g.num_edges - e
    if n_trim > 0:
        deg = g.degrees().copy()
        order = rng.permutation(len(und_src))
        keep = np.ones(len(und_src), dtype=bool)
        removed = 0
        for idx in order:
            if removed == n_trim:
                break
            a, b = und_src[idx], und_dst[idx]
            if deg[a] > 1 and deg[b] > 1:
                keep[idx] = False
                deg[a] -= 1
                deg[b] -= 1
                removed += 1
        if removed < n_trim:
            raise RuntimeError("shaped_graph: could not trim to target")
        g = build_graph(
            np.stack([und_src[keep], und_dst[keep]], axis=1)
        )
        if g.num_nodes != n or g.num_edges != e:
            raise RuntimeError("shaped_graph trim broke the target shape")
    return g


def planted_partition(
    num_communities: int,
    nodes_per_community: int,
    p_in: float = 0.3,
    p_out: float = 0.01,
    seed: int = 0,
) -> tuple:
    """Planted-partition graph; returns (Graph, labels[N])."""
    rng = np.random.default_rng(seed)
    n = num_communities * nodes_per_community
    labels = np.repeat(np.arange(num_communities), nodes_per_community)
    # sample within-community edges
    edges = []
    for c in range(num_communities):
        base = c * nodes_per_community
        iu, ju = np.triu_indices(nodes_per_community, k=1)
        mask = rng.random(len(iu)) < p_in
        edges.append(np.stack([base + iu[mask], base + ju[mask]], axis=1))
    # sample cross edges sparsely
    m_out = int(p_out * n * n / 2)
    if m_out:
        s = rng.integers(0, n, m_out)
        d = rng.integers(0, n, m_out)
        mask = labels[s] != labels[d]
        edges.append(np.stack([s[mask], d[mask]], axis=1))
    all_edges = np.concatenate(edges, axis=0)
    return build_graph(all_edges), labels
