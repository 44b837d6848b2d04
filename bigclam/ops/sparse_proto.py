"""CPU prototype of the sparsity-adaptive sweep (docs/sparse_sweep_design.md).

Computes one full BigCLAM sweep decision per node — gradient, Armijo step
pick and projected commit — touching ONLY per-node active sets, never a
K-length array:

  S'_u = supp(gacc_u) ∪ supp(fu)       (gacc support ⊆ union of neighbor
                                        supports; off-S' the gradient is
                                        exactly -sumF)
  S_u  = supp(fu) ∪ {k ∈ S'_u: g_u[k] > 0}   (shared candidate support)
  gg_u = GG + Σ_{S'_u} (g² - sumF²),   GG = Σ_K sumF²  (once per sweep)

All clamps give exact zeros, so the sets are exact, not thresholded —
tests/test_sparse_proto.py asserts both the set-containment claims and
value equality against the dense torch reference (ops/reference.py).

This is the readable algorithmic contract; ops/sparse_cpu.py is the
vectorized full-sweep realization (same math, scipy CSR), and the next
round's GPU path follows the same shape (rows as index/value segments,
candidate scoring as MFMA tiles over the gathered fv[:, S_u] panel).
"""
from __future__ import annotations

from typing import Dict, List, Tuple

import numpy as np

from ..config import BigClamConfig


def row_support(F: np.ndarray) -> List[np.ndarray]:
    """Per-row sorted nonzero column indices."""
    return [np.flatnonzero(F[u]) for u in range(F.shape[0])]


def sparse_sweep_node(
    u: int,
    F: np.ndarray,            # [N, K] fp32 (dense storage; reads are set-limited)
    supp: List[np.ndarray],   # row_support(F)
    indptr: np.ndarray,
    indices: np.ndarray,
    sumF: np.ndarray,         # [K] fp32
    GG: float,                # double: (sumF.astype(f64) ** 2).sum()
    cfg: BigClamConfig,
) -> Tuple[np.ndarray, np.ndarray, float, float, np.ndarray, np.ndarray]:
    """One node's sweep work on active sets only.

    Returns (S_prime, g_on_S_prime, llh_u, best_step, S, new_row_on_S):
    the gradient support and values, the node's local LLH, the accepted
    Armijo step (0 = none), and the committed row confined to S.
    """
    nbrs = indices[indptr[u]:indptr[u + 1]]
    fu_idx = supp[u]
    fu_val = F[u, fu_idx].astype(np.float32)

    # ---- gradient accumulation on the union of neighbor supports
    acc: Dict[int, np.float32] = {}
    llh_edges = 0.0
    for v in nbrs:
        sv = supp[v]
        fv = F[v, sv]
        # x = fu . fv over supp(fu) ∩ supp(fv)
        common, ia, ib = np.intersect1d(
            fu_idx, sv, assume_unique=True, return_indices=True
        )
        x = float(np.dot(fu_val[ia], fv[ib])) if len(common) else 0.0
        p = np.clip(np.exp(np.float32(-x)), cfg.min_p, cfg.max_p)
        w = np.float32(1.0) / (np.float32(1.0) - p)
        llh_edges += float(np.log1p(-p.astype(np.float64))) + x
        for k, val in zip(sv, fv):
            acc[k] = acc.get(k, np.float32(0.0)) + w * val

    s_prime = np.union1d(np.array(sorted(acc), dtype=np.int64), fu_idx)
    gacc = np.array([acc.get(k, np.float32(0.0)) for k in s_prime],
                    dtype=np.float32)
    fu_on_sp = F[u, s_prime].astype(np.float32)
    g_sp = gacc - sumF[s_prime] + fu_on_sp

    # node part of llh_u: -fu.sumF + fu.fu over supp(fu)
    llh_u = (
        llh_edges
        + float(-(fu_val.astype(np.float64) @ sumF[fu_idx].astype(np.float64)))
        + float(fu_val.astype(np.float64) @ fu_val.astype(np.float64))
    )

    # gg = Σ_K g² via the shared dense part: off-S' g == -sumF exactly
    g64 = g_sp.astype(np.float64)
    s64 = sumF[s_prime].astype(np.float64)
    gg = GG + float((g64 * g64 - s64 * s64).sum())

    # ---- candidate support and the Armijo ladder (descending; first hit)
    S = np.union1d(fu_idx, s_prime[g_sp > 0])
    pos = np.searchsorted(s_prime, S)
    g_S = np.where(np.isin(S, s_prime), g_sp[np.clip(pos, 0, len(s_prime) - 1)],
                   np.float32(0.0)).astype(np.float32)
    fu_S = F[u, S].astype(np.float32)
    best = 0.0
    new_row = fu_S
    for j in range(cfg.ls_steps + 1):
        s_val = np.float32(cfg.beta ** j)
        cand = np.clip(fu_S + s_val * g_S, cfg.min_f, cfg.max_f)
        # trial edge term: dot(cand, fv) over S ∩ supp(fv)
        trial = 0.0
        for v in nbrs:
            sv = supp[v]
            common, ia, ib = np.intersect1d(
                S, sv, assume_unique=True, return_indices=True
            )
            x = float(np.dot(cand[ia], F[v, sv][ib])) if len(common) else 0.0
            p = np.clip(np.exp(np.float32(-x)), cfg.min_p, cfg.max_p)
            trial += float(np.log1p(-p.astype(np.float64))) + x
        # node term: cand.(fu - sumF) over S (the kernel identity)
        trial += float(
            cand.astype(np.float64)
            @ (fu_S.astype(np.float64) - sumF[S].astype(np.float64))
        )
        if trial >= llh_u + cfg.alpha * float(s_val) * gg:
            best = float(s_val)
            new_row = cand
            break
    return s_prime, g_sp, llh_u, best, S, new_row
