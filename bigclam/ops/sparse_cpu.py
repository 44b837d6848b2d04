"""Vectorized sparse CPU sweep (scipy.sparse CSR F) — the full-sweep
realization of docs/sparse_sweep_design.md.

One call = one complete algorithmic sweep (K1 fused grad+LLH, K2
16-candidate Armijo ladder, K3 projected commit) with every per-node
quantity computed on the EXACT active pattern S'_u = supp(gacc_u) ∪
supp(fu) — candidates are identically zero outside it (the design doc's
containment lemma), and the dense ``-sumF`` part of ``gg`` folds into
one global ``GG = Σ_K sumF²`` per sweep.

Semantics mirror ops/reference.py bitwise-in-structure (same clamps,
descending first-accept ladder, Jacobi node-term identity
``cand·(fu − sumF)``); equality is asserted to fp tolerance in
tests/test_sparse_cpu.py, with step picks allowed to differ only below
the β⁵ noise rung (summation-order noise, same as the GPU tests).

Alignment: fu is re-expressed on the union pattern by a global
sorted-key scatter (exact — scipy's ``A + 0-pattern`` trick prunes
explicit zeros and cannot be relied on), so all per-entry math runs on
aligned ``.data`` arrays.
"""
from __future__ import annotations

from typing import Tuple

import numpy as np
import scipy.sparse as sp

from ..config import BigClamConfig


def _rowsum(data: np.ndarray, indptr: np.ndarray) -> np.ndarray:
    """Per-row sums of CSR data (float64)."""
    out = np.add.reduceat(
        np.concatenate([data.astype(np.float64), [0.0]]),
        indptr[:-1],
    )
    out[np.diff(indptr) == 0] = 0.0
    return out


def sparse_sweep(
    Fs: sp.csr_matrix,
    indptr: np.ndarray,
    indices: np.ndarray,
    sumF: np.ndarray,
    cfg: BigClamConfig,
) -> Tuple[sp.csr_matrix, np.ndarray, np.ndarray]:
    """Returns (F_new, best_step [N], llh_nodes [N] float64)."""
    N, K = Fs.shape
    Fs = Fs.tocsr().astype(np.float32)
    Fs.sort_indices()
    src = np.repeat(np.arange(N, dtype=np.int64), np.diff(indptr))
    dst = indices.astype(np.int64)

    # ---- K1: edge dots, weights, gradient accumulation
    Fsrc = Fs[src]
    Fdst = Fs[dst]
    x = np.asarray(Fsrc.multiply(Fdst).sum(axis=1)).ravel().astype(np.float32)
    p = np.clip(np.exp(-x), np.float32(cfg.min_p), np.float32(cfg.max_p))
    w = np.float32(1.0) / (np.float32(1.0) - p)
    llh_edges = np.bincount(
        src, weights=np.log1p(-p.astype(np.float64)) + x.astype(np.float64),
        minlength=N,
    )
    W = sp.csr_matrix((w, (src, dst)), shape=(N, N))
    gacc = (W @ Fs).tocsr()
    gacc.sort_indices()

    # ---- llh_u = edge part - fu.sumF + fu.fu
    sumF32 = sumF.astype(np.float32)
    fu_dot_sumF = Fs @ sumF32.astype(np.float64)
    fu_sq = _rowsum(Fs.data * Fs.data, Fs.indptr)
    llh_u = llh_edges - fu_dot_sumF + fu_sq

    # ---- active pattern S' = supp(gacc) ∪ supp(fu); aligned g, fu.
    # scipy prunes explicit zeros on addition, so fu is re-expressed on
    # U's pattern by a global sorted-key scatter (rows ascending and
    # within-row indices sorted make row*K+col keys globally sorted).
    U = (gacc + Fs).tocsr()
    U.sort_indices()
    U_rows = np.repeat(np.arange(N, dtype=np.int64), np.diff(U.indptr))
    U_keys = U_rows * K + U.indices
    F_rows = np.repeat(np.arange(N, dtype=np.int64), np.diff(Fs.indptr))
    F_keys = F_rows * K + Fs.indices
    pos = np.searchsorted(U_keys, F_keys)
    assert np.array_equal(U_keys[pos], F_keys)  # supp(F) ⊆ U pattern
    fu_al = np.zeros_like(U.data)
    fu_al[pos] = Fs.data
    g_data = (U.data - sumF32[U.indices]).astype(np.float32)

    # ---- gg = Σ_K g² = GG + Σ_{S'} (g² - sumF²)   (float64, like the
    # dense reference's grad.double() reduction)
    GG = float((sumF32.astype(np.float64) ** 2).sum())
    g64 = g_data.astype(np.float64)
    s64 = sumF32[U.indices].astype(np.float64)
    gg = GG + _rowsum(g64 * g64 - s64 * s64, U.indptr)

    # ---- 16-candidate ladder: trial LLH per rung on the S' pattern
    ladder = [np.float32(cfg.beta ** j) for j in range(cfg.ls_steps + 1)]
    n_l = len(ladder)
    trials = np.empty((n_l, N), dtype=np.float64)
    cands = []
    fu_minus_sumF = fu_al.astype(np.float64) - s64
    for j, s_val in enumerate(ladder):
        C = U.copy()
        C.data = np.clip(
            fu_al + s_val * g_data, np.float32(cfg.min_f),
            np.float32(cfg.max_f),
        )
        cands.append(C)
        xj = np.asarray(C[src].multiply(Fdst).sum(axis=1)).ravel().astype(
            np.float32
        )
        pj = np.clip(np.exp(-xj), np.float32(cfg.min_p), np.float32(cfg.max_p))
        edge_j = np.bincount(
            src,
            weights=np.log1p(-pj.astype(np.float64)) + xj.astype(np.float64),
            minlength=N,
        )
        node_j = _rowsum(C.data.astype(np.float64) * fu_minus_sumF, C.indptr)
        trials[j] = edge_j + node_j

    thresh = llh_u[None, :] + cfg.alpha * np.array(
        ladder, dtype=np.float64
    )[:, None] * gg[None, :]
    accepts = trials >= thresh
    first = accepts.argmax(axis=0)  # first True (descending ladder) or 0
    any_acc = accepts.any(axis=0)
    best = np.where(
        any_acc, np.array(ladder, dtype=np.float32)[first], np.float32(0.0)
    )

    # ---- commit: accepted rows take their candidate, others keep fu
    F_new = sp.csr_matrix(Fs.shape, dtype=np.float32)
    keep = (~any_acc).astype(np.float32)
    F_new = Fs.multiply(keep[:, None]).tocsr()
    for j in range(n_l):
        mask = (any_acc & (first == j)).astype(np.float32)
        if mask.any():
            F_new = (F_new + cands[j].multiply(mask[:, None])).tocsr()
    F_new.eliminate_zeros()  # clamped-to-zero entries leave the pattern
    F_new.sort_indices()
    return F_new, best, llh_u
