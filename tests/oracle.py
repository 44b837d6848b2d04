"""Loop-based fp64 NumPy oracle for BigCLAM — the correctness anchor.

Implements exactly the reference math (SURVEY.md §2.6-§2.10, §2.14;
reference: codes/bigclamv3-7.scala:89-204, codes/Bigclamv2.scala:223-230):
per-node gradient with the 1/(1-exp(-x)) weighting and MIN_P/MAX_P clamps,
per-node local LLH with the sumF trick, the 16-candidate Armijo ladder
evaluated Jacobi-style against a stale F snapshot, the projected commit,
and delta-threshold community extraction.

Deliberately slow and simple (python loops over nodes); used only in tests
on tiny graphs.
"""
from __future__ import annotations

import numpy as np

MIN_P = 1e-4
MAX_P = 0.9999
MIN_F = 0.0
MAX_F = 1000.0
ALPHA = 0.05
BETA = 0.1
LS_STEPS = 15


def clamp_p(x):
    return np.clip(np.exp(-x), MIN_P, MAX_P)


def project(row):
    return np.clip(row, MIN_F, MAX_F)


def node_llh(F, sumF, indptr, indices, u, Fu=None, sumF_u=None):
    """Local log-likelihood of node u (reference codes/bigclamv3-7.scala:137-150).

    llh_u = sum_{v in N(u)} [log(1 - clamp(exp(-Fu.Fv))) + Fu.Fv]
            - Fu.sumF' + Fu.Fu
    where Fu/sumF' may be overridden (line-search trial evaluation).
    """
    fu = F[u] if Fu is None else Fu
    sf = sumF if sumF_u is None else sumF_u
    acc = 0.0
    for v in indices[indptr[u] : indptr[u + 1]]:
        x = float(fu @ F[v])
        p = float(np.clip(np.exp(-x), MIN_P, MAX_P))
        acc += np.log(1.0 - p) + x
    return acc - float(fu @ sf) + float(fu @ fu)


def node_grad_llh(F, sumF, indptr, indices, u):
    """Gradient and local LLH of node u (codes/bigclamv3-7.scala:138-150)."""
    fu = F[u]
    K = F.shape[1]
    grad_acc = np.zeros(K, dtype=np.float64)
    llh_acc = 0.0
    for v in indices[indptr[u] : indptr[u + 1]]:
        fv = F[v]
        x = float(fu @ fv)
        p = float(np.clip(np.exp(-x), MIN_P, MAX_P))
        llh_acc += np.log(1.0 - p) + x
        grad_acc += fv / (1.0 - p)
    grad = grad_acc - sumF + fu
    llh = llh_acc - float(fu @ sumF) + float(fu @ fu)
    return grad, llh


def full_llh(F, sumF, indptr, indices):
    """Total objective: sum of node_llh over all nodes (edges counted twice)."""
    return sum(
        node_llh(F, sumF, indptr, indices, u) for u in range(len(indptr) - 1)
    )


def line_search(F, sumF, indptr, indices, u, grad, llh):
    """Best accepted Armijo step for node u (codes/bigclamv3-7.scala:153-163).

    All candidates evaluated against the same stale F; accept iff
    llh(F_u') >= llh + alpha * s * (grad . grad); keep the max accepted s.
    Returns 0.0 if no candidate is accepted.
    """
    gg = float(grad @ grad)
    best = 0.0
    for i in range(LS_STEPS + 1):
        s = BETA ** i
        fu_new = project(F[u] + s * grad)
        sf_new = sumF - F[u] + fu_new
        trial = node_llh(F, sumF, indptr, indices, u, Fu=fu_new, sumF_u=sf_new)
        if trial >= llh + ALPHA * s * gg:
            best = max(best, s)
            break  # ladder is descending; the first accept is the max
    return best


def sweep(F, sumF, indptr, indices):
    """One synchronous gradient/line-search sweep over all nodes.

    Jacobi semantics: every node's gradient and trial evaluations read the
    same stale snapshot; updates land simultaneously afterwards
    (SURVEY.md §2.8).  Returns (F_new, sumF_new, llh_after, steps[N]).
    """
    n = len(indptr) - 1
    F_new = F.copy()
    steps = np.zeros(n)
    for u in range(n):
        grad, llh = node_grad_llh(F, sumF, indptr, indices, u)
        s = line_search(F, sumF, indptr, indices, u, grad, llh)
        steps[u] = s
        if s > 0.0:
            F_new[u] = project(F[u] + s * grad)
    sumF_new = sumF + (F_new - F).sum(axis=0)
    llh_after = full_llh(F_new, sumF_new, indptr, indices)
    return F_new, sumF_new, llh_after, steps


def fit(F, indptr, indices, tol=1e-4, max_sweeps=50):
    """Convergence loop (MBSGD, codes/bigclamv3-7.scala:206-225)."""
    sumF = F.sum(axis=0)
    llh_old = 0.0
    history = []
    for _ in range(max_sweeps):
        F, sumF, llh, _ = sweep(F, sumF, indptr, indices)
        history.append(llh)
        if llh_old != 0.0 and abs(1.0 - llh / llh_old) < tol:
            break
        llh_old = llh
    return F, sumF, history


def extract_communities(F, num_edges, argmax_fallback=True):
    """Delta-threshold community assignment (codes/Bigclamv2.scala:223-230).

    eps = background edge density; delta = sqrt(-log(1-eps)); node u belongs
    to community c iff F[u,c] >= delta; if no community qualifies, u joins
    the argmax columns (ties included, matching the reference).  Deviation
    (documented): all-zero rows are assigned to NO community (the reference
    would assign them to every community — a latent quirk).
    """
    n = F.shape[0]
    eps = 2.0 * num_edges / (n * (n - 1))
    delta = np.sqrt(-np.log(1.0 - eps))
    members = []
    for u in range(n):
        row = F[u]
        fmax = row.max()
        if fmax >= delta:
            members.append(np.nonzero(row >= delta)[0])
        elif argmax_fallback and fmax > 0:
            members.append(np.nonzero(row == fmax)[0])
        else:
            members.append(np.array([], dtype=np.int64))
    return members, delta


def conductance(indptr, indices, u, total_degree):
    """Ego-net conductance of node u (codes/bigclamv3-7.scala:43-49).

    y = {u} ∪ N(u); z = multiset of neighbors of members of y;
    cut = |{z outside y}|; volS = |z| - cut; volT = Σdeg - volS - 2·cut;
    cond = cut/min(volS, volT) with the reference's 0/1 guards.
    """
    nbrs = indices[indptr[u] : indptr[u + 1]]
    y = set([u]) | set(int(v) for v in nbrs)
    cut = 0
    z_size = 0
    for m in y:
        for w in indices[indptr[m] : indptr[m + 1]]:
            z_size += 1
            if int(w) not in y:
                cut += 1
    vol_s = z_size - cut
    vol_t = total_degree - vol_s - 2 * cut
    if vol_s == 0:
        return 0.0
    if vol_t == 0:
        return 1.0
    return cut / min(vol_s, vol_t)


def armijo_margin_f64(graph, F, grad_u, u, s, cfg):
    """fp64 Armijo acceptance margin of step ``s`` for node ``u``:
    llh_u(clamp(F_u + s·grad_u)) − llh_u(F_u) − α·s·‖grad_u‖².
    A margin within fp32 noise of zero means accept/reject is a tie and
    fp32 implementations may legitimately disagree on the pick."""
    Fd = np.asarray(F, dtype=np.float64)
    sumF = Fd.sum(0)
    gu = np.asarray(grad_u, dtype=np.float64)
    nbrs = graph.indices[graph.indptr[u] : graph.indptr[u + 1]]

    def llh_u(fu):
        x = Fd[nbrs] @ fu
        p = np.clip(np.exp(-x), cfg.min_p, cfg.max_p)
        sf = sumF - Fd[u] + fu
        return float(np.sum(np.log(1 - p) + x) - fu @ sf + fu @ fu)

    fu2 = np.clip(Fd[u] + s * gu, cfg.min_f, cfg.max_f)
    return llh_u(fu2) - llh_u(Fd[u]) - cfg.alpha * s * (gu @ gu)
