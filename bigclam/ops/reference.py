"""Vectorized PyTorch reference implementations of the BigCLAM ops.

These are the CPU execution path (facebook_combined / gloo configs of
BASELINE.json) and the numerics reference the HIP kernels are tested
against.  Math follows the reference exactly (SURVEY.md §2.6-§2.10;
codes/bigclamv3-7.scala:89-204): clamps, edge-doubling convention, Jacobi
line search, sumF trick.

Edge-chunked so peak memory stays bounded at ``chunk x K`` regardless of
graph size.  LLH accumulation is fp64 to keep the convergence test stable
at large E.
"""
from __future__ import annotations

from typing import Tuple

import torch

from ..config import BigClamConfig

_DEF_CHUNK = 1 << 20


def _edge_src(indptr: torch.Tensor) -> torch.Tensor:
    """Expand CSR indptr to a per-edge source-row tensor."""
    deg = indptr[1:] - indptr[:-1]
    return torch.repeat_interleave(
        torch.arange(len(deg), device=indptr.device, dtype=torch.int64), deg
    )


def edge_grad_llh(
    F: torch.Tensor,
    indptr: torch.Tensor,
    indices: torch.Tensor,
    sumF: torch.Tensor,
    cfg: BigClamConfig,
    n_local: int = None,
    edge_src: torch.Tensor = None,
    chunk: int = _DEF_CHUNK,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """K1: per-node gradient + local LLH (codes/bigclamv3-7.scala:138-150).

    ``F`` is the shard row buffer [n_local + n_halo, K]; rows [0, n_local)
    are owned.  Returns (grad [n_local, K], llh [n_local] fp64).
    """
    n_local = n_local if n_local is not None else len(indptr) - 1
    K = F.shape[1]
    dev = F.device
    src = edge_src if edge_src is not None else _edge_src(indptr)
    Fl = F[:n_local].float()
    grad_acc = torch.zeros(n_local, K, device=dev, dtype=torch.float32)
    llh_acc = torch.zeros(n_local, device=dev, dtype=torch.float64)
    for e0 in range(0, len(indices), chunk):
        e1 = min(e0 + chunk, len(indices))
        s = src[e0:e1]
        d = indices[e0:e1].long()
        Fv = F[d].float()
        x = (Fl[s] * Fv).sum(-1)
        p = torch.clamp(torch.exp(-x), cfg.min_p, cfg.max_p)
        w = 1.0 / (1.0 - p)
        llh_e = torch.log1p(-p).double() + x.double()
        grad_acc.index_add_(0, s, Fv * w.unsqueeze(1))
        llh_acc.index_add_(0, s, llh_e)
    sf = sumF.float()
    grad = grad_acc - sf + Fl
    node_term = (-(Fl @ sf) + (Fl * Fl).sum(-1)).double()
    return grad, llh_acc + node_term


def linesearch(
    F: torch.Tensor,
    indptr: torch.Tensor,
    indices: torch.Tensor,
    sumF: torch.Tensor,
    grad: torch.Tensor,
    llh: torch.Tensor,
    cfg: BigClamConfig,
    n_local: int = None,
    edge_src: torch.Tensor = None,
    chunk: int = _DEF_CHUNK,
) -> torch.Tensor:
    """K2: best accepted Armijo step per node (codes/bigclamv3-7.scala:153-163).

    Jacobi semantics — trial rows are evaluated against the *unchanged*
    neighbor rows, with sumF adjusted only for u's own move.  The ladder is
    descending, so the first accepted candidate is the max accepted step.
    Nodes already accepted are masked out of later (smaller-step) passes.
    Returns best_step [n_local] (0 where no candidate accepted).
    """
    n_local = n_local if n_local is not None else len(indptr) - 1
    dev = F.device
    src = edge_src if edge_src is not None else _edge_src(indptr)
    Fl = F[:n_local].float()
    sf = sumF.float()
    gg = (grad.double() * grad.double()).sum(-1)
    best = torch.zeros(n_local, device=dev, dtype=torch.float32)
    remaining = torch.ones(n_local, device=dev, dtype=torch.bool)
    for s_val in [cfg.beta ** i for i in range(cfg.ls_steps + 1)]:
        if not bool(remaining.any()):
            break
        Fc = torch.clamp(Fl + s_val * grad, cfg.min_f, cfg.max_f)
        edge_acc = torch.zeros(n_local, device=dev, dtype=torch.float64)
        # only edges of still-searching nodes contribute
        emask_rows = remaining
        for e0 in range(0, len(indices), chunk):
            e1 = min(e0 + chunk, len(indices))
            srow = src[e0:e1]
            keep = emask_rows[srow]
            srow = srow[keep]
            if srow.numel() == 0:
                continue
            d = indices[e0:e1].long()[keep]
            x = (Fc[srow] * F[d].float()).sum(-1)
            p = torch.clamp(torch.exp(-x), cfg.min_p, cfg.max_p)
            edge_acc.index_add_(0, srow, torch.log1p(-p).double() + x.double())
        sf_adj = sf.unsqueeze(0) - Fl + Fc  # per-node sumF' = sumF - Fu + Fu'
        node_term = (-(Fc * sf_adj).sum(-1) + (Fc * Fc).sum(-1)).double()
        trial = edge_acc + node_term
        accept = remaining & (trial >= llh + cfg.alpha * s_val * gg)
        best = torch.where(accept, torch.full_like(best, s_val), best)
        remaining = remaining & ~accept
    return best


def apply_step(
    F_local: torch.Tensor,
    grad: torch.Tensor,
    steps: torch.Tensor,
    cfg: BigClamConfig,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """K3: projected commit + column-sum delta (codes/bigclamv3-7.scala:89-92,
    171-173).  Returns (F_new [n_local, K], delta_sumF [K] fp32)."""
    mask = (steps > 0).unsqueeze(1)
    Fl = F_local.float()
    F_new = torch.where(
        mask, torch.clamp(Fl + steps.unsqueeze(1) * grad, cfg.min_f, cfg.max_f), Fl
    )
    delta = (F_new - Fl).sum(dim=0)
    return F_new.to(F_local.dtype), delta


def full_llh(
    F: torch.Tensor,
    indptr: torch.Tensor,
    indices: torch.Tensor,
    sumF: torch.Tensor,
    cfg: BigClamConfig,
    n_local: int = None,
    edge_src: torch.Tensor = None,
    chunk: int = _DEF_CHUNK,
) -> torch.Tensor:
    """K4: total LLH over local nodes (codes/bigclamv3-7.scala:106-120,
    177-200).  fp64 scalar; sum over ranks is the global objective."""
    n_local = n_local if n_local is not None else len(indptr) - 1
    src = edge_src if edge_src is not None else _edge_src(indptr)
    Fl = F[:n_local].float()
    sf = sumF.float()
    acc = torch.zeros((), device=F.device, dtype=torch.float64)
    for e0 in range(0, len(indices), chunk):
        e1 = min(e0 + chunk, len(indices))
        s = src[e0:e1]
        d = indices[e0:e1].long()
        x = (Fl[s] * F[d].float()).sum(-1)
        p = torch.clamp(torch.exp(-x), cfg.min_p, cfg.max_p)
        acc = acc + (torch.log1p(-p).double() + x.double()).sum()
    node = (-(Fl @ sf) + (Fl * Fl).sum(-1)).double().sum()
    return acc + node
