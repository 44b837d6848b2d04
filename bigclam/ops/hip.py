"""GPU op wrappers around the in-tree HIP extension ``bigclam._C``.

The extension is built by ``setup.py build_ext --inplace`` (or
``__graft_entry__.build()``) with ``hipcc --offload-arch=gfx950`` and ships
in-tree so the gpurun snapshot carries it.  On a GPU box a missing extension
is a hard error — there is deliberately no silent eager fallback here
(see core/state.py).
"""
from __future__ import annotations

from typing import Tuple

import torch

from ..config import BigClamConfig

_C = None


def ensure_loaded():
    global _C
    if _C is None:
        try:
            from .. import _C as ext
        except ImportError as e:  # pragma: no cover
            raise RuntimeError(
                "bigclam._C HIP extension not built. Run "
                "`python setup.py build_ext --inplace` (gfx950) first. "
                f"Original error: {e}"
            ) from e
        _C = ext
    return _C


_xbuf_cache = {}


def _xbuf(nnz: int, device) -> torch.Tensor:
    """Per-edge dot buffer for the chunked K1 (reused across sweeps)."""
    key = str(device)
    t = _xbuf_cache.get(key)
    if t is None or t.shape[0] < nnz:
        t = torch.empty(nnz, device=device, dtype=torch.float32)
        _xbuf_cache[key] = t
    return t


def _use_chunked_k1(F: torch.Tensor) -> bool:
    """Chunked KD+KW replaces the one-pass K1 where gacc[K] LDS residency
    would cap occupancy at 1 block/CU (measured 235-247 ms/sweep at
    K=25000, profiles/r01_kernel_opt_log.md).  BIGCLAM_K1_CHUNKED=1/0
    forces it on/off (tests + dispatch measurements)."""
    import os

    env = os.environ.get("BIGCLAM_K1_CHUNKED")
    if env is not None:
        return env != "0"
    cap = 16384 if F.dtype == torch.bfloat16 else 8192
    return F.shape[1] > cap


def edge_grad_llh(
    F: torch.Tensor,
    indptr: torch.Tensor,
    indices: torch.Tensor,
    sumF: torch.Tensor,
    order: torch.Tensor,
    cfg: BigClamConfig,
    out: Tuple[torch.Tensor, torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """K1 over the nodes listed in ``order`` (grid = len(order)); rows of
    grad/llh not listed stay untouched, so interior/boundary subsets can be
    launched separately into the same ``out`` buffers (halo overlap)."""
    ext = ensure_loaded()
    n_local = len(indptr) - 1
    if out is None:
        grad = torch.empty(
            n_local, F.shape[1], device=F.device, dtype=torch.float32
        )
        llh = torch.empty(n_local, device=F.device, dtype=torch.float64)
    else:
        grad, llh = out
    if _use_chunked_k1(F):
        ext.edge_grad_llh_chunked(
            F, indptr, indices, sumF, order, grad, llh,
            _xbuf(indices.shape[0], F.device), cfg.min_p, cfg.max_p,
        )
    else:
        ext.edge_grad_llh(
            F, indptr, indices, sumF, order, grad, llh, cfg.min_p, cfg.max_p
        )
    return grad, llh


_ladder_cache = {}


def _ladder(cfg: BigClamConfig, device) -> torch.Tensor:
    key = (tuple(cfg.ladder()), str(device))
    t = _ladder_cache.get(key)
    if t is None:  # cached: a fresh HtoD copy per sweep costs a sync
        t = torch.tensor(cfg.ladder(), device=device, dtype=torch.float32)
        _ladder_cache[key] = t
    return t


def fused_grad_ls(
    F: torch.Tensor,
    indptr: torch.Tensor,
    indices: torch.Tensor,
    sumF: torch.Tensor,
    order: torch.Tensor,
    cfg: BigClamConfig,
    out=None,
    n_mfma: int = 0,
):
    """KF: fused K1 gradient+LLH and K2 line search in one per-node pass.

    Returns (grad [n,K], llh [n] f64, best_step [n]).  ``order`` may be a
    node subset (halo overlap) writing into shared ``out`` buffers.
    ``n_mfma``: the first n_mfma entries of ``order`` (the high-degree
    prefix of the degree-descending launch order) run the MFMA phase-B
    kernel; the rest the direct one."""
    ext = ensure_loaded()
    n_local = len(indptr) - 1
    if out is None:
        grad = torch.empty(
            n_local, F.shape[1], device=F.device, dtype=torch.float32
        )
        llh = torch.empty(n_local, device=F.device, dtype=torch.float64)
        best = torch.empty(n_local, device=F.device, dtype=torch.float32)
    else:
        grad, llh, best = out
    ext.fused_grad_ls(
        F, indptr, indices, sumF, order, grad, llh,
        _ladder(cfg, F.device), best,
        cfg.alpha, cfg.min_p, cfg.max_p, cfg.min_f, cfg.max_f,
        int(n_mfma),
    )
    return grad, llh, best


def linesearch(
    F: torch.Tensor,
    indptr: torch.Tensor,
    indices: torch.Tensor,
    sumF: torch.Tensor,
    grad: torch.Tensor,
    llh: torch.Tensor,
    order: torch.Tensor,
    cfg: BigClamConfig,
    out: torch.Tensor = None,
) -> torch.Tensor:
    """K2 over the nodes listed in ``order`` (subset-capable like K1);
    with ``out`` the results land in the caller's buffer at the listed
    positions."""
    ext = ensure_loaded()
    n_local = len(indptr) - 1
    best = out if out is not None else torch.empty(
        n_local, device=F.device, dtype=torch.float32
    )
    ladder = _ladder(cfg, F.device)
    ext.linesearch(
        F,
        indptr,
        indices,
        sumF,
        grad,
        llh,
        order,
        ladder,
        best,
        cfg.alpha,
        cfg.min_p,
        cfg.max_p,
        cfg.min_f,
        cfg.max_f,
    )
    return best


def apply_step(
    F_local: torch.Tensor,
    grad: torch.Tensor,
    steps: torch.Tensor,
    cfg: BigClamConfig,
) -> None:
    """K3: in-place projected commit on the owned rows."""
    ext = ensure_loaded()
    ext.apply_step(F_local, grad, steps, cfg.min_f, cfg.max_f)


def apply_step_colsum(
    F_local: torch.Tensor,
    grad: torch.Tensor,
    steps: torch.Tensor,
    partials: torch.Tensor,
    cfg: BigClamConfig,
) -> torch.Tensor:
    """K3+colsum fused (bf16 storage): commits F in place and returns the
    exact fp32 column sums of the updated shard (one pass over F instead
    of commit + fp32 materialize + torch reduce)."""
    ext = ensure_loaded()
    ext.apply_step_colsum(
        F_local, grad, steps, partials, cfg.min_f, cfg.max_f
    )
    return partials.sum(dim=0)


def seed_init_device(state, graph, seeds, include_seed: bool) -> None:
    """K6: device-side seed-init scatter — F[v, c] = 1 for v in
    N(seed_c) (codes/bigclamv3-7.scala:60-87).  The seeds' compact
    adjacency (k rows) uploads in a few hundred KB; works at any world
    size (off-shard rows are skipped in-kernel).  The caller handles
    pad columns (only exist when #seeds < k) and the sumF refresh."""
    import numpy as np

    ext = ensure_loaded()
    seeds = np.asarray(seeds, dtype=np.int64)
    deg = (graph.indptr[seeds + 1] - graph.indptr[seeds]).astype(np.int64)
    sindptr = np.concatenate([[0], np.cumsum(deg)])
    offs = np.arange(int(deg.sum()), dtype=np.int64) - np.repeat(
        np.cumsum(deg) - deg, deg
    )
    snbrs = graph.indices[
        np.repeat(graph.indptr[seeds], deg) + offs
    ].astype(np.int64)
    dev = state.device
    state.F.zero_()
    ext.seed_init(
        state.F_local,
        torch.from_numpy(sindptr).to(dev),
        torch.from_numpy(snbrs).to(dev),
        torch.from_numpy(seeds).to(dev),
        state.shard.start,
        state.shard.stop,
        include_seed,
    )


def conductance_full_graph(graph, device) -> "torch.Tensor":
    """K5: ego-net conductance of every node of the FULL graph on GPU.

    Used by the seed-init ranking; the ranking itself is a
    trivial host-side pass over the returned [N] fp64 vector.
    """
    import numpy as np

    ext = ensure_loaded()
    indptr = torch.from_numpy(graph.indptr.astype(np.int64)).to(device)
    indices = torch.from_numpy(graph.indices.astype(np.int32)).to(device)
    cond = torch.empty(graph.num_nodes, device=device, dtype=torch.float64)
    total_degree = float(len(graph.indices))
    ext.conductance(indptr, indices, cond, total_degree)
    return cond


def sparse_sweep_part(
    F: torch.Tensor,
    indptr: torch.Tensor,
    indices: torch.Tensor,
    sumF: torch.Tensor,
    order_sparse: torch.Tensor,
    soffset: torch.Tensor,
    sidx: torch.Tensor,
    sval: torch.Tensor,
    scount: torch.Tensor,
    epos: torch.Tensor,
    goffset: torch.Tensor,
    gpool_size: int,  # kept for API stability; pools come from the state
    cap: int,
    llh_out: torch.Tensor,
    best_out: torch.Tensor,
    cfg: BigClamConfig,
    state_pools=None,
):
    """KFS for the routed (sparse) nodes: one fused launch computing the
    compact gradient pools, llh per node, and the Armijo best step.
    Writes llh_out/best_out at the routed node positions; returns the
    commit pack for K3S."""
    ext = ensure_loaded()
    dev = F.device
    n_s = int(order_sparse.numel())
    # per-STATE grad pools (via state_pools): the commit consumes them at
    # the START of the next sweep (the pipelined carry), so a shared
    # module pool could be clobbered by another coexisting state
    gidx, gval = state_pools
    gcount = torch.empty(n_s, device=dev, dtype=torch.int32)
    GG = (sumF * sumF).sum().reshape(1)  # device scalar: no host sync
    ext.sparse_fused(
        F, indptr, indices, sumF, order_sparse, soffset, sidx, sval, scount,
        epos, goffset, gidx, gval, gcount, llh_out, GG, _ladder(cfg, dev),
        best_out, cap, cfg.alpha, cfg.min_p, cfg.max_p, cfg.min_f, cfg.max_f,
    )
    return {
        "order": order_sparse,
        "goffset": goffset,
        "gidx": gidx,
        "gval": gval,
        "gcount": gcount,
    }


def sparse_commit(F_local: torch.Tensor, pack: dict, best: torch.Tensor,
                  state, cfg: BigClamConfig):
    """K3S: projected commit confined to each routed node's active set,
    rewriting the committed rows' persistent support lists in place (the
    next sweep's KAF then skips them)."""
    ensure_loaded().sparse_commit(
        F_local, pack["order"], pack["goffset"], pack["gidx"], pack["gval"],
        pack["gcount"], best, state._sp_soffset, state._sp_sidx,
        state._sp_sval, state._sp_scount, state._sp_cap,
        cfg.min_f, cfg.max_f,
    )


def extract_membership(
    F_local: torch.Tensor, k_true: int, delta: float
) -> Tuple[torch.Tensor, torch.Tensor]:
    """K7: per-row community memberships of the local F shard on device.

    Returns ``(counts int32 [n], comms int32 [total])`` — row u's
    memberships are ``comms[offsets[u] : offsets[u] + counts[u]]`` in
    ascending community order (offsets = exclusive prefix sum of counts).
    Replaces the reference's driver-side threshold pass
    (codes/Bigclamv2.scala:226-229) without materializing N×K anywhere.
    """
    ext = ensure_loaded()
    n = F_local.shape[0]
    counts = torch.empty(n, device=F_local.device, dtype=torch.int32)
    ext.extract_count(F_local, k_true, delta, counts)
    offsets = torch.cumsum(counts, 0, dtype=torch.int64) - counts.to(torch.int64)
    total = int(counts.sum().item())
    comms = torch.empty(total, device=F_local.device, dtype=torch.int32)
    ext.extract_fill(F_local, k_true, delta, offsets, comms)
    return counts, comms


def full_llh(
    F: torch.Tensor,
    indptr: torch.Tensor,
    indices: torch.Tensor,
    sumF: torch.Tensor,
    order: torch.Tensor,
    cfg: BigClamConfig,
) -> torch.Tensor:
    ext = ensure_loaded()
    n_local = len(indptr) - 1
    llh = torch.empty(n_local, device=F.device, dtype=torch.float64)
    ext.llh_only(F, indptr, indices, sumF, order, llh, cfg.min_p, cfg.max_p)
    return llh.sum()
