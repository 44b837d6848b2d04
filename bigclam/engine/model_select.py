"""Model selection over K (v4 semantics, codes/bigclam4-7.scala:115-266).

Sweeps a geometric K grid (config.k_grid); for each K re-initializes F from
the one-time conductance seed ranking and fits to convergence; stops when
the LLH gain over the previous K flattens: ``(1 - LLH_new/LLH_old) < k_tol``
(no abs — codes/bigclam4-7.scala:259).  Returns the selected K.

Deviations (documented): the reference's ``LLHKold == null`` first-iteration
branch is dead code (always false on a Double) — we use an explicit
first-iteration flag; and we report the converged LLH rather than the
second-to-last sweep's (``SGDFindC`` returns the pre-convergence LLHold,
codes/bigclam4-7.scala:225-243 — a REPL quirk, not a capability).
"""
from __future__ import annotations

import dataclasses
from typing import Optional

from ..config import BigClamConfig, k_grid
from ..io.edgelist import Graph
from ..utils.metrics import MetricsLogger
from .trainer import Trainer


def select_k(
    graph: Graph,
    cfg: BigClamConfig,
    metrics: Optional[MetricsLogger] = None,
    init: str = "seed",
) -> dict:
    ks = k_grid(cfg.k_min, cfg.k_max, cfg.k_div)
    metrics = metrics or MetricsLogger(quiet=True)
    llh_old = None
    k_for_c = 0
    history = []
    seeds = None
    for k in ks:
        kcfg = dataclasses.replace(cfg, k=k)
        tr = Trainer(graph, kcfg, metrics=metrics)
        if seeds is not None:
            tr._seeds = seeds  # rank the seeds once, reuse per K
        res = tr.fit(init=init)
        seeds = tr._seeds
        llh = res.llh
        history.append({"k": k, "llh": llh, "sweeps": res.sweeps})
        metrics.log({"select_k": k, "llh": llh, "sweeps": res.sweeps})
        k_for_c = k  # if the grid exhausts without flattening, keep the
        # largest K tried (the reference loop simply runs out of grid)
        if llh_old is not None and (1.0 - llh / llh_old) < cfg.k_tol:
            break
        llh_old = llh
    return {"k": k_for_c, "history": history, "grid": ks}
