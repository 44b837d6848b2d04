"""Sparse-sweep prototype vs the dense torch reference (exact active
sets; docs/sparse_sweep_design.md)."""
import numpy as np
import pytest
import torch

from bigclam.config import BigClamConfig
from bigclam.core.shard import make_shard
from bigclam.core.state import ShardState
from bigclam.engine.trainer import Trainer
from bigclam.io import rmat_graph
from bigclam.ops import reference as ref_ops
from bigclam.ops.sparse_proto import row_support, sparse_sweep_node


def _sparse_state():
    """A partially-converged fit: rows carry exact zeros."""
    g = rmat_graph(8, 5.0, seed=13)
    cfg = BigClamConfig(k=48, device="cpu", max_sweeps=12, seed=9)
    tr = Trainer(g, cfg, rank=0, world_size=1, device=torch.device("cpu"))
    tr.init_F("seed")
    carry, _ = tr.prologue()
    for _ in range(12):
        carry, _, _ = tr.pipelined_sweep(carry)
    F = tr.state.F_local_k.float().numpy().copy()
    assert 0.0 < (F != 0).mean() < 0.9, "state should be partially sparse"
    return g, cfg, F


def test_sparse_sweep_matches_dense_reference():
    g, cfg, F = _sparse_state()
    Ft = torch.from_numpy(F)
    sumF = Ft.sum(0)
    indptr = torch.from_numpy(g.indptr)
    indices = torch.from_numpy(g.indices.astype(np.int64)).int()
    grad, llh = ref_ops.edge_grad_llh(Ft, indptr, indices, sumF, cfg)
    best = ref_ops.linesearch(Ft, indptr, indices, sumF, grad, llh, cfg)
    newF, _ = ref_ops.apply_step(Ft, grad, best, cfg)

    supp = row_support(F)
    sumF_np = sumF.numpy()
    GG = float((sumF_np.astype(np.float64) ** 2).sum())
    n_checked = n_step_agree = 0
    rng = np.random.default_rng(0)
    nodes = rng.choice(g.num_nodes, size=60, replace=False)
    for u in nodes:
        sp, g_sp, llh_u, b, S, new_row = sparse_sweep_node(
            int(u), F, supp, g.indptr, g.indices, sumF_np, GG, cfg
        )
        du = grad[u].numpy()
        # set-containment claims are EXACT: off-S' the gradient is -sumF
        off = np.setdiff1d(np.arange(F.shape[1]), sp)
        np.testing.assert_allclose(du[off], -sumF_np[off], rtol=0, atol=1e-5)
        # on-S' gradient values (fp32 accumulation-order tolerance)
        np.testing.assert_allclose(g_sp, du[sp], rtol=1e-4, atol=2e-4)
        # local LLH
        assert abs(llh_u - float(llh[u])) < 1e-6 * max(1.0, abs(float(llh[u])))
        # committed row: zero off S, equal on S when the steps agree.
        # At rungs <= beta^5 the Armijo margin (s*gg) drops below fp32
        # summation-order noise, so cross-implementation picks there are
        # allowed to differ (the same regime the HIP-vs-torch tests
        # cover with agreement fractions); any disagreement at a
        # MEANINGFUL step is a real bug and fails.
        rb = float(best[u])
        if b == rb:
            n_step_agree += 1
            ref_row = newF[u].numpy()
            np.testing.assert_allclose(new_row, ref_row[S], rtol=1e-4,
                                       atol=2e-4)
            offS = np.setdiff1d(np.arange(F.shape[1]), S)
            np.testing.assert_allclose(ref_row[offS], 0.0, rtol=0, atol=0)
        else:
            noise_rung = cfg.beta ** 5
            if max(b, rb) > noise_rung:
                # above the noise rung a disagreement must be a genuine
                # fp32 Armijo tie: fp64 margin within fp32 noise of zero
                from oracle import armijo_margin_f64

                m = armijo_margin_f64(g, F, grad[u].numpy(), int(u),
                                      max(b, rb), cfg)
                assert abs(m) < 5e-5 * max(1.0, abs(float(llh[u]))), (
                    f"node {u}: sparse step {b} vs dense {rb}, margin {m}"
                )
        n_checked += 1
    assert n_checked == 60
    assert n_step_agree >= 30, f"only {n_step_agree}/60 steps agree"
