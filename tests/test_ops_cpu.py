"""Torch reference ops (the CPU path) vs the fp64 NumPy oracle."""
import numpy as np
import pytest
import torch

import oracle
from bigclam.config import BigClamConfig
from bigclam.ops import reference as ops


def _setup(g, k, seed=0, scale=0.5):
    rng = np.random.default_rng(seed)
    F = rng.random((g.num_nodes, k)).astype(np.float32) * scale
    Ft = torch.from_numpy(F)
    sumF = Ft.sum(dim=0)
    indptr = torch.from_numpy(g.indptr)
    indices = torch.from_numpy(g.indices)
    return F.astype(np.float64), Ft, sumF, indptr, indices


def test_edge_grad_llh_matches_oracle(small_graph):
    g = small_graph
    cfg = BigClamConfig(k=5)
    F64, Ft, sumF, indptr, indices = _setup(g, 5, seed=1)
    grad, llh = ops.edge_grad_llh(Ft, indptr, indices, sumF, cfg)
    sumF64 = F64.sum(axis=0)
    for u in range(g.num_nodes):
        og, ol = oracle.node_grad_llh(F64, sumF64, g.indptr, g.indices, u)
        np.testing.assert_allclose(grad[u].numpy(), og, rtol=2e-4, atol=2e-4)
        assert abs(llh[u].item() - ol) < 1e-3 * max(1.0, abs(ol))


def test_linesearch_matches_oracle(small_graph):
    g = small_graph
    cfg = BigClamConfig(k=4)
    F64, Ft, sumF, indptr, indices = _setup(g, 4, seed=2, scale=0.4)
    grad, llh = ops.edge_grad_llh(Ft, indptr, indices, sumF, cfg)
    best = ops.linesearch(Ft, indptr, indices, sumF, grad, llh, cfg)
    sumF64 = F64.sum(axis=0)
    mismatches = 0
    for u in range(g.num_nodes):
        og, ol = oracle.node_grad_llh(F64, sumF64, g.indptr, g.indices, u)
        s = oracle.line_search(F64, sumF64, g.indptr, g.indices, u, og, ol)
        if not np.isclose(best[u].item(), s, rtol=1e-6, atol=1e-12):
            mismatches += 1
    # fp32 vs fp64 can flip a borderline Armijo accept; allow a rare flip
    assert mismatches <= max(1, g.num_nodes // 20)


def test_apply_step_projection_and_delta(small_graph):
    g = small_graph
    cfg = BigClamConfig(k=3)
    _, Ft, sumF, indptr, indices = _setup(g, 3, seed=3)
    grad, llh = ops.edge_grad_llh(Ft, indptr, indices, sumF, cfg)
    steps = torch.zeros(g.num_nodes)
    steps[::2] = 0.1
    F_new, delta = ops.apply_step(Ft, grad, steps, cfg)
    assert (F_new >= cfg.min_f).all() and (F_new <= cfg.max_f).all()
    np.testing.assert_allclose(
        delta.numpy(), (F_new - Ft).sum(dim=0).numpy(), rtol=1e-5, atol=1e-5
    )
    # untouched rows unchanged
    assert torch.equal(F_new[1::2], Ft[1::2])


def test_full_llh_matches_oracle(small_graph):
    g = small_graph
    cfg = BigClamConfig(k=6)
    F64, Ft, sumF, indptr, indices = _setup(g, 6, seed=4)
    t = ops.full_llh(Ft, indptr, indices, sumF, cfg)
    o = oracle.full_llh(F64, F64.sum(axis=0), g.indptr, g.indices)
    assert abs(t.item() - o) < 1e-3 * abs(o)


def test_chunking_invariance(small_graph):
    g = small_graph
    cfg = BigClamConfig(k=4)
    _, Ft, sumF, indptr, indices = _setup(g, 4, seed=5)
    g1 = ops.edge_grad_llh(Ft, indptr, indices, sumF, cfg, chunk=7)
    g2 = ops.edge_grad_llh(Ft, indptr, indices, sumF, cfg, chunk=10**9)
    np.testing.assert_allclose(g1[0].numpy(), g2[0].numpy(), rtol=1e-5, atol=1e-6)
    np.testing.assert_allclose(g1[1].numpy(), g2[1].numpy(), rtol=1e-9, atol=1e-9)


def test_bf16_storage_path(small_graph):
    g = small_graph
    cfg = BigClamConfig(k=4, dtype="bf16")
    _, Ft, sumF, indptr, indices = _setup(g, 4, seed=6)
    Fb = Ft.bfloat16()
    grad_b, llh_b = ops.edge_grad_llh(Fb, indptr, indices, sumF, cfg)
    grad_f, llh_f = ops.edge_grad_llh(Ft, indptr, indices, sumF, cfg)
    # bf16 storage: ~1% relative agreement with fp32
    np.testing.assert_allclose(
        grad_b.numpy(), grad_f.numpy(), rtol=0.05, atol=0.05
    )


def test_mfma_dispatch_prefix_invariant(monkeypatch):
    """The kernel dispatch assumes each launch-order list is degree-
    descending so the deg>=threshold nodes form a PREFIX of length
    n_mfma; verify on sharded states (interior/boundary included)."""
    import numpy as np
    import torch

    from bigclam.config import BigClamConfig
    from bigclam.core.shard import make_shard
    from bigclam.core.state import ShardState
    from bigclam.io import rmat_graph

    monkeypatch.setenv("BIGCLAM_MFMA_DEG", "8")
    g = rmat_graph(9, 6.0, seed=77)
    for ws, rank in [(1, 0), (3, 1)]:
        shard = make_shard(g, rank, ws)
        st = ShardState(
            shard, BigClamConfig(k=16, device="cpu"),
            device=torch.device("cpu"),
        )
        deg = shard.degrees()
        for order, n_hi in [
            (st.order, st.n_mfma),
            (st.order_interior, st.n_mfma_interior),
            (st.order_boundary, st.n_mfma_boundary),
        ]:
            o = order.cpu().numpy()
            assert np.all(deg[o[:n_hi]] >= 8)
            if n_hi < len(o):
                assert np.all(deg[o[n_hi:]] < 8)


def test_reference_ops_extreme_values_finite():
    """Numerical extremes: F rows at the clamp bound (1000) make
    x = Fu.Fv overflow exp(-x) to 0 -> p clamps to min_p; all outputs
    stay finite and llh uses the clamped p (no -inf)."""
    import numpy as np
    import torch

    from bigclam.config import BigClamConfig
    from bigclam.io import rmat_graph
    from bigclam.ops import reference as ref_ops

    g = rmat_graph(6, 4.0, seed=8)
    cfg = BigClamConfig(k=8, device="cpu")
    F = torch.full((g.num_nodes, 8), 1000.0)
    sumF = F.sum(0)
    indptr = torch.from_numpy(g.indptr)
    indices = torch.from_numpy(g.indices.astype("int64")).int()
    grad, llh = ref_ops.edge_grad_llh(F, indptr, indices, sumF, cfg)
    assert torch.isfinite(grad).all()
    assert torch.isfinite(llh).all()
    best = ref_ops.linesearch(F, indptr, indices, sumF, grad, llh, cfg)
    assert torch.isfinite(best).all()
    newF, _ = ref_ops.apply_step(F, grad, best, cfg)
    assert torch.isfinite(newF).all()
    assert (newF <= cfg.max_f).all() and (newF >= cfg.min_f).all()


def test_mfma_dispatch_defaults_by_k():
    """The measured K-aware dispatch defaults (state.py): bf16 routes ALL
    nodes to the MFMA kernel at kp <= 8192 only; above that the direct
    fused kernel (<= 16384) or the separate chunked path (> 16384) wins
    (profiles/r02_largek_dispatch.md); fp32 defaults to the direct
    kernel."""
    import torch

    from bigclam.config import BigClamConfig
    from bigclam.core.shard import make_shard
    from bigclam.core.state import ShardState
    from bigclam.io import rmat_graph

    g = rmat_graph(7, 4.0, seed=6)
    for dtype, k, expect_all in [
        ("bf16", 5000, True),
        ("bf16", 8500, False),
        ("bf16", 17000, False),
        ("fp32", 5000, False),
    ]:
        st = ShardState(
            make_shard(g, 0, 1),
            BigClamConfig(k=k, device="cpu", dtype=dtype),
            device=torch.device("cpu"),
        )
        n = int(st.order.numel())
        assert (st.n_mfma == n) if expect_all else (st.n_mfma == 0), (
            dtype, k, st.n_mfma)
