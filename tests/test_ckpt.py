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
    sweep0 = resume(str(tmp_path), tr2)
    assert sweep0 == res.sweeps
    np.testing.assert_allclose(
        tr2.state.F_local_k.numpy(), tr.state.F_local_k.numpy(), rtol=1e-6
    )
    np.testing.assert_allclose(
        tr2.state.sumF.numpy(), tr.state.sumF.numpy(), rtol=1e-5
    )
    out = tr2.sweep()
    assert np.isfinite(out["llh"])


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
