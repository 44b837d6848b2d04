"""Multi-process shard correctness on CPU (gloo, world_size 2-4).

The halo-exchange + allreduce path must produce results identical (to fp32
tolerance) to the single-shard run — RCCL on GPU exercises the exact same
code path (SURVEY.md §4 'distributed without a cluster').
"""
import json
import os
import sys
import tempfile

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from bigclam.config import BigClamConfig
from bigclam.core.shard import make_shard, partition_bounds
from bigclam.engine.trainer import Trainer
from bigclam.io import planted_partition


def _graph():
    g, _ = planted_partition(3, 14, p_in=0.5, p_out=0.03, seed=13)
    return g


def _single_run(n_sweeps=3):
    g = _graph()
    cfg = BigClamConfig(k=3, device="cpu", seed=5)
    tr = Trainer(g, cfg, rank=0, world_size=1, device=torch.device("cpu"))
    tr.init_F("random")
    llh = [tr.sweep()["llh"] for _ in range(n_sweeps)]
    return llh, tr.state.F_local_k.numpy().copy()


def _worker(rank, world_size, port, out_dir, n_sweeps):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world_size)
    try:
        g = _graph()
        cfg = BigClamConfig(k=3, device="cpu", seed=5)
        tr = Trainer(g, cfg, device=torch.device("cpu"))
        tr.init_F("random")
        llh = [tr.sweep()["llh"] for _ in range(n_sweeps)]
        F = tr.gather_F()
        if rank == 0:
            np.save(os.path.join(out_dir, "F.npy"), F.numpy())
            with open(os.path.join(out_dir, "llh.json"), "w") as f:
                json.dump(llh, f)
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("world_size", [2, 3, 4])
def test_sharded_equals_single(world_size, tmp_path):
    n_sweeps = 3
    llh1, F1 = _single_run(n_sweeps)
    port = 29600 + world_size
    mp.spawn(
        _worker,
        args=(world_size, port, str(tmp_path), n_sweeps),
        nprocs=world_size,
        join=True,
    )
    llh_w = json.load(open(tmp_path / "llh.json"))
    F_w = np.load(tmp_path / "F.npy")
    for a, b in zip(llh1, llh_w):
        assert abs(a - b) < 1e-6 * max(1.0, abs(a)), (llh1, llh_w)
    np.testing.assert_allclose(F_w, F1, rtol=1e-5, atol=1e-6)


def test_halo_plan_consistency():
    """Every rank's send list matches the peers' recv expectations."""
    g = _graph()
    for ws in (2, 3, 4):
        bounds = partition_bounds(g, ws)
        shards = [make_shard(g, r, ws, bounds) for r in range(ws)]
        for r in range(ws):
            for p in range(ws):
                if r == p:
                    continue
                sent = shards[r].plan.send_idx[p] + shards[r].start
                want = shards[p].halo_globals
                want_from_r = want[(want >= shards[r].start) & (want < shards[r].stop)]
                np.testing.assert_array_equal(sent, want_from_r)
                assert shards[p].plan.recv_counts[r] == len(sent)
        # local indices reference valid rows and reproduce the global CSR
        for s in shards:
            assert s.indices.min() >= 0
            assert s.indices.max() < s.n_rows
            # remap back to global and compare against the global graph rows
            back = np.where(
                s.indices < s.n_local,
                s.indices + s.start,
                # halo section
                s.halo_globals[np.clip(s.indices - s.n_local, 0, None)]
                if s.n_halo
                else s.indices,
            )
            for u_local in range(s.n_local):
                row = back[s.indptr[u_local] : s.indptr[u_local + 1]]
                np.testing.assert_array_equal(
                    np.sort(row), g.neighbors(u_local + s.start)
                )
