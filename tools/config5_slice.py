#!/usr/bin/env python3
"""BASELINE config #5 single-GPU slice.

Builds the 100M-edge R-MAT (scale 26, ef 1.5 -> N=15.0M, 100.3M undirected
edges), takes rank 0's ws=8 shard (edge-balanced contiguous ranges), puts
the full shard state (owned + halo F rows, bf16) in HBM and times real
K1 -> K2 -> K3 sweeps on it.  Measures what the r01 "288 GB sizing" was
arithmetic about: the worst-rank HBM footprint at K=10000 including the
halo buffer and the fp32 gradient.

  python tools/config5_slice.py --k 10000 --steps 3 --warmup 1
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np  # noqa: E402
import torch  # noqa: E402

from bigclam.config import BigClamConfig  # noqa: E402
from bigclam.core.shard import make_shard, partition_bounds  # noqa: E402
from bigclam.core.state import ShardState  # noqa: E402
from bigclam.io import rmat_graph  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--k", type=int, default=10000)
    ap.add_argument("--scale", type=int, default=26)
    ap.add_argument("--ef", type=float, default=1.5)
    ap.add_argument("--rank", type=int, default=0)
    ap.add_argument("--ws", type=int, default=8)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--dtype", default="bf16")
    args = ap.parse_args()
    assert torch.cuda.is_available()

    t0 = time.perf_counter()
    g = rmat_graph(args.scale, args.ef, seed=0)
    t_build = time.perf_counter() - t0
    print(json.dumps({
        "phase": "host_graph", "seconds": round(t_build, 1),
        "nodes": g.num_nodes, "undirected_edges": g.num_edges,
    }), flush=True)

    t0 = time.perf_counter()
    bounds = partition_bounds(g, args.ws)
    shard = make_shard(g, args.rank, args.ws, bounds)
    t_shard = time.perf_counter() - t0
    cfg = BigClamConfig(k=args.k, dtype=args.dtype, device="cuda", seed=7)
    st = ShardState(shard, cfg, device=torch.device("cuda"))
    # device-side init of ALL rows (halo rows stand in for peers' F)
    torch.manual_seed(3)
    blk = 65536
    scale = 1.0 / (args.k ** 0.5)
    for i in range(0, shard.n_rows, blk):
        m = min(blk, shard.n_rows - i)
        st.F[i : i + blk, : args.k] = (
            torch.rand(m, args.k, device="cuda") * scale
        ).to(st.storage_dtype)
    # stand-in global sumF: shard colsum scaled to the full node count
    # (blockwise — a full F.float() copy would be 266 GB)
    s = torch.zeros(st.kp, device="cuda", dtype=torch.float32)
    for i in range(0, shard.n_rows, blk):
        s += st.F[i : i + blk].float().sum(0)
    st.sumF = s * (g.num_nodes / shard.n_rows)
    torch.cuda.synchronize()
    alloc0 = torch.cuda.memory_allocated() / 1e9
    print(json.dumps({
        "phase": "shard_state", "shard_seconds": round(t_shard, 1),
        "n_local": shard.n_local, "n_halo": shard.n_halo,
        "nnz_directed": shard.nnz,
        "hbm_gb_after_init": round(alloc0, 1),
    }), flush=True)

    def sweep():
        # the pipelined engine's per-sweep compute: fused grad+LLH+line
        # search (dispatch per dtype/K), then the projected commit
        grad, llh, steps = st.fused_grad_ls_overlap(None)
        st.apply_step(grad, steps)
        return llh

    for _ in range(args.warmup):
        sweep()
    torch.cuda.synchronize()
    peak0 = torch.cuda.max_memory_allocated() / 1e9
    t0 = time.perf_counter()
    for _ in range(args.steps):
        llh = sweep()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.steps
    print(json.dumps({
        "phase": "timed", "ms_per_sweep": round(dt * 1000.0, 1),
        "edges_per_s_this_rank": shard.nnz / dt,
        "hbm_gb_peak": round(torch.cuda.max_memory_allocated() / 1e9, 1),
        "hbm_gb_peak_prewarm": round(peak0, 1),
        "llh_shard_finite": bool(torch.isfinite(llh).all().item()),
        "k": args.k, "dtype": args.dtype, "fused": st.fused_ok,
        "config": "BASELINE #5 rank0-of-8 slice, scrambled R-MAT s26 ef1.5",
    }), flush=True)


if __name__ == "__main__":
    main()
