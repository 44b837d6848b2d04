#!/usr/bin/env python3
"""Sparse-vs-dense sweep profiling at a converged state.

Converges the headline config on the dense path (10 sweeps), then times
N pipelined sweeps in the requested mode.  Run under rocprofv3 --stats
to attribute kernel time:

  rocprofv3 --kernel-trace --stats -d out -- python tools/sparse_prof.py --mode sparse
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from bigclam.config import BigClamConfig  # noqa: E402
from bigclam.engine.trainer import Trainer  # noqa: E402
from bigclam.io import shaped_graph  # noqa: E402
from bigclam.utils.metrics import MetricsLogger  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--mode", choices=["sparse", "dense"], default="sparse")
    ap.add_argument("--k", type=int, default=5000)
    ap.add_argument("--dtype", default="bf16")
    ap.add_argument("--nodes", type=int, default=334863)
    ap.add_argument("--edges", type=int, default=925872)
    ap.add_argument("--converge", type=int, default=10)
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--torchprof", action="store_true",
                    help="print a torch.profiler op table for 3 sweeps")
    args = ap.parse_args()
    assert torch.cuda.is_available()

    g = shaped_graph(args.nodes, args.edges, locality=0.7, seed=42)
    cfg = BigClamConfig(k=args.k, dtype=args.dtype, device="cuda", seed=7,
                        max_sweeps=args.converge, tol=0.0)
    os.environ["BIGCLAM_SPARSE"] = "0"
    tr = Trainer(g, cfg, rank=0, world_size=1, device=torch.device("cuda"),
                 metrics=MetricsLogger(rank=0, quiet=True))
    tr.fit(init="seed")
    nnz = float((tr.state.F_local_k != 0).float().mean().item())

    os.environ["BIGCLAM_SPARSE"] = "1" if args.mode == "sparse" else "0"
    carry, _ = tr.prologue()
    for _ in range(3):
        carry, _, _ = tr.pipelined_sweep(carry)
    torch.cuda.synchronize()
    if args.torchprof:
        from torch.profiler import ProfilerActivity, profile

        with profile(
            activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]
        ) as prof:
            for _ in range(3):
                carry, _, _ = tr.pipelined_sweep(carry)
            torch.cuda.synchronize()
        print(prof.key_averages().table(
            sort_by="cuda_time_total", row_limit=25))
    t0 = time.perf_counter()
    for _ in range(args.steps):
        carry, llh, _ = tr.pipelined_sweep(carry)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.steps
    routed = None
    if carry[2] is not None:
        routed = int(carry[2]["order"].numel())
    print(json.dumps({
        "mode": args.mode, "k": args.k, "dtype": args.dtype,
        "ms_per_sweep": round(dt * 1000.0, 3),
        "edges_per_s": g.num_directed_edges / dt,
        "f_nnz_frac": round(nnz, 5), "routed_nodes": routed,
        "n_local": tr.state.n_local, "llh": llh,
    }), flush=True)


if __name__ == "__main__":
    main()
