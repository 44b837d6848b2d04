#!/usr/bin/env python3
"""Flagship benchmark: edges/sec per gradient sweep (BASELINE.json metric).

One "step" = one full synchronous gradient/line-search sweep over every node
(K1 grad+llh -> K2 16-candidate Armijo -> K3 commit + ΔsumF allreduce ->
halo exchange -> K4 full LLH + scalar allreduce) — exactly the reference's
``backtrackingLineSearchs`` unit of work (codes/bigclamv3-7.scala:133-204).

Default config: com-Amazon-shaped synthetic power-law graph (335k nodes /
926k undirected edges — the real dataset is a missing blob upstream),
K=5000, random-init F.  The graph is global and fixed as GPUs are added
(row-sharded) -> strong scaling.  Both compute dtypes are measured every
run: the headline number is bf16 (fp32 accumulation everywhere — grad,
sumF, LLH in fp64) and the fp32-storage number is emitted alongside in
``variants`` so neither can be read as a precision-assisted claim.

Run (driver contract):
  python bench.py --gpus N --steps K --warmup W
  torchrun --nnodes=1 --nproc-per-node N bench.py --gpus N ...
"""
from __future__ import annotations

import argparse
import json
import os
import time

import numpy as np
import torch

from bigclam import comm
from bigclam.config import BigClamConfig
from bigclam.engine.trainer import Trainer
from bigclam.io import shaped_graph
from bigclam.utils.metrics import MetricsLogger


def _time_sweeps(graph, dtype, args, rank, world, device, use_cuda):
    """Build a trainer at ``dtype`` and time ``args.steps`` pipelined
    sweeps after ``args.warmup`` untimed ones.  Returns (elapsed_s, value,
    ms_per_step) with elapsed = MAX over ranks."""
    cfg = BigClamConfig(
        k=args.k,
        dtype=dtype,
        device="cuda" if use_cuda else "cpu",
        seed=7,
        ls_steps=args.ls_steps,
    )
    tr = Trainer(
        graph, cfg, device=device, metrics=MetricsLogger(rank=rank, quiet=True)
    )
    tr.init_F("random")

    def sync():
        if use_cuda:
            torch.cuda.synchronize()

    # pipelined loop: each timed step = one full algorithmic sweep
    # (projected commit of the carried Armijo steps -> sumF allreduce ->
    # halo exchange -> fused grad+LLH+16-candidate line-search pass ->
    # scalar LLH allreduce); the post-update LLH comes from the next grad
    # pass (see engine/trainer.py).
    carry, _ = tr.prologue()
    for _ in range(args.warmup):
        carry, _, _ = tr.pipelined_sweep(carry)
    comm.barrier()
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        carry, _, _ = tr.pipelined_sweep(carry)
    comm.barrier()
    sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    el = torch.tensor(
        [elapsed],
        dtype=torch.float64,
        device=device if world > 1 and use_cuda else "cpu",
    )
    if world > 1:
        import torch.distributed as dist

        dist.all_reduce(el, op=dist.ReduceOp.MAX)
    elapsed = float(el.item())
    ms_per_step = elapsed / args.steps * 1000.0
    edges_per_sec = graph.num_directed_edges / (elapsed / args.steps)
    del tr
    if use_cuda:
        torch.cuda.empty_cache()
    return elapsed, edges_per_sec, ms_per_step


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--k", type=int, default=5000)
    ap.add_argument("--nodes", type=int, default=334863)  # com-Amazon N
    ap.add_argument("--edges", type=int, default=925872)  # com-Amazon E
    ap.add_argument("--dtype", choices=["fp32", "bf16"], default="bf16")
    ap.add_argument("--graph-seed", type=int, default=42)
    ap.add_argument("--locality", type=float, default=0.7)
    ap.add_argument("--ls-steps", type=int, default=15,
                    help="Armijo ladder depth (reference: 15 -> 16 candidates)")
    ap.add_argument("--no-variant", action="store_true",
                    help="skip the second-dtype measurement")
    args = ap.parse_args()

    rank = comm.init_distributed()
    world = comm.get_world_size()
    use_cuda = torch.cuda.is_available()
    device = torch.device("cuda") if use_cuda else torch.device("cpu")

    graph = shaped_graph(
        args.nodes, args.edges, locality=args.locality, seed=args.graph_seed
    )
    _, value, ms_per_step = _time_sweeps(
        graph, args.dtype, args, rank, world, device, use_cuda
    )
    variants = {}
    if not args.no_variant:
        other = "fp32" if args.dtype == "bf16" else "bf16"
        try:
            _, v2, ms2 = _time_sweeps(
                graph, other, args, rank, world, device, use_cuda
            )
            variants[other] = {"value": v2, "ms_per_step": ms2}
        except Exception as e:  # e.g. K above a dtype's kernel coverage
            variants[other] = {"error": str(e)[:200]}

    if rank == 0:
        print(
            json.dumps(
                {
                    "metric": "edges/sec per grad iter",
                    "value": value,
                    "unit": "edges/s",
                    "n_gpus": world,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": ms_per_step,
                    "higher_is_better": True,
                    "scaling": "strong",
                    "vs_baseline": None,  # reference publishes no numbers
                    "dtype": args.dtype,
                    "data": "synthetic",
                    "variants": variants,
                    "config": {
                        "model": "bigclam",
                        "graph": "com-Amazon-shaped synthetic power-law (Chung-Lu)",
                        "nodes": graph.num_nodes,
                        "undirected_edges": graph.num_edges,
                        "directed_edge_visits_per_sweep": graph.num_directed_edges,
                        "k": args.k,
                        "init": "random",
                        "ladder": 16,
                        "sweep": "pipelined (grad+llh fused; post-update LLH = next grad pass)",
                        "parallelism": f"dp{world} row-sharded, RCCL halo",
                    },
                }
            ),
            flush=True,
        )


if __name__ == "__main__":
    main()
