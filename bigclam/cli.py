"""Command-line entry point: ``python -m bigclam <cmd>``.

The reference has no CLI/config surface at all (hard-coded paths and vars,
SURVEY.md §2.15); this exposes the full hyperparameter set with the
reference's defaults.

  fit        fit a K-community model on an edge list, write communities + ckpt
  select-k   model selection over the geometric K grid (v4 semantics)
  extract    re-extract communities from a checkpoint
  bench      one timed sweep loop (see also bench.py at the repo root)
"""
from __future__ import annotations

import argparse
import dataclasses
import json
import os
import sys

import numpy as np
import torch

from .config import BigClamConfig
from . import comm
from .engine.extract import extract_communities, write_communities
from .engine.model_select import select_k
from .engine.trainer import Trainer
from .io import load_graph, rmat_graph
from .utils.metrics import MetricsLogger


def _add_common(p: argparse.ArgumentParser):
    d = BigClamConfig()
    p.add_argument("--k", type=int, default=d.k)
    p.add_argument("--alpha", type=float, default=d.alpha)
    p.add_argument("--beta", type=float, default=d.beta)
    p.add_argument("--ls-steps", type=int, default=d.ls_steps)
    p.add_argument("--tol", type=float, default=d.tol)
    p.add_argument(
        "--max-sweeps", type=int, default=d.max_sweeps,
        help="sweep budget for THIS invocation (with --resume: additional "
        "sweeps on top of the checkpoint's sweep count)",
    )
    p.add_argument("--dtype", choices=["fp32", "bf16"], default=d.dtype)
    p.add_argument("--device", default=None, help="cuda|cpu (auto)")
    p.add_argument("--seed", type=int, default=d.seed)
    p.add_argument("--init", choices=["seed", "random"], default="seed")
    p.add_argument("--seed-rank-compat", action="store_true")
    p.add_argument("--checkpoint-every", type=int, default=0)
    p.add_argument("--checkpoint-dir", default=None)
    p.add_argument("--out", default=None)
    p.add_argument("--metrics", default=None, help="JSONL metrics path")
    p.add_argument("--quiet", action="store_true")


def _cfg_from(args) -> BigClamConfig:
    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    return BigClamConfig(
        k=args.k,
        alpha=args.alpha,
        beta=args.beta,
        ls_steps=args.ls_steps,
        tol=args.tol,
        max_sweeps=args.max_sweeps,
        dtype=args.dtype,
        device=device,
        seed=args.seed,
        seed_rank_compat=args.seed_rank_compat,
        checkpoint_every=args.checkpoint_every,
        checkpoint_dir=args.checkpoint_dir,
        out=args.out,
        k_min=getattr(args, "k_min", 1000),
        k_max=getattr(args, "k_max", 9000),
        k_div=getattr(args, "k_div", 100),
        k_tol=getattr(args, "k_tol", 1e-3),
    )


def _load(args):
    if args.edgelist.startswith("rmat:"):
        # rmat:<scale>:<edge_factor> synthetic graph
        _, scale, ef = args.edgelist.split(":")
        return rmat_graph(int(scale), float(ef))
    if args.edgelist.startswith("shaped:"):
        # shaped:<nodes>:<edges> — power-law graph with exactly that
        # shape (the bench/convergence configs, io/synthetic.py); named
        # presets for the headline shapes:
        #   shaped:amazon = 334863:925872, shaped:enron = 36692:183831,
        #   shaped:youtube = 1134890:2987624
        from .io import shaped_graph

        presets = {
            "amazon": (334863, 925872),
            "enron": (36692, 183831),
            "youtube": (1134890, 2987624),
        }
        parts = args.edgelist.split(":")[1:]
        if len(parts) == 1 and parts[0] in presets:
            n, e = presets[parts[0]]
        else:
            n, e = int(parts[0]), int(parts[1])
        return shaped_graph(n, e, locality=0.7, seed=42)
    return load_graph(args.edgelist)


def main(argv=None):
    ap = argparse.ArgumentParser(
        prog="bigclam",
        description="BigCLAM overlapping community detection, "
        "MI355X-native (see README.md)",
    )
    sub = ap.add_subparsers(dest="cmd", required=True)

    p_fit = sub.add_parser(
        "fit", help="fit a K-community model on an edge list, "
        "write communities + checkpoint")
    p_fit.add_argument("edgelist")
    _add_common(p_fit)
    p_fit.add_argument(
        "--resume",
        metavar="CKPT_DIR",
        default=None,
        help="resume the fit from a checkpoint directory (any world size)",
    )

    p_sel = sub.add_parser(
        "select-k", help="model selection over the geometric K grid "
        "(the reference v4 semantics)")
    p_sel.add_argument("edgelist")
    _add_common(p_sel)
    p_sel.add_argument("--k-min", dest="k_min", type=int, default=1000)
    p_sel.add_argument("--k-max", dest="k_max", type=int, default=9000)
    p_sel.add_argument("--k-div", dest="k_div", type=int, default=100)
    p_sel.add_argument("--k-tol", dest="k_tol", type=float, default=1e-3)

    p_ext = sub.add_parser(
        "extract", help="re-extract communities from a checkpoint")
    p_ext.add_argument("checkpoint_dir")
    p_ext.add_argument("edgelist")
    p_ext.add_argument("--out", required=True)

    sub.add_parser(
        "bench",
        help="flagship benchmark (forwards to bench.py)",
        add_help=False,
        description="flagship benchmark; all flags forwarded to bench.py",
    )

    if argv is None:
        argv = sys.argv[1:]
    if argv and argv[0] == "bench":
        # delegate to the repo-root bench harness (same contract the
        # scaling driver uses); works installed or from a checkout
        import runpy

        sys.argv = ["bench.py"] + list(argv[1:])
        here = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        bench = os.path.join(here, "bench.py")
        if not os.path.exists(bench):  # installed wheel: no repo-root bench.py
            print(
                "bigclam bench requires a repo checkout (bench.py at the "
                "repo root); run it from the source tree",
                file=sys.stderr,
            )
            return 2
        runpy.run_path(bench, run_name="__main__")
        return 0

    args = ap.parse_args(argv)
    rank = comm.init_distributed()

    if args.cmd == "extract":
        from .ckpt.checkpoint import load_full_F, load_meta

        g = _load(args)
        F = torch.from_numpy(load_full_F(args.checkpoint_dir))
        members = extract_communities(F, g.num_edges)
        write_communities(args.out, members, g.raw_ids)
        print(json.dumps({"communities": sum(1 for m in members if len(m))}))
        return 0

    cfg = _cfg_from(args)
    g = _load(args)
    metrics = MetricsLogger(args.metrics, rank=rank, quiet=args.quiet)

    if args.cmd == "select-k":
        out = select_k(g, cfg, metrics=metrics, init=args.init)
        if rank == 0:
            print(json.dumps(out))
        return 0

    tr = Trainer(g, cfg, metrics=metrics)
    if getattr(args, "resume", None):
        from .ckpt.checkpoint import resume as ckpt_resume

        sweep0, llh0 = ckpt_resume(args.resume, tr)
        metrics.log({"note": "resumed", "from_sweep": sweep0, "llh": llh0})
        res = tr.fit(skip_init=True, llh_old=llh0, sweep0=sweep0)
    else:
        res = tr.fit(init=args.init)
    if cfg.checkpoint_dir:  # every rank writes its own shard file
        from .ckpt.checkpoint import save_shard_checkpoint

        save_shard_checkpoint(cfg.checkpoint_dir, tr, res.sweeps, res.llh)
    if cfg.out:
        # sharded K7: per-rank device-side threshold, compacted lists to
        # rank 0 — no N×K gather
        from .engine.extract import (
            extract_communities_sharded,
            write_membership_pairs,
        )

        pairs = extract_communities_sharded(tr)
        if rank == 0:
            write_membership_pairs(cfg.out, pairs[0], pairs[1], g.raw_ids)
    if rank == 0:
        print(
            json.dumps(
                {
                    "llh": res.llh,
                    "sweeps": res.sweeps,
                    "converged": res.converged,
                    "n": g.num_nodes,
                    "edges": g.num_edges,
                    "k": cfg.k,
                }
            )
        )
    return 0


if __name__ == "__main__":
    sys.exit(main())
