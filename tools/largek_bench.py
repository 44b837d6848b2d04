#!/usr/bin/env python3
"""Large-K dispatch measurement.

Times full pipelined sweeps of the com-Amazon-shaped graph at a large K
under each dispatch variant (env toggles read at ShardState/launch time),
printing one JSON line per config.  Run on a GPU box:

  python tools/largek_bench.py --k 25000 --steps 5 --warmup 2
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

CONFIGS = [
    # label, dtype, env overrides
    ("bf16-default", "bf16", {}),
    ("bf16-mfma-all", "bf16", {"BIGCLAM_MFMA_DEG": "1"}),
    ("bf16-sep-chunked-k2staged", "bf16",
     {"BIGCLAM_MFMA_DEG": "0", "BIGCLAM_NO_FUSED": "1",
      "BIGCLAM_K1_CHUNKED": "1"}),
    ("bf16-sep-chunked-k2nostage", "bf16",
     {"BIGCLAM_MFMA_DEG": "0", "BIGCLAM_K2_NOSTAGE": "1",
      "BIGCLAM_NO_FUSED": "1", "BIGCLAM_K1_CHUNKED": "1"}),
    ("fp32-default", "fp32", {}),
    ("fp32-sep-chunked-k2tiled", "fp32",
     {"BIGCLAM_K2_TILED": "1", "BIGCLAM_NO_FUSED": "1",
      "BIGCLAM_K1_CHUNKED": "1"}),
    ("fp32-sep-onepass-k1", "fp32",
     {"BIGCLAM_K1_CHUNKED": "0", "BIGCLAM_NO_FUSED": "1"}),
]

ENV_KEYS = [
    "BIGCLAM_MFMA_DEG", "BIGCLAM_K2_NOSTAGE", "BIGCLAM_K2_TILED",
    "BIGCLAM_K1_CHUNKED", "BIGCLAM_NO_FUSED",
]


def dev_init(tr, k):
    """Device-side random init (avoids a 33 GB host array at K=25000)."""
    st = tr.state
    torch.manual_seed(7)
    st.F.zero_()
    n = st.n_local
    blk = 65536
    scale = 1.0 / (k ** 0.5)
    for i in range(0, n, blk):
        m = min(blk, n - i)
        st.F[i : i + blk, :k] = (
            torch.rand(m, k, device=st.device) * scale
        ).to(st.storage_dtype)
    st.sumF = st.F[:n].float().sum(0)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--k", type=int, default=25000)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--nodes", type=int, default=334863)
    ap.add_argument("--edges", type=int, default=925872)
    ap.add_argument("--only", default=None, help="substring filter on labels")
    args = ap.parse_args()

    assert torch.cuda.is_available()
    graph = shaped_graph(args.nodes, args.edges, locality=0.7, seed=42)
    dev = torch.device("cuda")

    for label, dtype, env in CONFIGS:
        if args.only and args.only not in label:
            continue
        for key in ENV_KEYS:
            os.environ.pop(key, None)
        os.environ.update(env)
        try:
            cfg = BigClamConfig(k=args.k, dtype=dtype, device="cuda", seed=7)
            tr = Trainer(graph, cfg, rank=0, world_size=1, device=dev,
                         metrics=MetricsLogger(rank=0, quiet=True))
            dev_init(tr, args.k)
            carry, _ = tr.prologue()
            for _ in range(args.warmup):
                carry, _, _ = tr.pipelined_sweep(carry)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(args.steps):
                carry, llh, _ = tr.pipelined_sweep(carry)
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / args.steps
            print(json.dumps({
                "label": label, "dtype": dtype, "k": args.k,
                "ms_per_sweep": dt * 1000.0,
                "edges_per_s": graph.num_directed_edges / dt,
                "llh": llh, "fused": tr.state.fused_ok,
                "env": env,
            }), flush=True)
            del tr, carry
        except Exception as e:
            print(json.dumps({"label": label, "error": str(e)[:300]}),
                  flush=True)
        torch.cuda.empty_cache()
    for key in ENV_KEYS:
        os.environ.pop(key, None)


if __name__ == "__main__":
    main()
