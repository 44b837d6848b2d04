"""F-matrix checkpoint / resume (SURVEY.md §5 checkpoint row).

The reference has no checkpointing — a crash loses the fit (the model lives
only as an in-memory RDD).  Our format (defined here, documented):

  <dir>/meta.json            {n, k, dtype, sweep, llh, world_size, bounds}
  <dir>/raw_ids.npy          int64 [N] internal->raw id map
  <dir>/sumF.npy             fp32 [K]
  <dir>/F_rank<r>.npy        fp32/bf16-as-uint16 [n_r, K] row-major shard

Shards are written per-rank (parallel, no gather) and can be merged or
re-sharded on load: resume works at any world size.
"""
from __future__ import annotations

import json
import os
from typing import Optional

import numpy as np
import torch


def save_shard_checkpoint(dirpath: str, trainer, sweep: int, llh: float):
    os.makedirs(dirpath, exist_ok=True)
    st = trainer.state
    r = trainer.rank
    F = st.F_local_k.contiguous()
    if F.dtype == torch.bfloat16:
        arr = F.view(torch.uint16).cpu().numpy()
        dtype = "bf16"
    else:
        arr = F.float().cpu().numpy()
        dtype = "fp32"
    np.save(os.path.join(dirpath, f"F_rank{r}.npy"), arr)
    if r == 0:
        np.save(os.path.join(dirpath, "raw_ids.npy"), trainer.graph.raw_ids)
        np.save(
            os.path.join(dirpath, "sumF.npy"),
            st.sumF[: trainer.cfg.k].cpu().numpy(),
        )
        meta = {
            "n": trainer.graph.num_nodes,
            "k": trainer.cfg.k,
            "dtype": dtype,
            "sweep": sweep,
            "llh": llh,
            "world_size": trainer.world_size,
            "bounds": [int(b) for b in trainer.bounds],
        }
        with open(os.path.join(dirpath, "meta.json"), "w") as f:
            json.dump(meta, f, indent=2)


def load_meta(dirpath: str) -> dict:
    with open(os.path.join(dirpath, "meta.json")) as f:
        return json.load(f)


def load_full_F(dirpath: str) -> np.ndarray:
    """Merge all shard files into the full [N, K] fp32 matrix."""
    meta = load_meta(dirpath)
    parts = []
    for r in range(meta["world_size"]):
        arr = np.load(os.path.join(dirpath, f"F_rank{r}.npy"))
        if meta["dtype"] == "bf16":
            arr = (
                torch.from_numpy(arr).view(torch.bfloat16).float().numpy()
            )
        parts.append(arr)
    return np.concatenate(parts, axis=0)


def load_F_slice(dirpath: str, start: int, stop: int) -> np.ndarray:
    """Global rows [start, stop) as fp32, reading only the shard files that
    overlap (memory-mapped) — never materializes the full N×K matrix."""
    meta = load_meta(dirpath)
    bounds = meta["bounds"]
    out = np.empty((stop - start, meta["k"]), dtype=np.float32)
    for r in range(meta["world_size"]):
        rs, re_ = int(bounds[r]), int(bounds[r + 1])
        lo, hi = max(start, rs), min(stop, re_)
        if lo >= hi:
            continue
        arr = np.load(
            os.path.join(dirpath, f"F_rank{r}.npy"), mmap_mode="r"
        )[lo - rs : hi - rs]
        if meta["dtype"] == "bf16":
            # copy: the mmap slice is read-only and non-contiguous rows
            # can't be viewed as bf16 anyway
            arr = (
                torch.from_numpy(np.array(arr, copy=True))
                .view(torch.bfloat16)
                .float()
                .numpy()
            )
        out[lo - start : hi - start] = arr
    return out


def resume(dirpath: str, trainer) -> tuple:
    """Load checkpoint rows into trainer's shard (any world size; each rank
    reads only its own slice).  Returns ``(sweep, llh)`` so the training
    loop can continue with the saved objective as ``llh_old`` (same
    convergence behavior as an uninterrupted run)."""
    meta = load_meta(dirpath)
    s = trainer.shard
    sl = load_F_slice(dirpath, s.start, s.stop)
    trainer.state.set_local_F(torch.from_numpy(sl))
    return int(meta["sweep"]), float(meta["llh"])
