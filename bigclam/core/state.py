"""Device-resident shard state: F rows, CSR, halo buffers, op dispatch.

Holds everything one rank keeps in HBM (F shard + halo section, local CSR,
sumF, launch-order array) and dispatches each op to the HIP extension on
GPU or the vectorized torch reference on CPU.

On a GPU box the HIP extension is REQUIRED — ops raise rather than silently
falling back to eager PyTorch (the native path must be the one that runs).
Set ``BIGCLAM_FORCE_TORCH_OPS=1`` to override for A/B numerics testing only.
"""
from __future__ import annotations

import os
from typing import Optional, Tuple

import numpy as np
import torch

from .. import comm
from ..config import BigClamConfig
from ..ops import reference as ref_ops
from .shard import GraphShard


def _hip_ops():
    from ..ops import hip as hip_ops  # lazy: requires built extension

    return hip_ops


class ShardState:
    def __init__(
        self,
        shard: GraphShard,
        cfg: BigClamConfig,
        device: Optional[torch.device] = None,
    ):
        self.shard = shard
        self.cfg = cfg
        if device is None:
            device = torch.device(cfg.device if torch.cuda.is_available() else "cpu")
        self.device = torch.device(device)
        self.storage_dtype = (
            torch.bfloat16 if cfg.dtype == "bf16" else torch.float32
        )
        # pad K so rows stay 16B-aligned for vector kernels (fp32: float4,
        # bf16: uint4 of 8 elements); pad columns are identically zero and
        # stay zero through every op (grad_pad = -sumF_pad + F_pad = 0,
        # clamp(0 + s*0) = 0).
        self.kp = (
            (cfg.k + 7) & ~7 if self.storage_dtype == torch.bfloat16 else (cfg.k + 3) & ~3
        )

        dev = self.device
        self.indptr = torch.from_numpy(shard.indptr).to(dev)
        self.indices = torch.from_numpy(shard.indices).to(dev)
        # launch order: degree-descending so hub blocks start first
        deg = shard.degrees()
        self.order = torch.from_numpy(
            np.argsort(-deg, kind="stable").astype(np.int32)
        ).to(dev)
        # per-edge source row (torch reference path); built lazily on CPU
        self._edge_src: Optional[torch.Tensor] = None

        # F buffer: owned rows [0, n_local) + halo rows [n_local, n_rows)
        self.F = torch.zeros(
            shard.n_rows, self.kp, device=dev, dtype=self.storage_dtype
        )
        self.sumF = torch.zeros(self.kp, device=dev, dtype=torch.float32)

        # halo plan tensors
        plan = shard.plan
        self.send_idx = torch.from_numpy(
            np.concatenate(plan.send_idx)
            if plan.total_send
            else np.empty(0, dtype=np.int64)
        ).to(dev)
        self.send_splits = plan.send_counts
        self.recv_splits = plan.recv_counts

        self.use_hip = self.device.type == "cuda" and not os.environ.get(
            "BIGCLAM_FORCE_TORCH_OPS"
        )
        if self.use_hip:
            _hip_ops().ensure_loaded()  # fail loudly if the .so is missing

    # ------------------------------------------------------------------ util
    @property
    def n_local(self) -> int:
        return self.shard.n_local

    @property
    def F_local(self) -> torch.Tensor:
        """Owned rows, padded to kp columns (kernel view)."""
        return self.F[: self.n_local]

    @property
    def F_local_k(self) -> torch.Tensor:
        """Owned rows, the true K columns (user/io view)."""
        return self.F[: self.n_local, : self.cfg.k]

    def edge_src(self) -> torch.Tensor:
        if self._edge_src is None:
            self._edge_src = ref_ops._edge_src(self.indptr)
        return self._edge_src

    # ------------------------------------------------------------- model init
    def set_local_F(self, F_local: torch.Tensor):
        """Install owned rows (true-K width) and (re)compute the global sumF."""
        self.F = torch.zeros(
            self.shard.n_rows,
            self.kp,
            device=self.device,
            dtype=self.storage_dtype,
        )
        self.F[: self.n_local, : self.cfg.k] = F_local.to(
            self.device, self.storage_dtype
        )
        self.sumF = self.F_local.float().sum(dim=0)
        comm.all_reduce_(self.sumF)

    # ---------------------------------------------------------- communication
    def halo_exchange(self):
        """C8: refresh halo rows with peers' current F rows (p2p all-to-all)."""
        if self.shard.world_size == 1 or self.shard.n_halo == 0:
            return
        send = self.F_local.index_select(0, self.send_idx)
        recv = self.F[self.n_local :]
        comm.all_to_all(recv, send, self.recv_splits, self.send_splits)

    # ------------------------------------------------------------------- ops
    def grad_llh(self) -> Tuple[torch.Tensor, torch.Tensor]:
        if self.use_hip:
            return _hip_ops().edge_grad_llh(
                self.F, self.indptr, self.indices, self.sumF, self.order, self.cfg
            )
        return ref_ops.edge_grad_llh(
            self.F,
            self.indptr,
            self.indices,
            self.sumF,
            self.cfg,
            n_local=self.n_local,
            edge_src=self.edge_src(),
        )

    def linesearch(self, grad: torch.Tensor, llh: torch.Tensor) -> torch.Tensor:
        if self.use_hip:
            return _hip_ops().linesearch(
                self.F,
                self.indptr,
                self.indices,
                self.sumF,
                grad,
                llh,
                self.order,
                self.cfg,
            )
        return ref_ops.linesearch(
            self.F,
            self.indptr,
            self.indices,
            self.sumF,
            grad,
            llh,
            self.cfg,
            n_local=self.n_local,
            edge_src=self.edge_src(),
        )

    def apply_step(self, grad: torch.Tensor, steps: torch.Tensor) -> torch.Tensor:
        """K3 commit: in-place projected F update, then refresh sumF by
        recomputing the exact column sums + all-reduce (C12).  Recomputing
        (one read pass) keeps the sumF == colsum(F) invariant exact instead
        of accumulating incremental deltas."""
        if self.use_hip:
            _hip_ops().apply_step(self.F_local, grad, steps, self.cfg)
        else:
            F_new, _ = ref_ops.apply_step(self.F_local, grad, steps, self.cfg)
            self.F[: self.n_local] = F_new
        self.sumF = self.F_local.float().sum(dim=0)
        comm.all_reduce_(self.sumF)
        return self.sumF

    def full_llh(self) -> torch.Tensor:
        """K4 + C14: global objective (fp64 scalar, all-reduced)."""
        if self.use_hip:
            llh = _hip_ops().full_llh(
                self.F, self.indptr, self.indices, self.sumF, self.order, self.cfg
            )
        else:
            llh = ref_ops.full_llh(
                self.F,
                self.indptr,
                self.indices,
                self.sumF,
                self.cfg,
                n_local=self.n_local,
                edge_src=self.edge_src(),
            )
        llh = llh.reshape(1)
        comm.all_reduce_(llh)
        return llh[0]
