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


class _CastOnWait:
    """Work-handle wrapper that copies the fp32 staging buffer into the
    bf16 halo section after the collective completes (gloo bf16 path)."""

    def __init__(self, work, src32: torch.Tensor, dst: torch.Tensor):
        self._work = work
        self._src32 = src32
        self._dst = dst

    def wait(self):
        if self._work is not None:
            self._work.wait()
        self._dst.copy_(self._src32)


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
        order_np = np.argsort(-deg, kind="stable").astype(np.int32)
        self.order = torch.from_numpy(order_np).to(dev)
        # interior/boundary split for halo-compute overlap: a node is
        # boundary iff any neighbor lives in the halo section (local row
        # index >= n_local).  K1 on interior nodes can run while the halo
        # all_to_all is in flight (halo fractions are large on power-law
        # graphs: Email-Enron ws=8 halo = 1.95x N).
        if shard.world_size > 1 and shard.n_halo > 0:
            remote = shard.indices >= shard.n_local
            # exact segment sums even for empty rows: cumulative count of
            # remote neighbors, differenced at row boundaries (reduceat
            # with clipped indptr misattributes the last edge when
            # trailing degree-0 rows exist)
            cs = np.concatenate([[0], np.cumsum(remote)])
            row_has_halo = (cs[shard.indptr[1:]] - cs[shard.indptr[:-1]]) > 0
            bnd = order_np[row_has_halo[order_np]]
            interior = order_np[~row_has_halo[order_np]]
            self.order_interior = torch.from_numpy(
                np.ascontiguousarray(interior)
            ).to(dev)
            self.order_boundary = torch.from_numpy(
                np.ascontiguousarray(bnd)
            ).to(dev)
        else:
            self.order_interior = self.order
            self.order_boundary = self.order[:0]
            interior = order_np
            bnd = order_np[:0]
        # MFMA split: each launch list is degree-descending, so the nodes
        # taking the MFMA phase-B kernel (deg >= threshold) are a prefix;
        # precompute the split points.  BIGCLAM_MFMA_DEG overrides
        # (0 = all nodes on the direct kernel).  Default is K-dependent,
        # measured in profiles/r01_kernel_opt_log.md and
        # r02_largek_dispatch.md: at kp <= 8192 (NSLOT<=4, spill-free)
        # MFMA-all wins at EVERY degree (16.4 vs 22.6 ms on the com-Amazon
        # CSR, mean degree 5.5); at 8192 < kp <= 16384 the NSLOT=8
        # template spills 20 B/lane and the direct kernel measures faster
        # (com-Youtube K=8385: 149 vs 159 ms); above 16384 the separate
        # chunked-K1 + unstaged-K2 path beats the 1-block/CU MFMA fused
        # kernel (145 vs 236 ms at K=25000 bf16), so MFMA is off there.
        mfma_all = self.storage_dtype == torch.bfloat16 and self.kp <= 8192
        thr = int(os.environ.get("BIGCLAM_MFMA_DEG", "1" if mfma_all else "0"))
        if thr == 1:
            # "all nodes": include any degree-0 rows too (the MFMA kernel
            # handles them; the direct kernel rejects K > 16384)
            self.n_mfma = int(order_np.size)
            self.n_mfma_interior = int(interior.size)
            self.n_mfma_boundary = int(bnd.size)
        elif thr > 0:
            self.n_mfma = int((deg[order_np] >= thr).sum())
            self.n_mfma_interior = int((deg[interior] >= thr).sum())
            self.n_mfma_boundary = int((deg[bnd] >= thr).sum())
        else:
            self.n_mfma = self.n_mfma_interior = self.n_mfma_boundary = 0
        self._halo_send: Optional[torch.Tensor] = None
        self._colsum_partials: Optional[torch.Tensor] = None
        self._sp_soffset: Optional[torch.Tensor] = None
        self._indices64: Optional[torch.Tensor] = None
        self._order64: Optional[torch.Tensor] = None
        self._last_nnz: Optional[torch.Tensor] = None  # from KAF scount
        self._sp_scount: Optional[torch.Tensor] = None
        self._sp_sidx: Optional[torch.Tensor] = None
        self._sp_sval: Optional[torch.Tensor] = None
        self._sp_gidx: Optional[torch.Tensor] = None
        self._sp_gval: Optional[torch.Tensor] = None
        self._sp_cap: Optional[int] = None
        self._dirty: Optional[torch.Tensor] = None  # uint8 [n_rows]
        self._kaf_valid = False  # persistent support lists match F
        self._side_stream = None  # hub-remainder overlap (sparse path)
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
            # no K cap: shapes beyond the fused kernels' LDS coverage
            # (fp32 > 8192, bf16 > 16384) take the chunked-K1 + K2 path
            # (ops/hip.py edge_grad_llh_chunked; measured dispatch table
            # in profiles/r02_largek_dispatch.md)

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
        self._kaf_valid = False
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
    def _halo_compress_on(self) -> bool:
        """Compressed halo: send (col, val) nonzeros instead of dense
        rows.  Engages when F is sparse enough that the packed stream
        (8 B/nnz) beats the dense row bytes — the xGMI volume drops with
        the density (~250x at the converged headline config).

        The decision MUST be identical on every rank (the two paths run
        different collectives), so it comes from an allreduce of
        (have-counts, nnz) — one tiny extra collective per sweep, only
        at ws > 1.  BIGCLAM_HALO_COMPRESS=1/0 forces (same env on every
        rank under torchrun)."""
        env = os.environ.get("BIGCLAM_HALO_COMPRESS")
        if env is not None:
            return env != "0"
        flag = torch.zeros(2, dtype=torch.float64, device=self.device)
        if self._last_nnz is not None:
            flag[0] = 1.0
            flag[1] = float(self._last_nnz.item())
        comm.all_reduce_(flag)
        if float(flag[0].item()) < self.shard.world_size:
            return False  # some rank has no fresh counts: stay dense
        density = float(flag[1].item()) / max(
            1, self.shard.n_total * self.kp
        )
        # transport = ~8 B/nnz vs kp*esize dense; engage with ~2x margin
        return density < 0.1 * self.F.element_size()

    def _halo_exchange_compressed(self):
        """C8 compressed: two all-to-alls (per-row nnz counts, then the
        packed col/val streams) + a scatter into the zeroed halo section.
        Synchronous (the volumes are tiny once this path engages)."""
        st = self.shard
        dev = self.device
        Fs = self.F_local.index_select(0, self.send_idx).float()
        mask = Fs != 0
        c_send = mask.sum(1).to(torch.int64)
        c_recv = torch.empty(st.n_halo, device=dev, dtype=torch.int64)
        comm.all_to_all(
            c_recv, c_send, self.recv_splits, self.send_splits
        )
        nz = mask.nonzero(as_tuple=False)  # row-major: rows grouped by peer
        cols = nz[:, 1].to(torch.int32)
        vals = Fs[nz[:, 0], nz[:, 1]]
        # per-peer element splits (host): segment sums of the row counts
        cs_s = torch.cat(
            [torch.zeros(1, dtype=torch.int64, device=dev),
             torch.cumsum(c_send, 0)]
        )
        cs_r = torch.cat(
            [torch.zeros(1, dtype=torch.int64, device=dev),
             torch.cumsum(c_recv, 0)]
        )
        sb = np.concatenate([[0], np.cumsum(self.send_splits)])
        rb = np.concatenate([[0], np.cumsum(self.recv_splits)])
        vs = (cs_s[sb[1:]] - cs_s[sb[:-1]]).cpu().tolist()
        vr_t = cs_r[rb[1:]] - cs_r[rb[:-1]]
        vr = vr_t.cpu().tolist()
        total_r = int(sum(vr))
        cols_r = torch.empty(total_r, device=dev, dtype=torch.int32)
        vals_r = torch.empty(total_r, device=dev, dtype=torch.float32)
        comm.all_to_all(cols_r, cols, vr, vs)
        comm.all_to_all(vals_r, vals, vr, vs)
        # unpack: zero the halo section, scatter (row, col) = val
        halo = self.F[self.n_local :]
        halo.zero_()
        rows_r = torch.repeat_interleave(
            torch.arange(st.n_halo, device=dev, dtype=torch.int64), c_recv
        )
        halo[rows_r, cols_r.to(torch.int64)] = vals_r.to(self.storage_dtype)
        return None

    def halo_exchange(self, async_op: bool = False):
        """C8: refresh halo rows with peers' current F rows (p2p all-to-all).

        With ``async_op`` returns the Work handle (None when nothing to
        exchange); the send buffer stays referenced until the next call.
        When F is sparse (tracked by the KAF counts), the compressed
        variant sends only the nonzeros.
        """
        if self.shard.world_size == 1 or self.shard.n_halo == 0:
            return None
        if self._halo_compress_on():
            return self._halo_exchange_compressed()
        send = self.F_local.index_select(0, self.send_idx)
        recv = self.F[self.n_local :]
        if self.storage_dtype == torch.bfloat16 and comm.backend() == "gloo":
            # gloo has no bf16 collectives: route the exchange through fp32
            # (RCCL takes bf16 natively — no cast on the GPU path)
            send = send.float()
            recv32 = torch.empty_like(recv, dtype=torch.float32)
            self._halo_send = send
            work = comm.all_to_all(
                recv32, send, self.recv_splits, self.send_splits,
                async_op=async_op,
            )
            if async_op:
                return _CastOnWait(work, recv32, recv)
            recv.copy_(recv32)
            return None
        self._halo_send = send  # keep alive while the collective is in flight
        return comm.all_to_all(
            recv, send, self.recv_splits, self.send_splits, async_op=async_op
        )

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

    @property
    def fused_ok(self) -> bool:
        """The fused K1+K2 kernel covers fp32 rows up to K=8192 and bf16
        rows up to K=16384 (LDS holds grad + fu); other shapes use the
        separate kernels.  BIGCLAM_NO_FUSED=1 forces the separate path
        (dispatch measurements)."""
        if not self.use_hip:
            return False
        if os.environ.get("BIGCLAM_NO_FUSED"):
            return False
        if self.storage_dtype == torch.float32:
            return self.kp <= 8192
        if self.kp <= 16384:
            return True
        # beyond the direct kernel's K cap, the MFMA kernel (fu via LDS
        # only, K*6 bytes <= 160 KB) must cover EVERY node
        return self.kp <= 26000 and self.n_mfma == int(self.order.numel())

    def fused_grad_ls_overlap(
        self, halo_work
    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        """(grad, llh_nodes, best_step) of the current post-halo state —
        the fused KF kernel where supported (with interior/boundary halo
        overlap), else K1 (+overlap) followed by K2."""
        if self.fused_ok:
            ops = _hip_ops()
            n, kp = self.n_local, self.F.shape[1]
            grad = torch.empty(
                n, kp, device=self.device, dtype=torch.float32
            )
            llh = torch.empty(n, device=self.device, dtype=torch.float64)
            best = torch.empty(n, device=self.device, dtype=torch.float32)
            out = (grad, llh, best)
            if (
                halo_work is not None
                and self.order_boundary.numel() > 0
                and self.order_interior.numel() > 0
            ):
                ops.fused_grad_ls(
                    self.F, self.indptr, self.indices, self.sumF,
                    self.order_interior, self.cfg, out=out,
                    n_mfma=self.n_mfma_interior,
                )
                halo_work.wait()
                ops.fused_grad_ls(
                    self.F, self.indptr, self.indices, self.sumF,
                    self.order_boundary, self.cfg, out=out,
                    n_mfma=self.n_mfma_boundary,
                )
            else:
                if halo_work is not None:
                    halo_work.wait()
                ops.fused_grad_ls(
                    self.F, self.indptr, self.indices, self.sumF,
                    self.order, self.cfg, out=out,
                    n_mfma=self.n_mfma,
                )
            return grad, llh, best
        grad, llh = self.grad_llh_overlap(halo_work)
        return grad, llh, self.linesearch(grad, llh)

    # ------------------------------------------------- sparse-adaptive path
    @property
    def sparse_cap(self) -> int:
        """Active-set bound under which a node routes to the KFS sparse
        kernel.  kp/4 keeps the compact work well under the dense
        kernels' while letting moderate-degree nodes route
        (bound ≈ deg·density·K); hubs always stay on the dense path.
        At large K the cap additionally shrinks to what fits next to the
        dense-K LDS accumulator (gacc K·4 + pools cap·12/16 + bitmap
        within the 160 KB workgroup LDS, ~4 KB margin for statics)."""
        env = os.environ.get("BIGCLAM_SPARSE_CAP")
        if env is not None:
            return int(env)
        per = 16 if self.storage_dtype == torch.bfloat16 else 20
        nw4 = ((self.kp + 31) // 32) * 4
        # KFS v10's compact accumulator: LDS = bitmap + cap*per.  Target
        # ~32 KB -> 4-5 blocks/CU: a measured cap sweep at K=25000 put
        # the 28.7 KB / 5-block point at 14.2 ms/sweep vs 21.5 at the
        # 78 KB / 2-block sizing — occupancy beats routing the last few
        # hub-adjacent nodes.  (The K=5000 headline stays at cap = K/4,
        # 20.6 KB, 7 blocks/CU.)
        lds_cap = (32 * 1024 - 1024 - nw4) // per
        return max(0, min(self.kp // 4, lds_cap) & ~7)

    @property
    def sparse_allowed(self) -> bool:
        """Active-column sweep (docs/sparse_sweep_design.md): exact
        per-node routing by active-set bound; pays off once F sparsifies
        (measured 0.4-4% converged density at the headline configs).
        KFS's LDS cost is bitmap + compact-set arrays (occupancy-aware
        cap), so there is no hard K bound from the kernel itself; the
        dense-node remainder runs the fused kernel where it covers, else
        the separate chunked path — both subset-capable.
        The cap-strided support/grad pools cost
        ~16·n_rows·cap bytes — at HBM-filling shapes (e.g. the 100M-edge
        R-MAT ws=8 slice: 266 GB of pools next to a 207 GB model) that
        budget doesn't exist, so the path stays off there.
        BIGCLAM_SPARSE=0 disables."""
        if (
            not self.use_hip
            or self.sparse_cap < 64
            or os.environ.get("BIGCLAM_SPARSE", "1") == "0"
        ):
            return False
        pool_bytes = 16 * self.F.shape[0] * self.sparse_cap
        total = torch.cuda.get_device_properties(self.device).total_memory
        return pool_bytes < 0.15 * total

    @staticmethod
    def sparse_bounds(scount: torch.Tensor, indptr: torch.Tensor,
                      indices64: torch.Tensor, n_local: int):
        """Exact per-node active-set bound ``s_u + Σ_{v∈N(u)} s_v`` and the
        per-edge support prefix (the kernel's staging offsets).  Pure
        tensor math — works on CPU for the ws>1 routing tests even though
        the kernels are GPU-only."""
        sc = scount.to(torch.int64)
        sv_edges = sc[indices64]
        cs = torch.cat(
            [torch.zeros(1, device=scount.device, dtype=torch.int64),
             torch.cumsum(sv_edges, 0)]
        )
        bound = sc[:n_local] + cs[indptr[1:]] - cs[indptr[:-1]]
        return bound, cs

    def grad_ls_auto(self, halo_work):
        """Per-sweep adaptive dispatch: route nodes whose active-set bound
        (own support + sum of neighbor supports, exact upper bound on
        |S_u|) fits under ``sparse_cap`` to the fused sparse kernel
        (KFS), the rest to the dense path (fused kernel, or chunked
        K1 + K2 subsets above its caps).  Returns
        (grad, llh_nodes, best, sparse_pack|None)."""
        if not self.sparse_allowed:
            self._last_nnz = None
            g, l, b = self.fused_grad_ls_overlap(halo_work)
            return g, l, b, None
        try:
            return self._grad_ls_sparse(halo_work)
        except RuntimeError as e:
            if halo_work is not None:
                try:  # Work.wait() is idempotent; make sure the halo
                    halo_work.wait()  # landed before the dense fallback
                except Exception:
                    pass
            # The sparse path is an OPTIMIZATION of the dense sweep (same
            # math); if it ever fails at runtime (e.g. an untested shape
            # on a fresh topology), degrade loudly to the dense path for
            # the rest of the session instead of killing an unattended
            # multi-GPU run.  Kernel-load failures still raise (they hit
            # the dense path identically).
            import warnings

            warnings.warn(
                f"sparse sweep failed ({e}); falling back to the dense "
                "path for this session (BIGCLAM_SPARSE=0)"
            )
            os.environ["BIGCLAM_SPARSE"] = "0"
            self._last_nnz = None
            g, l, b = self.fused_grad_ls_overlap(None)
            return g, l, b, None

    def _grad_ls_sparse(self, halo_work):
        ops = _hip_ops()
        # NOTE: the halo wait happens mid-KAF below — the local-section
        # scan overlaps the in-flight all_to_all (disjoint rows)
        dev = self.device
        n_rows = self.F.shape[0]
        cap = self.sparse_cap
        # Single host sync per sweep (the boolean order split below);
        # pools use a fixed per-row stride of `cap` so their sizes are
        # shape-derived, not data-dependent.
        if self._sp_soffset is None or self._sp_cap != cap:
            # per-STATE persistent buffers: the incremental lists must
            # survive across sweeps of THIS state without another state
            # (e.g. select-k's next-K trainer) clobbering them — a
            # shared module pool would.  Rebuilt if the cap env changes.
            self._sp_cap = cap
            self._sp_soffset = (
                torch.arange(n_rows, device=dev, dtype=torch.int64) * cap
            )
            self._indices64 = self.indices.long()
            self._order64 = self.order.long()
            self._sp_scount = torch.empty(
                n_rows, device=dev, dtype=torch.int32
            )
            self._sp_sidx = torch.empty(
                n_rows * cap, device=dev, dtype=torch.int32
            )
            self._sp_sval = torch.empty(
                n_rows * cap, device=dev, dtype=torch.float32
            )
            self._sp_gidx = torch.empty(
                self.n_local * cap, device=dev, dtype=torch.int32
            )
            self._sp_gval = torch.empty(
                self.n_local * cap, device=dev, dtype=torch.float32
            )
            self._dirty = torch.ones(n_rows, device=dev, dtype=torch.uint8)
            self._kaf_valid = False
        soffset = self._sp_soffset
        scount = self._sp_scount
        sidx = self._sp_sidx
        sval = self._sp_sval
        # single KAF pass: counts AND lists for rows <= cap.  Incremental:
        # rows unchanged since the last commit keep their persistent
        # entries (the commit marks accepted rows dirty; halo rows are
        # always dirty — refreshed by the exchange every sweep).
        # ws > 1: the LOCAL section scans while the halo all_to_all is in
        # flight (the exchange writes ONLY the halo rows — disjoint), then
        # the halo section scans after the wait.
        if self._kaf_valid:
            dirty = self._dirty
        else:
            dirty = torch.empty(0, device=dev, dtype=torch.uint8)
        ext = ops.ensure_loaded()
        n_loc = self.n_local
        if halo_work is not None and n_rows > n_loc:
            ext.sparse_support(
                self.F[:n_loc], soffset[:n_loc], scount[:n_loc], sidx,
                sval, cap, True,
                dirty[:n_loc] if dirty.numel() else dirty,
            )
            halo_work.wait()
            halo_work = None
            ext.sparse_support(
                self.F[n_loc:], soffset[n_loc:], scount[n_loc:], sidx,
                sval, cap, True,
                torch.empty(0, device=dev, dtype=torch.uint8),
            )
        else:
            if halo_work is not None:
                halo_work.wait()
                halo_work = None
            ext.sparse_support(
                self.F, soffset, scount, sidx, sval, cap, True, dirty
            )
        self._kaf_valid = True
        self._last_nnz = scount[: self.n_local].sum()
        bound, cs = self.sparse_bounds(
            scount, self.indptr, self._indices64, self.n_local
        )
        is_sparse = bound <= cap
        om = is_sparse[self._order64]
        order_s = self.order[om].contiguous()  # <- the one host sync
        n_s = int(order_s.numel())
        # engage only when a real fraction of nodes routes: at low K /
        # moderate density the bound exceeds cap for most nodes and the
        # routing overhead loses (Enron-shaped K=500: 16% routed, sparse
        # 1.28 vs dense 1.09 ms — r17 measurement)
        if n_s < max(64, self.n_local // 4):
            g, l, b = self.fused_grad_ls_overlap(None)
            return g, l, b, None
        order_d = self.order[~om].contiguous()
        goffset = torch.arange(n_s, device=dev, dtype=torch.int64) * cap
        n, kp = self.n_local, self.kp
        grad = torch.empty(n, kp, device=dev, dtype=torch.float32)
        llh = torch.empty(n, device=dev, dtype=torch.float64)
        best = torch.empty(n, device=dev, dtype=torch.float32)
        n_d = int(order_d.numel())
        if n_d:
            # the hub remainder is a FEW blocks (often one giant hub
            # whose serial block latency is ~1 ms) — run it on a side
            # stream so it overlaps the 300k-block KFS launch below.
            # Independent: both only READ F/sumF; grad/llh/best writes
            # are row-disjoint by the routing split.
            if self._side_stream is None:
                self._side_stream = torch.cuda.Stream(device=dev)
            self._side_stream.wait_stream(torch.cuda.current_stream(dev))
            with torch.cuda.stream(self._side_stream):
                if self.fused_ok:
                    n_mfma_d = (
                        n_d if self.n_mfma == int(self.order.numel()) else 0
                    )
                    ops.fused_grad_ls(
                        self.F, self.indptr, self.indices, self.sumF,
                        order_d, self.cfg, out=(grad, llh, best),
                        n_mfma=n_mfma_d,
                    )
                else:
                    # large-K hub remainder: chunked K1 + K2 subsets
                    ops.edge_grad_llh(
                        self.F, self.indptr, self.indices, self.sumF,
                        order_d, self.cfg, out=(grad, llh),
                    )
                    ops.linesearch(
                        self.F, self.indptr, self.indices, self.sumF, grad,
                        llh, order_d, self.cfg, out=best,
                    )
        # grad pools sized at the n_local bound so the backing buffers
        # allocate ONCE (growing them as more nodes route caused 60-90 ms
        # hipMalloc spikes mid-fit)
        pack = ops.sparse_sweep_part(
            self.F, self.indptr, self.indices, self.sumF, order_s,
            soffset, sidx, sval, scount, cs, goffset,
            self.n_local * cap, cap, llh, best, self.cfg,
            state_pools=(self._sp_gidx, self._sp_gval),
        )
        if n_d:
            torch.cuda.current_stream(dev).wait_stream(self._side_stream)
        pack["best"] = best
        # dense commit must skip sparse rows (their grad rows are unset)
        steps_dense = best.clone()
        steps_dense[order_s.long()] = 0.0
        pack["steps_dense"] = steps_dense
        return grad, llh, best, pack

    def apply_commit(self, grad, steps, pack):
        """Commit a sweep's accepted steps: K3S for sparse-routed rows
        (first), then the dense commit + exact column-sum refresh (which
        reads the POST-commit F for every row)."""
        accepted = steps  # full best: sparse + dense accepted rows
        if pack is not None:
            # K3S commits AND rewrites the sparse rows' support lists,
            # so only DENSE-committed rows need a KAF rescan
            ops = _hip_ops()
            ops.sparse_commit(self.F_local, pack, pack["best"], self,
                              self.cfg)
            steps = pack["steps_dense"]
            if self._kaf_valid and os.environ.get(
                "BIGCLAM_SPARSE_COLSUM", "1"
            ) != "0":
                # list-based sumF refresh: the persistent lists are
                # current for everything except the dense-committed
                # (dirty) and over-cap rows, which the kernel reads
                # dense — exact for the current F, ~60x fewer bytes
                ops.apply_step(self.F_local, grad, steps, self.cfg)
                self._dirty[: self.n_local] = (steps > 0).to(torch.uint8)
                ns = (self.n_local + 511) // 512
                if (
                    self._colsum_partials is None
                    or self._colsum_partials.shape[0] != ns
                ):
                    self._colsum_partials = torch.empty(
                        ns, self.kp, device=self.device,
                        dtype=torch.float32,
                    )
                ops.ensure_loaded().sparse_colsum(
                    self.F_local, self._sp_soffset, self._sp_sidx,
                    self._sp_sval, self._sp_scount, self._dirty,
                    self._sp_cap, self._colsum_partials,
                )
                self.sumF = self._colsum_partials.sum(dim=0)
                comm.all_reduce_(self.sumF)
                return
        self.apply_step(grad, steps)
        # incremental KAF bookkeeping (halo rows stay always-dirty)
        if self._dirty is not None:
            dirty_rows = steps if pack is not None else accepted
            self._dirty[: self.n_local] = (dirty_rows > 0).to(torch.uint8)
            self._kaf_valid = True

    def grad_llh_overlap(self, halo_work) -> Tuple[torch.Tensor, torch.Tensor]:
        """K1 overlapped with the in-flight halo exchange: interior nodes
        (no halo neighbors) run while the all_to_all completes, boundary
        nodes after ``halo_work.wait()``.  Falls back to wait-then-full
        when there is nothing to overlap (world 1, or the CPU reference
        path, which is vectorized over all nodes at once)."""
        if (
            halo_work is not None
            and self.use_hip
            and self.order_boundary.numel() > 0
            and self.order_interior.numel() > 0
        ):
            ops = _hip_ops()
            grad = torch.empty(
                self.n_local, self.F.shape[1], device=self.device,
                dtype=torch.float32,
            )
            llh = torch.empty(
                self.n_local, device=self.device, dtype=torch.float64
            )
            ops.edge_grad_llh(
                self.F, self.indptr, self.indices, self.sumF,
                self.order_interior, self.cfg, out=(grad, llh),
            )
            halo_work.wait()
            ops.edge_grad_llh(
                self.F, self.indptr, self.indices, self.sumF,
                self.order_boundary, self.cfg, out=(grad, llh),
            )
            return grad, llh
        if halo_work is not None:
            halo_work.wait()
        return self.grad_llh()

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
        self._kaf_valid = False  # re-validated by apply_commit
        if self.use_hip and self.storage_dtype == torch.bfloat16:
            # fused commit + column sums: one pass over F (the separate
            # flow re-read F via an fp32 materialization, ~3.5 ms/sweep
            # at com-Amazon K=5000)
            ns = (self.n_local + 511) // 512
            if self._colsum_partials is None or (
                self._colsum_partials.shape[0] != ns
            ):
                self._colsum_partials = torch.empty(
                    ns, self.kp, device=self.device, dtype=torch.float32
                )
            self.sumF = _hip_ops().apply_step_colsum(
                self.F_local, grad, steps, self._colsum_partials, self.cfg
            )
        elif self.use_hip:
            _hip_ops().apply_step(self.F_local, grad, steps, self.cfg)
            self.sumF = self.F_local.float().sum(dim=0)
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
