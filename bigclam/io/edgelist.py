"""Edge-list ingest: parse, ID-remap, undirected adjacency.

Input contract (same as the reference's ``GraphLoader.edgeListFile``,
codes/bigclamv3-7.scala:26): one edge per line, ``src<ws>dst``, lines starting
with ``#`` are comments.  Node IDs are arbitrary non-negative integers.

Unlike the reference — which keeps raw (possibly non-contiguous) GraphX vertex
IDs and papers over the resulting lookup holes with zero rows
(``Flookup``, codes/bigclamv3-7.scala:94-104) — we remap IDs to a dense
``0..N-1`` internal space and keep the ``internal -> raw`` table for output.

Deduplication: SNAP files store undirected graphs either as both directions
(Email-Enron) or one direction (facebook_combined).  The reference's
``collectNeighborIds(Either)`` therefore yields neighbor arrays with
duplicates for both-direction files, silently doubling every edge term.  We
normalize instead: the canonical internal form is the simple undirected graph
(self-loops dropped, duplicates merged); every undirected edge appears in both
endpoints' CSR rows, so each edge is counted exactly twice per LLH sweep —
the convention of the reference's single-direction datasets.
"""
from __future__ import annotations

import io
import os
from dataclasses import dataclass

import numpy as np


@dataclass
class Graph:
    """Undirected graph in CSR form over dense internal IDs.

    ``indptr``/``indices`` cover both directions: ``indices[indptr[u]:indptr[u+1]]``
    are u's neighbors.  ``raw_ids[i]`` is the original file ID of internal
    node ``i``.  ``num_edges`` is the undirected edge count
    (= len(indices) // 2).
    """

    indptr: np.ndarray  # int64 [N+1]
    indices: np.ndarray  # int32 [2E]
    raw_ids: np.ndarray  # int64 [N]

    @property
    def num_nodes(self) -> int:
        return len(self.indptr) - 1

    @property
    def num_edges(self) -> int:
        return len(self.indices) // 2

    @property
    def num_directed_edges(self) -> int:
        return len(self.indices)

    def degrees(self) -> np.ndarray:
        return np.diff(self.indptr)

    def neighbors(self, u: int) -> np.ndarray:
        return self.indices[self.indptr[u] : self.indptr[u + 1]]


def parse_edge_array(path: str) -> np.ndarray:
    """Read an edge-list file into an int64 [M, 2] array.

    Fast path: the native mmap'd multithreaded parser (bigclam._io_native);
    falls back to pandas' C engine, then to pure python.
    """
    try:
        from .. import _io_native

        return np.asarray(_io_native.parse_edgelist(path))
    except ImportError:
        pass
    try:
        import pandas as pd

        df = pd.read_csv(
            path,
            sep=r"\s+",
            comment="#",
            header=None,
            usecols=[0, 1],
            dtype=np.int64,
            engine="c",
        )
        return df.values
    except Exception:
        rows = []
        with io.open(path, "r") as f:
            for line in f:
                line = line.strip()
                if not line or line.startswith("#"):
                    continue
                a, b = line.split()[:2]
                rows.append((int(a), int(b)))
        return np.asarray(rows, dtype=np.int64).reshape(-1, 2)


def _native_build_threshold() -> int:
    v = os.environ.get("BIGCLAM_NATIVE_CSR")
    if v == "0":
        return 1 << 62  # disabled
    if v == "1":
        return 0  # always
    return 2_000_000  # default: large arrays only


def build_graph(edges: np.ndarray, drop_self_loops: bool = True) -> Graph:
    """Build the canonical undirected CSR from a raw [M, 2] edge array.

    Large arrays take the parallel native builder (io_native.cpp
    ``build_csr`` — bucket-sorted dedupe + atomic-cursor scatter), which
    produces EXACTLY the same Graph as the numpy path below (tested in
    tests/test_io.py); numpy remains the reference and the fallback for
    sparse 64-bit id spaces.  ``BIGCLAM_NATIVE_CSR=1/0`` forces it
    on/off."""
    edges = np.asarray(edges, dtype=np.int64).reshape(-1, 2)
    if len(edges) >= _native_build_threshold():
        try:
            from .. import _io_native

            if len(edges) == 0 or edges.max() < (1 << 31):
                indptr, indices, raw_ids = _io_native.build_csr(
                    np.ascontiguousarray(edges), drop_self_loops
                )
                return Graph(
                    indptr=np.asarray(indptr),
                    indices=np.asarray(indices),
                    raw_ids=np.asarray(raw_ids),
                )
        except ImportError:
            pass
    if drop_self_loops:
        edges = edges[edges[:, 0] != edges[:, 1]]
    # dense ID remap
    raw_ids, remapped = np.unique(edges.reshape(-1), return_inverse=True)
    e = remapped.reshape(-1, 2).astype(np.int64)
    n = len(raw_ids)
    # canonicalize (min, max) and dedupe
    lo = np.minimum(e[:, 0], e[:, 1])
    hi = np.maximum(e[:, 0], e[:, 1])
    key = lo * n + hi
    uniq = np.unique(key)
    lo = (uniq // n).astype(np.int64)
    hi = (uniq % n).astype(np.int64)
    # both directions; one int64 key sort gives row-major + sorted rows
    # (np.sort on the key is ~2x np.lexsort((dst, src)) at 100M+ edges,
    # and np.bincount beats np.add.at by ~10x)
    key2 = np.concatenate([lo * n + hi, hi * n + lo])
    key2.sort(kind="stable")
    src = key2 // n
    indices = (key2 % n).astype(np.int32)
    counts = np.bincount(src, minlength=n)
    indptr = np.zeros(n + 1, dtype=np.int64)
    np.cumsum(counts, out=indptr[1:])
    return Graph(indptr=indptr, indices=indices, raw_ids=raw_ids)


def load_graph(path: str) -> Graph:
    if not os.path.exists(path):
        raise FileNotFoundError(path)
    return build_graph(parse_edge_array(path))
