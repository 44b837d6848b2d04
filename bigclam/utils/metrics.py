"""Structured per-sweep metrics: JSONL + stdout (SURVEY.md §5 observability).

The reference's entire observability stack is a ``println`` of sweep index
and LLH (codes/bigclamv3-7.scala:216); here every sweep logs LLH, the
relative change, accepted-step histogram, edges/sec and the per-phase time
breakdown, as one JSON line.
"""
from __future__ import annotations

import json
import sys
import time
from typing import Optional


class MetricsLogger:
    def __init__(self, path: Optional[str] = None, rank: int = 0, quiet: bool = False):
        self.rank = rank
        self.quiet = quiet
        self._f = open(path, "a") if (path and rank == 0) else None

    def log(self, record: dict):
        if self.rank != 0:
            return
        record = dict(record, ts=time.time())
        line = json.dumps(record)
        if self._f:
            self._f.write(line + "\n")
            self._f.flush()
        if not self.quiet:
            print(line, file=sys.stderr, flush=True)

    def close(self):
        if self._f:
            self._f.close()
            self._f = None


class PhaseTimer:
    """Wall-clock phase timer; on CUDA devices synchronizes at boundaries
    only when enabled (bench mode keeps it off inside the timed region)."""

    def __init__(self, sync: bool = False):
        self.sync = sync
        self.times = {}
        self._t0 = None
        self._name = None

    def _now(self):
        if self.sync:
            import torch

            if torch.cuda.is_available():
                torch.cuda.synchronize()
        return time.perf_counter()

    def start(self, name: str):
        self._name = name
        self._t0 = self._now()

    def stop(self):
        if self._name is not None:
            dt = self._now() - self._t0
            self.times[self._name] = self.times.get(self._name, 0.0) + dt
            self._name = None

    def phase(self, name: str):
        timer = self

        class _Ctx:
            def __enter__(self):
                timer.start(name)

            def __exit__(self, *a):
                timer.stop()

        return _Ctx()
