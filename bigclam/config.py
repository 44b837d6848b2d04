"""BigCLAM configuration.

The complete hyperparameter surface of the reference scripts (hard-coded
``var``s there — see SURVEY.md §2.15; reference: codes/bigclamv3-7.scala:14-24,
121-123, 217 and codes/bigclam4-7.scala:14-28) exposed as one dataclass.
Defaults are the reference's values.
"""
from __future__ import annotations

import dataclasses
import json
from dataclasses import dataclass, field
from typing import Optional


@dataclass
class BigClamConfig:
    # --- model size -------------------------------------------------------
    #: number of communities (columns of F). Reference: codes/bigclamv3-7.scala:15
    k: int = 100

    # --- optimizer (projected gradient ascent + Armijo backtracking) ------
    #: Armijo sufficient-decrease constant (codes/bigclamv3-7.scala:121)
    alpha: float = 0.05
    #: geometric step-ladder ratio (codes/bigclamv3-7.scala:122)
    beta: float = 0.1
    #: line-search depth: candidates are {1, beta, ..., beta**ls_steps}
    #: (reference ``MaxIter``, codes/bigclamv3-7.scala:123)
    ls_steps: int = 15
    #: relative-LLH convergence tolerance (codes/bigclamv3-7.scala:217)
    tol: float = 1e-4
    #: hard cap on gradient sweeps (the reference loops unboundedly; we bound)
    max_sweeps: int = 1000

    # --- numeric clamps (codes/bigclamv3-7.scala:20-24) -------------------
    min_p: float = 1e-4
    max_p: float = 0.9999
    min_f: float = 0.0
    max_f: float = 1000.0
    #: declared but never used by the reference; kept for config parity
    eps_comm_force: float = 1e-6

    # --- model selection over K (codes/bigclam4-7.scala:16-20, 259) -------
    k_min: int = 1000
    k_max: int = 9000
    k_div: int = 100
    k_tol: float = 1e-3

    # --- seeding / init ---------------------------------------------------
    #: RNG seed for random pad rows and R-MAT generation
    seed: int = 0
    #: reproduce the reference's lowest-ID tuple-min seed ranking bug
    #: (codes/bigclamv3-7.scala:51 — Scala tuple min orders by node id first).
    #: Default False = intended min-conductance semantics.
    seed_rank_compat: bool = False
    #: include the seed node itself in its community's init row
    #: (v2 behavior, codes/Bigclamv2.scala:70); v3 uses neighbors only.
    init_include_seed: bool = False

    # --- execution --------------------------------------------------------
    #: compute dtype for F storage: "fp32" or "bf16" (accumulation is fp32)
    dtype: str = "fp32"
    #: device: "cuda" or "cpu"
    device: str = "cuda"
    #: checkpoint every n sweeps (0 = disabled)
    checkpoint_every: int = 0
    checkpoint_dir: Optional[str] = None
    #: output path for community assignments
    out: Optional[str] = None

    def __post_init__(self):
        if not 0 <= self.ls_steps <= 15:
            raise ValueError(
                "ls_steps must be in [0, 15] (ladder of at most 16 "
                f"candidates, the GPU kernels' MAX_LS); got {self.ls_steps}"
            )
        if self.dtype not in ("fp32", "bf16"):
            raise ValueError(f"dtype must be fp32 or bf16, got {self.dtype!r}")

    def ladder(self) -> list:
        """The Armijo candidate-step ladder, largest first.

        Reference builds ``List(1.0)`` then prepends beta^i for i=1..MaxIter
        (codes/bigclamv3-7.scala:125-130); evaluation keeps the max accepted
        step, so ordering here is descending for first-accept-wins scans.
        """
        return [self.beta ** i for i in range(self.ls_steps + 1)]

    def to_json(self) -> str:
        return json.dumps(dataclasses.asdict(self), indent=2)

    @classmethod
    def from_json(cls, s: str) -> "BigClamConfig":
        d = json.loads(s)
        known = {f.name for f in dataclasses.fields(cls)}
        return cls(**{k: v for k, v in d.items() if k in known})


def k_grid(k_min: int, k_max: int, k_div: int) -> list:
    """Geometric K grid for model selection.

    Mirrors codes/bigclam4-7.scala:116-133: ratio = exp(log(k_max/k_min)/k_div),
    each step multiplies by the ratio (floor), stepping by at least 1, with
    k_max appended.  Deviation (documented): the reference computes
    ``maxCom/minCom`` with *integer* division before the log — we use true
    division (identical for the exercised 9000/1000 grid).
    """
    import math

    if k_min >= k_max:
        return [k_min]
    ratio = math.exp(math.log(k_max / k_min) / k_div)
    ks = [k_min]
    x = k_min
    while True:
        xt = int(x * ratio)
        if xt == x:
            xt += 1
        x = xt
        if x >= k_max:
            break
        ks.append(x)
    ks.append(k_max)
    return ks
