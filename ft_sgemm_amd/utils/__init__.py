"""Host utilities: deterministic matrix generation, verification, timing table.

Re-designed equivalents of the reference host-utils layer
(/root/reference/utils/utils.cu): `generate_random_matrix` (utils.cu:23,
values in (-0.9, 0.9)), `verify_matrix` (utils.cu:61-77, fail iff abs diff
> 1e-2 AND rel diff > 1e-2), `cpu_gemm` (utils.cu:79).  The reference's known
bugs (verify_vector returning a function pointer at utils.cu:58, the E_
off-by-one at sgemm.cu:66) are intentionally not reproduced.
"""

from __future__ import annotations

import numpy as np

ABS_TOL = 1e-2
REL_TOL = 1e-2


def generate_random_matrix(m: int, n: int, seed: int | None = None,
                           rng: np.random.Generator | None = None) -> np.ndarray:
    """Column-major m x n fp32 matrix with entries uniform in (-0.9, 0.9).

    Small magnitudes keep fp32 accumulation error far below the ABFT
    threshold across the whole N=1024..6144 sweep (reference draws the
    same range, utils.cu:23-31).
    """
    if rng is None:
        rng = np.random.default_rng(10 if seed is None else seed)
    return np.asfortranarray(rng.uniform(-0.9, 0.9, size=(m, n)).astype(np.float32))


def verify_matrix(ref: np.ndarray, got: np.ndarray,
                  abs_tol: float = ABS_TOL, rel_tol: float = REL_TOL):
    """Reference tolerance semantics: an element FAILS only when its absolute
    difference exceeds abs_tol AND its relative difference exceeds rel_tol
    (utils.cu:61-77).  Returns (ok, first_bad_index_or_None, max_abs_diff)."""
    ref = np.asarray(ref, dtype=np.float64)
    got = np.asarray(got, dtype=np.float64)
    diff = np.abs(ref - got)
    rel = diff / np.maximum(np.abs(ref), 1e-30)
    bad = (diff > abs_tol) & (rel > rel_tol)
    if not bad.any():
        return True, None, float(diff.max(initial=0.0))
    idx = tuple(int(x) for x in np.argwhere(bad)[0])
    return False, idx, float(diff.max(initial=0.0))


def gflops(m: int, n: int, k: int, reps: int, elapsed_s: float) -> float:
    """GFLOPS with the reference protocol: 2*M*N*K*reps / time
    (sgemm.cu:431-435; reps=5 there)."""
    if elapsed_s <= 0:
        return float("inf")
    return 2.0 * m * n * k * reps / elapsed_s / 1e9


class PerfTable:
    """Accumulates and prints the reference-format sweep table
    (README.md:38-53: rows = kernels, cols = matrix sizes)."""

    def __init__(self, sizes):
        self.sizes = list(sizes)
        self.rows = []  # (name, {size: gflops})

    def add(self, name: str, values: dict):
        self.rows.append((name, dict(values)))

    def render(self) -> str:
        out = []
        header = "Matrix Size|" + "|".join(f"{s:8d}" for s in self.sizes) + "|"
        out.append(header)
        for name, vals in self.rows:
            cells = "|".join(
                f"{vals[s]:8.0f}" if s in vals else " " * 8 for s in self.sizes)
            out.append(f"{name:>11s}|{cells}|")
        return "\n".join(out)
