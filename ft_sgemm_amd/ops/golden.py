"""CPU golden model: SGEMM + offline row/column-checksum ABFT.

This is SURVEY.md section-7 config 0 — the semantic anchor for every HIP
kernel.  It mirrors the maths of the fused device pipeline
(/root/reference/kernel/ft_sgemm/include_code_gen/ft_sgemm_huge.cuh, traced
in SURVEY.md section 2.3) at whole-matrix granularity:

  C        = A @ B^T                 A: MxK, B: NxK, all column-major
  cr       = C @ e   maintained as  sum_k A[:,k] * sB[k],  sB[k] = sum_j B[j,k]
  cc       = e^T C   maintained as  sum_k sA[k] * B[:,k]^T, sA[k] = sum_i A[i,k]
  residual_row = rowsum(C) - cr ;  residual_col = colsum(C) - cc
  a corrupted element (i,j) shows up at the intersection
  |residual_row[i]| > tau AND |residual_col[j]| > tau and is corrected by
  subtracting residual_row[i] (exactly the reference's branch-free
  correction, ft_sgemm_huge.cuh:422-485, with explicit sign convention:
  residual = computed - checksum = +error).

Everything operates on numpy fp32 arrays; fp64 is used only for the
`exact` reference in tests.
"""

from __future__ import annotations

from dataclasses import dataclass

import numpy as np

from ..kernel_table import ERR_BOUND, ERROR_INJECT, N_INJECT


def sgemm_golden(a: np.ndarray, b: np.ndarray, c: np.ndarray,
                 alpha: float = 1.0, beta: float = 0.0) -> np.ndarray:
    """C = alpha * A @ B^T + beta * C (fp32), reference semantics
    (sgemm.cu:108 cublasSgemm OP_N, OP_T)."""
    m, k = a.shape
    n, kb = b.shape
    assert k == kb and c.shape == (m, n)
    prod = (a.astype(np.float32) @ b.astype(np.float32).T).astype(np.float32)
    return (np.float32(alpha) * prod + np.float32(beta) * c).astype(np.float32)


@dataclass
class Checksums:
    cr: np.ndarray  # length M: maintained row checksum  C @ e
    cc: np.ndarray  # length N: maintained col checksum  e^T C


def abft_encode(a: np.ndarray, b: np.ndarray) -> Checksums:
    """Maintained checksums of the (alpha/beta-free) product A @ B^T."""
    s_b = b.sum(axis=0, dtype=np.float32)          # sB[k] = sum_j B[j,k]
    s_a = a.sum(axis=0, dtype=np.float32)          # sA[k] = sum_i A[i,k]
    cr = (a.astype(np.float32) @ s_b).astype(np.float32)       # M
    cc = (b.astype(np.float32) @ s_a).astype(np.float32)       # N
    return Checksums(cr=cr, cc=cc)


def abft_residuals(prod: np.ndarray, cks: Checksums):
    """residual = computed - checksum (== +injected error at a corrupted
    element's row and column, roundoff-sized elsewhere)."""
    rr = prod.sum(axis=1, dtype=np.float32) - cks.cr
    rc = prod.sum(axis=0, dtype=np.float32) - cks.cc
    return rr, rc


def abft_detect_correct(prod: np.ndarray, cks: Checksums,
                        tau: float = ERR_BOUND):
    """Locate and correct corrupted elements at row x col residual
    intersections.  Returns (corrected, locations).  Branch-free form of the
    device correction: C[i,j] -= (|rr_i|>tau && |rc_j|>tau) * rr_i."""
    rr, rc = abft_residuals(prod, cks)
    row_bad = np.abs(rr) > tau
    col_bad = np.abs(rc) > tau
    mask = np.outer(row_bad, col_bad)
    corrected = prod - mask * rr[:, None]
    locs = [tuple(int(x) for x in ij) for ij in np.argwhere(mask)]
    return corrected.astype(np.float32), locs


def inject_faults(prod: np.ndarray, k: int, magnitude: float = ERROR_INJECT,
                  n_inject: int = N_INJECT, seed: int = 0):
    """Model of the in-kernel injector (ft_sgemm_huge.cuh:324-327): n_inject
    single-element corruptions of fixed magnitude at deterministic rotating
    positions.  In the offline model the K-period collapses to n_inject
    distinct (i,j) sites."""
    rng = np.random.default_rng(seed)
    m, n = prod.shape
    out = prod.copy()
    sites = set()
    while len(sites) < min(n_inject, m * n):
        sites.add((int(rng.integers(m)), int(rng.integers(n))))
    for (i, j) in sites:
        out[i, j] += np.float32(magnitude)
    return out, sorted(sites)


def ft_sgemm_golden(a, b, c, alpha=1.0, beta=0.0, tau=ERR_BOUND,
                    inject=True, seed: int = 0, n_windows: int = N_INJECT):
    """End-to-end offline ABFT GEMM, windowed exactly like the fused device
    pipeline: K is processed in ~20 windows; each window accumulates its
    panel product AND its checksum contributions, injects (at most) one
    fault, then detects/locates/corrects before the next window — ABFT's
    single-fault-per-verification-interval guarantee (the reason the device
    kernels verify every K/20 columns, ft_sgemm_huge.cuh:324-327)."""
    m, k = a.shape
    n = b.shape[0]
    rng = np.random.default_rng(seed)
    nwin = max(1, min(n_windows, k))
    bounds = np.linspace(0, k, nwin + 1, dtype=int)
    prod = np.zeros((m, n), dtype=np.float32)
    cks = Checksums(cr=np.zeros(m, dtype=np.float32),
                    cc=np.zeros(n, dtype=np.float32))
    injected_sites, located = [], []
    for w in range(nwin):
        ap = a[:, bounds[w]:bounds[w + 1]].astype(np.float32)
        bp = b[:, bounds[w]:bounds[w + 1]].astype(np.float32)
        prod += ap @ bp.T
        cks.cr += ap @ bp.sum(axis=0, dtype=np.float32)
        cks.cc += bp @ ap.sum(axis=0, dtype=np.float32)
        if inject:
            site = (int(rng.integers(m)), int(rng.integers(n)))
            prod[site] += np.float32(ERROR_INJECT)
            injected_sites.append(site)
        prod, locs = abft_detect_correct(prod, cks, tau)
        located.extend(locs)
    out = (np.float32(alpha) * prod + np.float32(beta) * c).astype(np.float32)
    return out, injected_sites, located


def abft_ratio_locate(prod: np.ndarray, a: np.ndarray, b: np.ndarray,
                      seg: int, tau: float = ERR_BOUND):
    """Golden model of the DEVICE locate scheme (csrc/ft_kernels.hpp):
    per seg-row band of C, maintain per-column plain (cc) and
    row-index-weighted (cw) checksums from the precomputed segment sums of
    A; a fault's column has |rc| > tau, its magnitude is rc, and its row
    within the band is round(rw / rc).  Returns (corrected, locations) with
    locations in global (i, j) coordinates."""
    m, n = prod.shape
    out = prod.copy()
    locs = []
    w = np.arange(seg, dtype=np.float32)
    for s0 in range(0, m, seg):
        band = out[s0:s0 + seg]
        ab = a[s0:s0 + seg].astype(np.float32)
        sa = ab.sum(axis=0, dtype=np.float32)           # plain segment sums
        saw = w @ ab                                    # weighted
        cc = (b.astype(np.float32) @ sa).astype(np.float32)     # N
        cw = (b.astype(np.float32) @ saw).astype(np.float32)    # N
        rc = band.sum(axis=0, dtype=np.float32) - cc
        rw = (w @ band).astype(np.float32) - cw
        for j in np.nonzero(np.abs(rc) > tau)[0]:
            # Fail-safe parity with the device kernel (ft_kernels.hpp
            # locate_correct): the device only subtracts where an
            # accumulator row MATCHES the computed index, so a nonsense
            # ratio (residual noise, or two same-column faults summed —
            # rc = e1+e2, row = round((r1*e1+r2*e2)/(e1+e2))) corrects
            # nothing out of band.  Mirror that guard instead of letting
            # numpy wrap a negative i onto the wrong row:
            # an out-of-band index is recorded as detected-but-uncorrected
            # location (-1, j).
            i = int(np.rint(rw[j] / rc[j]))
            if 0 <= i < band.shape[0]:
                band[i, j] -= rc[j]
                locs.append((s0 + i, int(j)))
            else:
                locs.append((-1, int(j)))
    return out.astype(np.float32), locs


def baseline_ft_check(a, b, panel_k: int = 256, tau: float = ERR_BOUND):
    """Non-fused baseline semantics (include/baseline_ft_sgemm.cuh:1-33):
    per 256-wide K panel, compare e^T(A@B^T) against (e^T A)@B^T and
    (A@B^T)e against A@(B^T e); produce the scalar verdicts the cuBLAS
    Sdot chain produces.  Returns the max |residual| seen (0-ish when
    fault-free)."""
    k = a.shape[1]
    worst = 0.0
    for k0 in range(0, k, panel_k):
        ap = a[:, k0:k0 + panel_k]
        bp = b[:, k0:k0 + panel_k]
        prod = (ap.astype(np.float32) @ bp.astype(np.float32).T)
        rr = prod.sum(axis=1, dtype=np.float32) - ap @ bp.sum(axis=0, dtype=np.float32)
        rc = prod.sum(axis=0, dtype=np.float32) - bp @ ap.sum(axis=0, dtype=np.float32)
        worst = max(worst, float(np.abs(rr).max(initial=0)),
                    float(np.abs(rc).max(initial=0)))
    return worst
