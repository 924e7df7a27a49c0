import os
import numpy as np

from ft_sgemm_amd.utils import PerfTable, generate_random_matrix, gflops, verify_matrix
from ft_sgemm_amd.kernel_table import (KERNEL_NAMES, KERNEL_TABLE,
                                       PERF_SWEEP_IDS, TILING, threads)


def test_generate_random_matrix_range_and_determinism():
    m1 = generate_random_matrix(64, 32, seed=10)
    m2 = generate_random_matrix(64, 32, seed=10)
    assert m1.dtype == np.float32 and m1.shape == (64, 32)
    assert np.array_equal(m1, m2)
    assert m1.max() < 0.9 and m1.min() > -0.9
    assert m1.flags.f_contiguous  # column-major


def test_verify_matrix_reference_semantics():
    ref = np.array([[100.0, 0.001]], dtype=np.float32)
    # abs diff 0.5 but rel diff 0.005 < 1e-2 -> passes (AND semantics)
    got = np.array([[100.5, 0.001]], dtype=np.float32)
    ok, _, _ = verify_matrix(ref, got)
    assert ok
    # abs diff 0.005 < 1e-2, rel enormous -> still passes
    got2 = np.array([[100.0, 0.006]], dtype=np.float32)
    ok2, _, _ = verify_matrix(ref, got2)
    assert ok2
    # both exceeded -> fails
    got3 = np.array([[103.0, 0.001]], dtype=np.float32)
    ok3, idx, _ = verify_matrix(ref, got3)
    assert not ok3 and idx == (0, 0)


def test_gflops_protocol():
    # 2*M*N*K*reps / t  (sgemm.cu:431-435)
    assert abs(gflops(1000, 1000, 1000, 5, 1.0) - 10.0) < 1e-9


def test_kernel_table_parity():
    assert KERNEL_NAMES[0] == "cublas"
    assert KERNEL_NAMES[6] == "kernel_sgemm_huge"
    assert KERNEL_NAMES[10] == "abft_baseline"
    assert KERNEL_NAMES[16] == "abft_kernel_huge"
    assert PERF_SWEEP_IDS == [0, 1, 2, 3, 4, 5, 6, 10, 11, 12, 13, 14, 15, 16]
    assert 7 not in KERNEL_TABLE and 9 not in KERNEL_TABLE
    # wavefront-64 blocks; sizes tuned for CDNA4 (large got a second wave,
    # probe-measured +6% plain / +11% fused) — the reference's 32-thread
    # warp sizing does not transfer
    assert [threads(t) for t in TILING] == [64, 64, 128, 128, 128, 256]
    for t in TILING.values():
        assert t["bm"] % t["wm"] == 0 and t["bn"] % t["wn"] == 0


def test_perf_table_format():
    tb = PerfTable([1024, 1536])
    tb.add("cublas", {1024: 4695.4, 1536: 5357.1})
    text = tb.render()
    lines = text.splitlines()
    assert lines[0].startswith("Matrix Size|")
    assert "4695" in lines[1] and lines[1].endswith("|")


def test_choose_tier():
    from ft_sgemm_amd.ops import choose_tier
    # big even shapes fill the huge grid
    assert choose_tier(4096, 4096, 4096) == "huge"
    assert choose_tier(8192, 8192, 8192) == "huge"
    assert choose_tier(3584, 3584, 3584) == "huge"
    # the large tier wins the 2048-3072 band (measured, cli_sweep_r2)
    assert choose_tier(3072, 3072, 3072) == "large"
    assert choose_tier(2048, 2048, 2048) == "large"
    # below ~1024 large blocks the medium grid wins
    assert choose_tier(1024, 1024, 1024) == "medium"
    assert choose_tier(512, 64, 256) == "tall"
    assert choose_tier(64, 512, 256) == "wide"
    assert choose_tier(64, 64, 64) == "medium"
    assert choose_tier(32, 32, 32) == "medium"
    assert choose_tier(16, 16, 32) == "small"
    # shapes no tier divides -> None, and the auto entry points fall back
    # to the rocBLAS path instead of raising
    assert choose_tier(17, 16, 32) is None
    assert choose_tier(100, 100, 100) is None


def test_make_tables_from_sweep_log(tmp_path):
    """The judged-table generator parses a CLI sweep log and derives
    per-tier overheads + per-column best-of-rows ratios."""
    import subprocess
    import sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    log = os.path.join(root, "profiles", "cli_sweep_r2.log")
    r = subprocess.run(
        [sys.executable, os.path.join(root, "tools", "make_tables.py"), log,
         str(tmp_path)], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-500:]
    ovh = open(tmp_path / "overhead_table.txt").read()
    assert "huge" in ovh and "ref@4096" in ovh
    ratios = open(tmp_path / "sweep_ratios.txt").read()
    assert "best plain" in ratios and " 4096 " in ratios


def test_compare_reference_runs():
    import subprocess
    import sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, os.path.join(root, "tools", "compare_reference.py"),
         os.path.join(root, "profiles", "cli_sweep_r2.log")],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-500:]
    assert "fused-ABFT overhead" in r.stdout


def test_streamk_partition_properties():
    """Property test of the stream-K work partition + fixup mapping
    (csrc/ft_streamk.hpp): balanced contiguous ranges cover [0, total)
    exactly once; for every SPLIT tile, g0's tail-partial slot plus the
    head-partial slots of (g0, gl] reconstruct exactly the tile's units;
    fully-owned tiles have no slot writers; no workgroup writes more than
    one head and one tail slot."""
    import random
    rng = random.Random(20260914)

    def wg_of(u, q, r):
        return u // (q + 1) if u < r * (q + 1) else r + (u - r * (q + 1)) // q

    for _ in range(300):
        ntiles = rng.randint(1, 80)
        upt = rng.choice([1, 2, 3, 8, 16, 48, 72])
        total = ntiles * upt
        G = rng.randint(1, 64)
        G = min(G, total)
        q, r = divmod(total, G)

        # walk each workgroup's segments exactly as the kernel does
        covered = [0] * total
        head_writers, tail_writers = {}, {}   # g -> (tile, units)
        for g in range(G):
            u = g * q + min(g, r)
            u_end = u + q + (1 if g < r else 0)
            first = True
            while u < u_end:
                tile = u // upt
                w_lo = u - tile * upt
                seg = min(u_end - u, upt - w_lo)
                for x in range(u, u + seg):
                    covered[x] += 1
                full = (w_lo == 0) and (seg == upt)
                if not full:
                    if w_lo != 0:
                        assert first, "head partial must be the 1st segment"
                        assert g not in head_writers
                        head_writers[g] = (tile, seg)
                    else:
                        assert g not in tail_writers
                        tail_writers[g] = (tile, seg)
                u += seg
                first = False

        assert covered == [1] * total, "units must be covered exactly once"

        # fixup reconstruction per tile
        for t in range(ntiles):
            g0 = wg_of(t * upt, q, r)
            gl = wg_of(t * upt + upt - 1, q, r)
            if g0 == gl:
                # fully owned: nobody wrote a partial for this tile
                assert all(tw[0] != t for tw in head_writers.values())
                assert all(tw[0] != t for tw in tail_writers.values())
                continue
            units = tail_writers[g0][1]  # g0's tail partial
            for gc in range(g0 + 1, gl + 1):
                assert head_writers[gc][0] == t
                units += head_writers[gc][1]
            assert units == upt, f"tile {t}: {units} != {upt}"
