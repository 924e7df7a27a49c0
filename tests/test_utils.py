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
    # straggler grid (3072: 288 blocks over 512 slots) -> large
    assert choose_tier(3072, 3072, 3072) == "large"
    # small/skinny shapes
    assert choose_tier(1024, 1024, 1024) == "large"
    assert choose_tier(512, 64, 256) == "tall"
    assert choose_tier(64, 512, 256) == "wide"
    assert choose_tier(64, 64, 64) == "large"
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
