"""CLI binary integration tests (reference driver parity: sgemm.cu flow).

The verification-pass test needs a GPU; the generator-consistency test runs
anywhere and guards drift between ft_sgemm_amd/kernel_table.py and the
committed csrc/generated/ files.
"""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_generator_output_is_committed():
    """csrc/generated/* must match what gen_kernels.py produces from the
    current kernel table (the generator is the source of truth)."""
    import tempfile
    before = {}
    gen_dir = os.path.join(ROOT, "csrc", "generated")
    for f in sorted(os.listdir(gen_dir)):
        before[f] = open(os.path.join(gen_dir, f)).read()
    subprocess.check_call([sys.executable,
                          os.path.join(ROOT, "csrc", "codegen",
                                       "gen_kernels.py")], cwd=ROOT)
    after = {f: open(os.path.join(gen_dir, f)).read()
             for f in sorted(os.listdir(gen_dir))}
    assert before == after, "run csrc/codegen/gen_kernels.py and commit"


@pytest.mark.gpu
def test_cli_verify_and_json(tmp_path):
    """bin/ft_sgemm end to end at N=512: all 17 kernel ids verify against
    rocBLAS (FT ids with the always-on injector), and FT_SGEMM_JSON emits
    one valid record per sweep cell."""
    exe = os.path.join(ROOT, "bin", "ft_sgemm")
    assert os.path.exists(exe), "bin/ft_sgemm not built (make cli)"
    jpath = tmp_path / "sweep.json"
    env = dict(os.environ, FT_SGEMM_JSON=str(jpath))
    out = subprocess.run([exe, "512", "512", "512", "0", "16"],
                         capture_output=True, text=True, env=env,
                         timeout=300)
    assert out.returncode == 0, out.stderr[-500:]
    verified = [l for l in out.stdout.splitlines() if "verified" in l]
    assert len(verified) == 17, out.stdout
    assert "MISMATCH" not in out.stdout
    recs = [json.loads(l) for l in open(jpath)]
    assert len(recs) == 14  # sweep ids {0,1..6,10,11..16} x 1 size
    by_id = {r["kernel_id"]: r for r in recs}
    assert by_id[16]["inject"] is True
    assert all(r["gflops"] > 0 for r in recs)
