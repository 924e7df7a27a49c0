"""Multi-process CPU tests of the distributed block-row SGEMM (gloo,
world_size 2) — the same code path the 8-GPU RCCL run takes."""

import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from ft_sgemm_amd.parallel.distributed import (block_row_sgemm, local_shard,
                                               torch_gemm_fn)

M, N, K = 128, 96, 256
PANEL = 64


def _make_full(seed=10):
    g = torch.Generator().manual_seed(seed)
    a = torch.rand((K, M), generator=g) * 1.8 - 0.9   # col-major A (MxK)
    b = torch.rand((K, N), generator=g) * 1.8 - 0.9   # col-major B (NxK)
    return a, b


def _worker(rank, world, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        a, b = _make_full()
        mlo, mhi = local_shard(M, rank, world)
        nlo, nhi = local_shard(N, rank, world)
        a_loc = a[:, mlo:mhi].contiguous()
        b_loc = b[:, nlo:nhi].contiguous()
        c_loc = torch.zeros((N, mhi - mlo))
        block_row_sgemm(a_loc, b_loc, c_loc, panel_k=PANEL,
                        gemm_fn=torch_gemm_fn, alpha=1.0, beta=0.0)
        ref = b.transpose(0, 1) @ a[:, mlo:mhi]
        err = (c_loc - ref).abs().max().item()
        q.put((rank, err))
    finally:
        dist.destroy_process_group()


def test_block_row_sgemm_world2():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = 29873
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get() for _ in range(2)]
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    for rank, err in results:
        assert err < 1e-4, f"rank {rank} max err {err}"


def test_block_row_sgemm_world1_panels():
    a, b = _make_full()
    c = torch.zeros((N, M))
    block_row_sgemm(a, b, c, panel_k=PANEL, gemm_fn=torch_gemm_fn,
                    alpha=2.0, beta=0.0)
    ref = 2.0 * (b.transpose(0, 1) @ a)
    assert (c - ref).abs().max().item() < 1e-4


def test_block_row_beta_accumulate():
    a, b = _make_full()
    c0 = torch.rand((N, M))
    c = c0.clone()
    block_row_sgemm(a, b, c, panel_k=PANEL, gemm_fn=torch_gemm_fn,
                    alpha=1.0, beta=-1.5)
    ref = b.transpose(0, 1) @ a - 1.5 * c0
    assert (c - ref).abs().max().item() < 1e-4


def test_local_shard():
    assert local_shard(32768, 3, 8) == (3 * 4096, 4 * 4096)
    with pytest.raises(AssertionError):
        local_shard(100, 0, 3)


def _bench_worker(rank, world, port, mode, q):
    import io
    import subprocess
    import sys
    env = dict(os.environ, RANK=str(rank), WORLD_SIZE=str(world),
               LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
               MASTER_PORT=str(port))
    r = subprocess.run(
        [sys.executable, "bench.py", "--device", "cpu", "--size", "128",
         "--steps", "3", "--warmup", "1", "--mode", mode],
        capture_output=True, text=True, env=env,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    q.put((rank, r.returncode, r.stdout, r.stderr))


@pytest.mark.parametrize("mode", ["replicated", "blockrow"])
def test_bench_contract_world2_cpu(mode):
    """End-to-end smoke of the driver's bench contract at world_size 2 over
    gloo: rank 0 must print exactly one valid JSON line with the whole-job
    aggregate, rank 1 nothing."""
    import json
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = 29981 if mode == "replicated" else 29982
    procs = [ctx.Process(target=_bench_worker, args=(r, 2, port, mode, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, rc, out, err = q.get()
        results[rank] = (rc, out, err)
    for p in procs:
        p.join(timeout=180)
    for rank, (rc, out, err) in results.items():
        assert rc == 0, f"rank {rank} failed: {err[-800:]}"
    payload = [l for l in results[0][1].splitlines() if l.startswith("{")]
    assert len(payload) == 1, results[0][1]
    j = json.loads(payload[0])
    assert j["n_gpus"] == 2
    assert j["metric"] == "fused_abft_sgemm_gflops"
    assert j["scaling"] == ("weak" if mode == "replicated" else "strong")
    assert not [l for l in results[1][1].splitlines() if l.startswith("{")]
