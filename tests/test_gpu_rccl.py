"""On-hardware RCCL tests (single GPU, world-size 1).

The 8-GPU run is driver-territory; what a 1-GPU box CAN prove is the RCCL
branch itself: nccl(=RCCL) backend init, device-tensor
all_gather_into_tensor with async_op=True, and the block-row K-panel loop
consuming the gathered buffer with the fused-ABFT MFMA kernel on the same
stream (VERDICT r01 next #4a — this path had never executed on a GPU in
round 1, only gloo/CPU equivalents).
"""

import os
import socket

import pytest
import torch
import torch.distributed as dist

pytestmark = pytest.mark.gpu

from ft_sgemm_amd import ops
from ft_sgemm_amd.parallel import block_row_sgemm


def _free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.fixture()
def nccl_world1():
    assert ops.have_extension(), "HIP extension must be built on a GPU box"
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(_free_port())
    dist.init_process_group("nccl", rank=0, world_size=1)
    torch.cuda.set_device(0)
    yield
    dist.destroy_process_group()


def test_rccl_allgather_device_tensor(nccl_world1):
    """all_gather_into_tensor on device fp32 tensors over RCCL."""
    shard = torch.arange(2048 * 64, device="cuda", dtype=torch.float32)
    shard = shard.view(2048, 64)
    out = torch.empty((1 * 2048, 64), device="cuda", dtype=torch.float32)
    work = dist.all_gather_into_tensor(out, shard, async_op=True)
    work.wait()
    torch.cuda.synchronize()
    assert torch.equal(out, shard)


def test_rccl_blockrow_fused_abft(nccl_world1):
    """The production block-row path through an initialized RCCL group:
    rotating gather buffers, async all-gather, rank-chunk views, fused-ABFT
    MFMA panel GEMMs with injection — verified against plain torch fp32."""
    n = 2048
    a, b, c = ops.make_operands(n, n, n)
    ref = ops.torch_reference(a, b, c, 1.0, -1.5)

    def gemm_fn(ap, bp, cl, al, be):
        ops.ft_sgemm("huge", ap, bp, cl, al, be, inject=True)

    block_row_sgemm(a, b, c, panel_k=512, gemm_fn=gemm_fn, alpha=1.0,
                    beta=-1.5)
    torch.cuda.synchronize()
    diff = (ref - c).abs()
    rel = diff / ref.abs().clamp_min(1e-30)
    assert not ((diff > 1e-2) & (rel > 1e-2)).any(), (
        f"max abs diff {diff.max().item():.4e}")
