"""Large-tensor path: multi-hundred-MB roundtrips per transport.

CPU runs a small sweep; the GPU-marked sweep covers the 2 GB reference
config (tests/test_large_tensors.py:27-125) over HIP IPC.
"""

import pytest
import torch

from benchmarks.large_tensor_sweep import sweep


async def test_cpu_sweep_shm():
    rows = await sweep([4, 32], "cpu", "shared_memory", None, repeats=1)
    assert len(rows) == 2


async def test_cpu_sweep_rpc():
    rows = await sweep([4, 32], "cpu", "rpc", None, repeats=1)
    assert len(rows) == 2


@pytest.mark.gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
async def test_gpu_sweep_ipc_2gb():
    rows = await sweep([64, 512, 2048], "cuda", "hip_ipc", None, repeats=2)
    # 2 GB tensors must exceed 100 GB/s through the store on MI355X
    big = rows[-1]
    assert big["put_MBps"] > 100_000, f"put too slow: {big}"
    assert big["get_MBps"] > 100_000, f"get too slow: {big}"
