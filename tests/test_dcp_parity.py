"""Resharding parity against torch.distributed.checkpoint (DCP) — the
reference's ground-truth oracle (tests/test_state_dict.py:206-265):
save with DCP under layout A, reload under layout B via BOTH DCP and the
store, and require bit-identical local shards."""

import asyncio
import tempfile
import uuid

import pytest
import torch

import torchstore_amd as ts
from torchstore_amd.runtime import spawn_actors, close_connections
from torchstore_amd.strategy import LocalRankStrategy
from tests.utils import DTensorWorker


async def _spawn_world(n, controller, tag):
    pg_file = tempfile.mktemp(prefix=f"ts-pg-{tag}-{uuid.uuid4().hex[:6]}")
    return await asyncio.to_thread(
        spawn_actors, n, DTensorWorker, f"dtw-{tag}", n, pg_file, controller
    )


async def _parity_case(
    put_world, put_mesh, put_placements, get_world, get_mesh, get_placements,
    shape=(16, 16),
):
    controller = await ts.initialize(
        num_storage_volumes=put_world,
        strategy=LocalRankStrategy(),
        storage_device="cpu",
    )
    ckpt = tempfile.mkdtemp(prefix="ts-dcp-")
    put_m = get_m = None
    try:
        put_m = await _spawn_world(put_world, controller, "put")
        get_m = await _spawn_world(get_world, controller, "get")
        res = await put_m.dcp_save_and_push.call(
            ckpt, "sd", shape, put_mesh, put_placements
        )
        assert all(r == "ok" for r in res)
        res = await get_m.dcp_load_and_compare.call(
            ckpt, "sd", shape, get_mesh, get_placements
        )
        assert all(r == "ok" for r in res)
    finally:
        for m in (put_m, get_m):
            if m is not None:
                await m.stop()
        await ts.shutdown()
        await close_connections()


async def test_dcp_parity_grow_world():
    await _parity_case(2, (2,), ["0"], 4, (4,), ["0"])


async def test_dcp_parity_dim_change():
    await _parity_case(2, (2,), ["0"], 2, (2,), ["1"])


async def test_dcp_parity_2d_transpose():
    await _parity_case(4, (2, 2), ["0", "1"], 4, (2, 2), ["1", "0"])


@pytest.mark.slow
async def test_dcp_parity_2d_to_1d_shrink():
    await _parity_case(4, (2, 2), ["0", "r"], 2, (2,), ["1"])
