"""DTensor put/get with automatic resharding across mismatched layouts.

Mirrors the reference's resharding matrix (tests/test_resharding_basic.py):
Shard(d) → Shard(d'), world growth/shrink, 1-D↔2-D meshes, replication,
and the partial-commit gate.  Put world and get world are independent
actor meshes sharing one store.
"""

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
    mesh = await asyncio.to_thread(
        spawn_actors,
        n,
        DTensorWorker,
        f"dtw-{tag}",
        n,
        pg_file,
        controller,
    )
    return mesh


async def _reshard_case(
    put_world,
    put_mesh_shape,
    put_placements,
    get_world,
    get_mesh_shape,
    get_placements,
    shape=(16, 16),
):
    controller = await ts.initialize(
        num_storage_volumes=put_world,
        strategy=LocalRankStrategy(),
        storage_device="cpu",
    )
    put_mesh = get_mesh = None
    try:
        put_mesh = await _spawn_world(put_world, controller, "put")
        get_mesh = await _spawn_world(get_world, controller, "get")
        res = await put_mesh.put_dtensor.call(
            "w", shape, put_mesh_shape, put_placements
        )
        assert all(r == "ok" for r in res)
        res = await get_mesh.get_dtensor.call(
            "w", shape, get_mesh_shape, get_placements
        )
        assert all(r == "ok" for r in res)
    finally:
        for m in (put_mesh, get_mesh):
            if m is not None:
                await m.stop()
        await ts.shutdown()
        await close_connections()


async def test_shard0_to_shard1():
    await _reshard_case(2, (2,), ["0"], 2, (2,), ["1"])


async def test_shard0_grow_world():
    await _reshard_case(2, (2,), ["0"], 4, (4,), ["0"])


async def test_shard1_shrink_world():
    await _reshard_case(4, (4,), ["1"], 2, (2,), ["1"])


async def test_replicate_to_shard():
    await _reshard_case(2, (2,), ["r"], 2, (2,), ["0"])


async def test_shard_to_replicate():
    await _reshard_case(2, (2,), ["0"], 2, (2,), ["r"])


async def test_2d_to_1d():
    await _reshard_case(4, (2, 2), ["0", "1"], 2, (2,), ["0"])


async def test_1d_to_2d():
    await _reshard_case(2, (2,), ["1"], 4, (2, 2), ["r", "0"])


async def test_2d_to_2d_transposed():
    await _reshard_case(4, (2, 2), ["0", "1"], 4, (2, 2), ["1", "0"])


async def test_fsdp_style_shard_replicate():
    # Shard(0) + Replicate: the fully_shard default layout
    await _reshard_case(4, (2, 2), ["0", "r"], 2, (2,), ["0"])


async def test_uneven_shards_world3():
    """16 rows over 3 ranks: chunks of 6/6/4 (torch chunk semantics) —
    offsets and intersections must handle the ragged tail."""
    await _reshard_case(3, (3,), ["0"], 2, (2,), ["0"], shape=(16, 16))


async def test_uneven_shards_dim_change():
    await _reshard_case(3, (3,), ["0"], 3, (3,), ["1"], shape=(16, 12))


async def test_uneven_grow_world():
    await _reshard_case(2, (2,), ["1"], 3, (3,), ["1"], shape=(8, 10))


async def test_full_tensor_get_from_shards():
    """A rank outside any mesh fetches the assembled full tensor."""
    controller = await ts.initialize(
        num_storage_volumes=2,
        strategy=LocalRankStrategy(),
        storage_device="cpu",
    )
    put_mesh = None
    try:
        put_mesh = await _spawn_world(2, controller, "put")
        res = await put_mesh.put_dtensor.call("w", (16, 16), (2,), ["0"])
        assert all(r == "ok" for r in res)
        from tests.utils import make_full_tensor

        out = await ts.get("w")
        assert torch.equal(out, make_full_tensor((16, 16)))
    finally:
        if put_mesh is not None:
            await put_mesh.stop()
        await ts.shutdown()
        await close_connections()


async def test_partial_commit_blocks_get():
    """A sharded key with a missing coordinate is invisible + error mentions it."""
    controller = await ts.initialize(
        num_storage_volumes=2,
        strategy=LocalRankStrategy(),
        storage_device="cpu",
    )
    put_mesh = None
    try:
        put_mesh = await _spawn_world(2, controller, "put")
        # rank 1 skips its shard put
        results = await asyncio.gather(
            put_mesh.handles[0].put_dtensor.call_one(
                "w", (16, 16), (2,), ["0"], False
            ),
            put_mesh.handles[1].put_dtensor.call_one(
                "w", (16, 16), (2,), ["0"], True
            ),
        )
        assert results == ["ok", "skipped"]
        with pytest.raises(KeyError, match="partially committed"):
            await ts.get("w")
        assert not await ts.exists("w")
        assert await ts.keys() == []
        # completing the commit makes it readable (both ranks participate in
        # the DTensor collectives; rank 0's re-put is an idempotent overwrite)
        res = await put_mesh.put_dtensor.call("w", (16, 16), (2,), ["0"])
        assert res == ["ok", "ok"]
        out = await ts.get("w")
        assert out.shape == (16, 16)
    finally:
        if put_mesh is not None:
            await put_mesh.stop()
        await ts.shutdown()
        await close_connections()


async def test_repush_with_new_mesh_replaces_old_epoch():
    """Re-pushing a key under a DIFFERENT mesh must fully replace the old
    sharding — stale old-layout shards must never serve reads."""
    controller = await ts.initialize(
        num_storage_volumes=4,
        strategy=LocalRankStrategy(),
        storage_device="cpu",
    )
    m2 = m4 = None
    try:
        m2 = await _spawn_world(2, controller, "ep2")
        m4 = await _spawn_world(4, controller, "ep4")
        res = await m2.put_dtensor.call("w", (16, 16), (2,), ["0"], False, 1.0)
        assert all(r == "ok" for r in res)
        # new epoch: same key, 4-way mesh, DIFFERENT values
        res = await m4.put_dtensor.call("w", (16, 16), (4,), ["1"], False, 2.0)
        assert all(r == "ok" for r in res)
        from tests.utils import make_full_tensor

        out = await ts.get("w")
        assert torch.equal(out, make_full_tensor((16, 16)) * 2.0)
    finally:
        for m in (m2, m4):
            if m is not None:
                await m.stop()
        await ts.shutdown()
        await close_connections()


async def test_fully_replicated_dtensor_demoted_to_plain():
    """Reference EP/MoE semantics (test_tensor_slice.py:400-506): a fully
    Replicate DTensor is stored as a PLAIN tensor — no commit gate, no
    shard bookkeeping — and reads back as the full tensor immediately."""
    controller = await ts.initialize(
        num_storage_volumes=2,
        strategy=LocalRankStrategy(),
        storage_device="cpu",
    )
    put_mesh = None
    try:
        put_mesh = await _spawn_world(2, controller, "rep")
        # ONLY rank 0 puts (ranks_to_skip semantics): a replicated entry
        # must be readable without every coordinate committing
        results = await asyncio.gather(
            put_mesh.handles[0].put_dtensor.call_one(
                "e", (8, 8), (2,), ["r"], False
            ),
            put_mesh.handles[1].put_dtensor.call_one(
                "e", (8, 8), (2,), ["r"], True
            ),
        )
        assert results == ["ok", "skipped"]
        out = await ts.get("e")  # readable despite rank 1 skipping
        assert out.shape == (8, 8)
        # index kind: plain TENSOR, not TENSOR_SLICE
        from torchstore_amd.controller import ObjectType

        located = await controller.locate.call_one(["e"])
        kinds = {i.object_type for i in located["e"].values()}
        assert kinds == {ObjectType.TENSOR}
    finally:
        if put_mesh is not None:
            await put_mesh.stop()
        await ts.shutdown()
        await close_connections()
