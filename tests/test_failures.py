"""Failure semantics: dead volumes, idempotent deletes, tolerant exists,
reset, and teardown robustness (SURVEY §5.3 — the reference's level of
fault handling plus cleaner surfacing)."""

import asyncio

import pytest
import torch

import torchstore_amd as ts
from torchstore_amd.strategy import SingletonStrategy
from torchstore_amd.transport import TransportType


async def test_dead_volume_surfaces_connection_error():
    await ts.initialize(
        num_storage_volumes=1,
        strategy=SingletonStrategy(transport=TransportType.RPC),
        storage_device="cpu",
    )
    try:
        await ts.put("k", torch.ones(4))
        # kill the volume process out from under the store
        session = ts.api._sessions["default"]
        for p in session.volume_mesh._procs:
            p.terminate()
            p.join(timeout=10)
        with pytest.raises((ConnectionError, OSError)):
            await ts.get("k")
    finally:
        await ts.shutdown()


async def test_exists_never_raises_for_missing():
    await ts.initialize(
        num_storage_volumes=1,
        strategy=SingletonStrategy(),
        storage_device="cpu",
    )
    try:
        assert not await ts.exists("missing/deeply/nested")
        assert await ts.keys("missing") == []
    finally:
        await ts.shutdown()


async def test_reset_clears_everything():
    await ts.initialize(
        num_storage_volumes=1,
        strategy=SingletonStrategy(),
        storage_device="cpu",
    )
    try:
        await ts.put("a", torch.ones(4))
        session = ts.api._sessions["default"]
        await session.controller.teardown.call_one()
        assert await ts.keys() == []
        with pytest.raises(KeyError):
            await ts.get("a")
        # the store is usable again after teardown
        await ts.put("b", torch.ones(2))
        assert (await ts.get("b")).eq(1).all()
    finally:
        await ts.shutdown()


async def test_controller_stats():
    await ts.initialize(
        num_storage_volumes=1, strategy=SingletonStrategy(), storage_device="cpu"
    )
    try:
        await ts.put("a", torch.ones(4))
        await ts.put("b", {"x": 1})
        session = ts.api._sessions["default"]
        stats = await session.controller.stats.call_one()
        assert stats["keys"] == 2 and stats["volumes"] == 1
    finally:
        await ts.shutdown()


async def test_double_shutdown_is_safe():
    await ts.initialize(
        num_storage_volumes=1, strategy=SingletonStrategy(), storage_device="cpu"
    )
    await ts.shutdown()
    await ts.shutdown()  # second time: no-op


async def test_client_before_initialize_raises():
    with pytest.raises(RuntimeError, match="not initialized"):
        ts.client("never-made")


async def test_two_named_stores_isolated():
    """Two stores in one process: key spaces and lifecycles are independent."""
    await ts.initialize(
        num_storage_volumes=1, strategy=SingletonStrategy(),
        storage_device="cpu", store_name="alpha",
    )
    await ts.initialize(
        num_storage_volumes=1, strategy=SingletonStrategy(),
        storage_device="cpu", store_name="beta",
    )
    try:
        await ts.put("k", torch.ones(4), store_name="alpha")
        await ts.put("k", torch.full((4,), 2.0), store_name="beta")
        assert (await ts.get("k", store_name="alpha")).eq(1).all()
        assert (await ts.get("k", store_name="beta")).eq(2).all()
        assert not await ts.exists("other", store_name="alpha")
        await ts.shutdown("alpha")
        # beta unaffected by alpha's teardown
        assert (await ts.get("k", store_name="beta")).eq(2).all()
    finally:
        await ts.shutdown("beta")
        await ts.shutdown("alpha")  # idempotent


@pytest.mark.slow
async def test_repeated_lifecycle_no_leaks():
    """Repeated bring-up/teardown in one process must not leak processes,
    connections, or store state."""
    import multiprocessing

    for i in range(4):
        await ts.initialize(
            num_storage_volumes=1, strategy=SingletonStrategy(),
            storage_device="cpu",
        )
        await ts.put("cycle", torch.full((8,), float(i)))
        assert (await ts.get("cycle")).eq(float(i)).all()
        await ts.shutdown()
    # no stray actor children left behind
    leftovers = [p for p in multiprocessing.active_children()]
    assert leftovers == [], leftovers


async def test_volume_stats_endpoint():
    """Per-volume observability: entry/byte counts and tier occupancy."""
    await ts.initialize(
        num_storage_volumes=1,
        strategy=SingletonStrategy(),
        storage_device="cpu",
        storage_capacity_gb=1e-6,  # 1 kB primary -> second put spills
    )
    try:
        await ts.put("a", torch.ones(128))          # 512 B -> primary
        await ts.put("b", torch.ones(1024))         # 4 kB -> spilled
        await ts.put("o", {"x": 1})
        c = ts.client()
        await c._ensure_volumes()
        v = next(iter(c._volumes.values()))
        stats = await v.handle.stats.call_one()
        agg = await ts.stats()
        assert agg["controller"]["keys"] == 3
        assert agg["volumes"][0]["entries"] == 3
        assert stats["entries"] == 3
        assert stats["tensor_entries"] == 2 and stats["object_entries"] == 1
        assert stats["tensor_bytes"] == 512 + 4096
        assert stats["tier_primary_used"] == 512
        assert stats["tier_capacity"] == 1000
    finally:
        await ts.shutdown()
