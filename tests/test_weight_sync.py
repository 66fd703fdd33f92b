"""Direct weight sync plan/pull logic on CPU via the fake memory codec
(mirrors the reference's MockRDMABuffer test strategy)."""

import pytest
import torch

import torchstore_amd as ts
from torchstore_amd.strategy import SingletonStrategy
from torchstore_amd.transport import TransportType
from torchstore_amd.types import TensorSlice
from torchstore_amd import weight_sync
from torchstore_amd.weight_sync import (
    DirectWeightSyncDest,
    DirectWeightSyncSource,
    FakeMemoryCodec,
    WeightHandle,
)


@pytest.fixture
def fake_codec():
    codec = FakeMemoryCodec()
    weight_sync.set_codec(codec)
    yield codec
    weight_sync.set_codec(None)


async def _with_store(body):
    await ts.initialize(
        num_storage_volumes=1,
        strategy=SingletonStrategy(transport=TransportType.RPC),
        storage_device="cpu",
    )
    try:
        await body()
    finally:
        await ts.shutdown()


async def test_exact_match_zero_copy_and_liveness(fake_codec):
    async def body():
        c = ts.client()
        w = torch.randn(64, 64)
        b = torch.randn(64)
        src = DirectWeightSyncSource(c, "sync", rank=0, world_size=1)
        await src.push({"m": {"w": w, "b": b}})

        dst_sd = {"m": {"w": torch.zeros(64, 64), "b": torch.zeros(64)}}
        dest = DirectWeightSyncDest(c, "sync")
        await dest.pull(dst_sd)
        assert torch.equal(dst_sd["m"]["w"], w)
        assert torch.equal(dst_sd["m"]["b"], b)

        # optimizer-style in-place update is visible WITHOUT re-push
        with torch.no_grad():
            w.add_(1.0)
        await dest.pull(dst_sd)
        assert torch.equal(dst_sd["m"]["w"], w)
        # plan was cached: second pull did not refetch handles
        assert fake_codec.read_count == 4

    await _with_store(body)


async def test_sharded_to_full_reshard(fake_codec):
    """Two Shard(0) source shards assemble into a full dest tensor."""

    async def body():
        c = ts.client()
        full = torch.arange(64, dtype=torch.float32).reshape(8, 8)
        handles = []
        for r in range(2):
            shard = full[r * 4 : (r + 1) * 4].clone()
            handles.append(
                WeightHandle(
                    name="w",
                    desc=fake_codec.export(shard),
                    slice=TensorSlice(
                        offsets=(r * 4, 0), local_shape=(4, 8),
                        global_shape=(8, 8), coordinates=(r,), mesh_shape=(2,),
                    ),
                )
            )
        await c.put("sync/rank_0", handles)
        await c.put("sync/num_ranks", 1)

        dst = {"w": torch.zeros(8, 8)}
        dest = DirectWeightSyncDest(c, "sync")
        await dest.pull(dst)
        assert torch.equal(dst["w"], full)

    await _with_store(body)


async def test_partial_overlap_column_shards(fake_codec):
    """Column-sharded (TP-style) sources into a full dest: every scatter
    destination view is strided, exercising the reshard copy path."""

    async def body():
        c = ts.client()
        full = torch.randn(8, 8)
        handles = []
        for r in range(2):
            shard = full[:, r * 4 : (r + 1) * 4].clone()
            handles.append(
                WeightHandle(
                    name="w",
                    desc=fake_codec.export(shard),
                    slice=TensorSlice(
                        offsets=(0, r * 4), local_shape=(8, 4),
                        global_shape=(8, 8), coordinates=(r,), mesh_shape=(2,),
                    ),
                )
            )
        await c.put("sync/rank_0", handles)
        await c.put("sync/num_ranks", 1)

        dest = DirectWeightSyncDest(c, "sync")
        dst = {"w": torch.zeros(8, 8)}
        await dest.pull(dst)
        assert torch.equal(dst["w"], full)
        # column shards land via pitched reads of only the overlap bytes
        assert all(op.kind == "2d" for op in dest._plan)
        assert sum(op.nbytes for op in dest._plan) == full.numel() * 4

    await _with_store(body)


async def test_replicated_dedup(fake_codec):
    async def body():
        c = ts.client()
        t = torch.randn(4, 4)
        handles = []
        for r in range(3):  # 3 replicas of the same region
            handles.append(
                WeightHandle(
                    name="w",
                    desc=fake_codec.export(t.clone()),
                    slice=TensorSlice(
                        offsets=(0, 0), local_shape=(4, 4), global_shape=(4, 4),
                        coordinates=(r,), mesh_shape=(3,),
                    ),
                )
            )
        await c.put("sync/rank_0", handles)
        await c.put("sync/num_ranks", 1)
        dest = DirectWeightSyncDest(c, "sync")
        dst = {"w": torch.zeros(4, 4)}
        await dest.pull(dst)
        assert fake_codec.read_count == 1  # deduped to one read
        assert torch.equal(dst["w"], t)

    await _with_store(body)


async def test_transfer_dtype_staging_and_refresh(fake_codec):
    async def body():
        c = ts.client()
        master = torch.randn(32, 32, dtype=torch.float32)
        src = DirectWeightSyncSource(
            c, "sync", transfer_dtype=torch.bfloat16, rank=0, world_size=1
        )
        await src.push({"w": master})

        dst = {"w": torch.zeros(32, 32, dtype=torch.bfloat16)}
        dest = DirectWeightSyncDest(c, "sync")
        await dest.pull(dst)
        assert torch.equal(dst["w"], master.to(torch.bfloat16))

        # stale until refresh (staging buffer holds the old cast)
        with torch.no_grad():
            master.mul_(2.0)
        await dest.pull(dst)
        assert not torch.equal(dst["w"], master.to(torch.bfloat16))
        await src.push({"w": master})  # push == refresh after registration
        await dest.pull(dst)
        assert torch.equal(dst["w"], master.to(torch.bfloat16))

    await _with_store(body)


async def test_dtype_mismatch_raises(fake_codec):
    async def body():
        c = ts.client()
        src = DirectWeightSyncSource(c, "sync", rank=0, world_size=1)
        await src.push({"w": torch.randn(8, 8)})
        dest = DirectWeightSyncDest(c, "sync")
        with pytest.raises(TypeError):
            await dest.pull({"w": torch.zeros(8, 8, dtype=torch.bfloat16)})

    await _with_store(body)


async def test_localshard_source_and_dest(fake_codec):
    """The bench's mesh-free layout path: LocalShard entries on BOTH ends
    (regression: these were silently skipped as non-Tensor leaves)."""
    from torchstore_amd.types import LocalShard
    from torchstore_amd.weight_sync import DirectWeightSyncSource

    async def body():
        c = ts.client()
        full = torch.randn(8, 4)
        sources = []
        for r in range(2):
            src = DirectWeightSyncSource(c, "sync", rank=r, world_size=2)
            sd = {
                "w": LocalShard(
                    tensor=full[r * 4 : (r + 1) * 4].clone(),
                    slice=TensorSlice(
                        offsets=(r * 4, 0), local_shape=(4, 4),
                        global_shape=(8, 4), coordinates=(r,), mesh_shape=(2,),
                    ),
                )
            }
            await src.push(sd)
            sources.append(src)

        dest = DirectWeightSyncDest(c, "sync")
        # dest wants rows 2..6 — spans both source shards
        dst = {
            "w": LocalShard(
                tensor=torch.zeros(4, 4),
                slice=TensorSlice(
                    offsets=(2, 0), local_shape=(4, 4), global_shape=(8, 4),
                    coordinates=(0,), mesh_shape=(2,),
                ),
            )
        }
        await dest.pull(dst)
        assert len(dest._plan) == 2
        assert torch.equal(dst["w"].tensor, full[2:6])

    await _with_store(body)


async def test_missing_source_raises(fake_codec):
    async def body():
        c = ts.client()
        dest = DirectWeightSyncDest(c, "nope")
        with pytest.raises(RuntimeError, match="no direct weight sync source"):
            await dest.pull({"w": torch.zeros(2)})

    await _with_store(body)


async def test_coverage_gap_raises(fake_codec):
    """Source shards that do NOT tile the wanted region must raise instead
    of silently leaving stale weights (verify coverage)."""
    from torchstore_amd.types import LocalShard

    async def body():
        c = ts.client()
        # source registers only rows 0..32 of a 64-row parameter
        top = torch.randn(32, 16)
        src = DirectWeightSyncSource(c, "gap", rank=0, world_size=1)
        await src.push({
            "w": LocalShard(
                tensor=top,
                slice=TensorSlice(
                    offsets=(0, 0), local_shape=(32, 16),
                    global_shape=(64, 16), coordinates=(0,), mesh_shape=(2,),
                ),
            )
        })
        dest = DirectWeightSyncDest(c, "gap")
        full = {"w": torch.zeros(64, 16)}
        with pytest.raises(RuntimeError, match="do not tile"):
            await dest.pull(full)
        # a dest asking only for the covered half succeeds
        half = {
            "w": LocalShard(
                tensor=torch.zeros(32, 16),
                slice=TensorSlice(
                    offsets=(0, 0), local_shape=(32, 16),
                    global_shape=(64, 16), coordinates=(0,), mesh_shape=(2,),
                ),
            )
        }
        await dest.pull(half)
        assert torch.equal(half["w"].tensor, top)

    await _with_store(body)


async def test_plan_cache_dest_identity(fake_codec):
    """The cached plan is keyed by IDENTITY of the dest dict: a different
    dict (even with equal shapes) forces a rebuild into the new tensors."""

    async def body():
        c = ts.client()
        w = torch.randn(16, 16)
        src = DirectWeightSyncSource(c, "ident", rank=0, world_size=1)
        await src.push({"w": w})
        dest = DirectWeightSyncDest(c, "ident")
        d1 = {"w": torch.zeros(16, 16)}
        await dest.pull(d1)
        assert torch.equal(d1["w"], w)
        n_reads = fake_codec.read_count
        await dest.pull(d1)  # same dict -> cached plan
        d2 = {"w": torch.zeros(16, 16)}
        await dest.pull(d2)  # new dict -> rebuilt plan targets d2's tensor
        assert torch.equal(d2["w"], w)
        assert fake_codec.read_count == n_reads + 2

    await _with_store(body)


async def test_direct_rdma_alias(fake_codec):
    """Reference-compat kwarg: direct_rdma=True behaves as direct=True."""

    async def body():
        w = torch.randn(8, 8)
        await ts.put_state_dict({"w": w}, "alias", direct_rdma=True,
                                rank=0, world_size=1)
        dest = {"w": torch.zeros(8, 8)}
        out = await ts.get_state_dict("alias", dest, direct_rdma=True)
        assert torch.equal(out["w"], w)

    await _with_store(body)
