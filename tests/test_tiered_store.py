"""TieredStore: HBM-primary placement with a host overflow tier.

Fills the StorageImpl seam (reference storage_volume.py:102-143 names a
tiered backend as the explicit extension point but never builds one).
CPU tests drive the watermark accounting directly; placement semantics
are device-agnostic.
"""

import pytest
import torch

import torchstore_amd as ts
from torchstore_amd.storage import TieredStore
from torchstore_amd.strategy import SingletonStrategy
from torchstore_amd.types import Request, TensorSlice


def _req(key, slice_=None):
    return Request(key=key, tensor_slice=slice_)


def test_placement_watermark():
    s = TieredStore("cpu", capacity_bytes=1000, spill_device="cpu")
    a = torch.ones(100, dtype=torch.float32)  # 400 B
    s.put(_req("a"), a)
    assert s.primary_used == 400
    b = torch.ones(100, dtype=torch.float32)
    s.put(_req("b"), b)
    assert s.primary_used == 800
    c = torch.ones(100, dtype=torch.float32)  # would exceed -> spilled
    s.put(_req("c"), c)
    assert s.primary_used == 800
    # all three readable regardless of tier
    for k in ("a", "b", "c"):
        assert torch.equal(s.fetch(_req(k)), torch.ones(100))
    # freeing primary makes room again
    s.delete("a")
    assert s.primary_used == 400
    s.put(_req("d"), torch.ones(100, dtype=torch.float32))
    assert s.primary_used == 800


def test_overwrite_accounting_stable():
    s = TieredStore("cpu", capacity_bytes=1000, spill_device="cpu")
    for i in range(10):
        s.put(_req("k"), torch.full((100,), float(i)))
    assert s.primary_used == 400
    assert s.fetch(_req("k")).eq(9.0).all()
    s.reset()
    assert s.primary_used == 0 and s.keys() == []


def test_shard_entries_and_epoch_release():
    s = TieredStore("cpu", capacity_bytes=900, spill_device="cpu")

    def shard(coord, mesh=(2,)):
        return TensorSlice(
            offsets=(coord[0] * 50,), local_shape=(50,),
            global_shape=(100,), coordinates=coord, mesh_shape=mesh,
        )

    s.put(_req("w", shard((0,))), torch.ones(50))   # 200 B
    s.put(_req("w", shard((1,))), torch.ones(50))   # 400 B
    assert s.primary_used == 400
    s.put(_req("x"), torch.ones(100))               # 800 B
    spilled = torch.ones(50)
    s.put(_req("y"), spilled)                        # spills
    assert s.primary_used == 800
    # new sharding epoch releases the old shards' primary bytes
    new = TensorSlice(
        offsets=(0,), local_shape=(100,), global_shape=(100,),
        coordinates=(0,), mesh_shape=(1,),
    )
    s.put(_req("w", new), torch.ones(100))
    # old 2 shards (400 B) released, new shard placed... if it fit
    assert s.primary_used <= 900
    got = s.fetch(_req("w", new))
    assert got.numel() == 100


async def test_end_to_end_spill_roundtrip():
    """Store with a tiny primary capacity: later keys spill but remain
    fully readable through the normal client path."""
    await ts.initialize(
        num_storage_volumes=1,
        strategy=SingletonStrategy(),
        storage_device="cpu",
        storage_capacity_gb=1e-6,  # 1 kB primary: nearly everything spills
    )
    try:
        big = torch.randn(1024)  # 4 kB > capacity -> spilled
        await ts.put("spill/a", big)
        out = await ts.get("spill/a")
        assert torch.equal(out, big)
        dest = torch.zeros_like(big)
        await ts.get("spill/a", dest)
        assert torch.equal(dest, big)
    finally:
        await ts.shutdown()
