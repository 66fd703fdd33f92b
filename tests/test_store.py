"""End-to-end store semantics on CPU: put/get/delete/keys/exists,
objects, batches, both RPC and SHM transports."""

import asyncio
import os

import pytest
import torch

import torchstore_amd as ts
from torchstore_amd.strategy import SingletonStrategy
from torchstore_amd.transport import TransportType


@pytest.fixture(
    params=[
        TransportType.RPC,
        TransportType.SHARED_MEMORY,
        TransportType.GLOO,
    ]
)
def transport(request):
    return request.param


async def _with_store(transport, body, num_volumes=1):
    strategy = SingletonStrategy(transport=transport)
    await ts.initialize(
        num_storage_volumes=num_volumes,
        strategy=strategy,
        storage_device="cpu",
    )
    try:
        await body()
    finally:
        await ts.shutdown()


async def test_put_get_tensor(transport):
    async def body():
        t = torch.randn(64, 32)
        await ts.put("w", t)
        out = await ts.get("w")
        assert torch.equal(out, t)
        # in-place get into a preallocated dest
        dest = torch.zeros_like(t)
        got = await ts.get("w", dest)
        assert got is dest and torch.equal(dest, t)

    await _with_store(transport, body)


async def test_put_get_object(transport):
    async def body():
        await ts.put("config", {"lr": 0.1, "name": "x"})
        out = await ts.get("config")
        assert out == {"lr": 0.1, "name": "x"}

    await _with_store(transport, body)


async def test_overwrite(transport):
    async def body():
        await ts.put("k", torch.ones(8))
        await ts.put("k", torch.full((8,), 2.0))
        out = await ts.get("k")
        assert out.eq(2).all()
        # shape change on overwrite
        await ts.put("k", torch.full((4, 4), 3.0))
        out = await ts.get("k")
        assert out.shape == (4, 4) and out.eq(3).all()

    await _with_store(transport, body)


async def test_batches(transport):
    async def body():
        items = {f"t{i}": torch.randn(16, 16) for i in range(8)}
        items["meta"] = {"step": 3}
        await ts.put_batch(items)
        out = await ts.get_batch({k: None for k in items})
        for k, v in items.items():
            if isinstance(v, torch.Tensor):
                assert torch.equal(out[k], v)
            else:
                assert out[k] == v

    await _with_store(transport, body)


async def test_missing_key_raises(transport):
    async def body():
        with pytest.raises(KeyError):
            await ts.get("nope")

    await _with_store(transport, body)


async def test_delete_and_exists(transport):
    async def body():
        await ts.put("a/b", torch.ones(4))
        assert await ts.exists("a/b")
        assert not await ts.exists("a/c")
        await ts.delete("a/b")
        assert not await ts.exists("a/b")
        with pytest.raises(KeyError):
            await ts.get("a/b")
        with pytest.raises(KeyError):
            await ts.delete("a/b")
        await ts.delete("a/b", missing_ok=True)

    await _with_store(transport, body)


async def test_keys_prefix(transport):
    async def body():
        await ts.put_batch(
            {
                "m/l0/w": torch.ones(2),
                "m/l1/w": torch.ones(2),
                "other": torch.ones(2),
            }
        )
        ks = sorted(await ts.keys("m"))
        assert ks == ["m/l0/w", "m/l1/w"]
        assert sorted(await ts.keys()) == ["m/l0/w", "m/l1/w", "other"]
        await ts.delete_batch(["m/l0/w", "m/l1/w"])
        assert await ts.keys("m") == []

    await _with_store(transport, body)


async def test_slice_get_of_full_tensor(transport):
    """Fetching a sub-region of a stored full tensor via a LocalShard dest."""
    from torchstore_amd.types import LocalShard, TensorSlice

    async def body():
        t = torch.arange(64, dtype=torch.float32).reshape(8, 8)
        await ts.put("big", t)
        dest = LocalShard(
            tensor=torch.zeros(3, 5),
            slice=TensorSlice(
                offsets=(2, 1), local_shape=(3, 5), global_shape=(8, 8),
                coordinates=(0,), mesh_shape=(1,),
            ),
        )
        await ts.get("big", dest)
        assert torch.equal(dest.tensor, t[2:5, 1:6])

    await _with_store(transport, body)


async def test_two_volumes_round_robin():
    """Multiple volumes with rank strategy; same process writes to one."""
    from torchstore_amd.strategy import LocalRankStrategy

    strategy = LocalRankStrategy(transport=TransportType.RPC)
    await ts.initialize(
        num_storage_volumes=2, strategy=strategy, storage_device="cpu"
    )
    try:
        await ts.put("x", torch.ones(4))
        assert torch.equal(await ts.get("x"), torch.ones(4))
    finally:
        await ts.shutdown()


async def test_host_strategy():
    """Volume id = hostname; clients pick their host's volume."""
    from torchstore_amd.strategy import HostStrategy

    await ts.initialize(
        num_storage_volumes=1,
        strategy=HostStrategy(),
        storage_device="cpu",
    )
    try:
        c = ts.client()
        await c._ensure_volumes()
        import socket

        vid = next(iter(c._volumes))
        assert vid == (os.environ.get("HOSTNAME") or socket.gethostname())
        ref = c._volume_ref(vid)
        assert ref.is_local
        await ts.put("h", torch.full((4,), 3.0))
        assert (await ts.get("h")).eq(3).all()
    finally:
        await ts.shutdown()


async def test_large_tensor_roundtrip(transport):
    async def body():
        t = torch.randn(32 << 20 // 4)  # 32 Mi floats = 128 MB
        await ts.put("big", t)
        out = await ts.get("big")
        assert torch.equal(out, t)

    await _with_store(transport, body)


@pytest.mark.parametrize(
    "strategy_name,num_volumes",
    [("singleton", 1), ("local_rank", 2), ("host", 1)],
)
@pytest.mark.parametrize(
    "matrix_transport", [TransportType.RPC, TransportType.SHARED_MEMORY]
)
async def test_transport_strategy_matrix(strategy_name, num_volumes, matrix_transport):
    """The reference's transport x strategy product (tests/utils.py:63-69):
    core put/get/batch semantics must hold for every combination."""
    from torchstore_amd.strategy import (
        HostStrategy,
        LocalRankStrategy,
        SingletonStrategy,
    )

    strategy = {
        "singleton": SingletonStrategy,
        "local_rank": LocalRankStrategy,
        "host": HostStrategy,
    }[strategy_name](transport=matrix_transport)
    await ts.initialize(
        num_storage_volumes=num_volumes, strategy=strategy, storage_device="cpu"
    )
    try:
        t = torch.randn(32, 32)
        await ts.put("m/x", t)
        assert torch.equal(await ts.get("m/x"), t)
        await ts.put_batch({"m/y": torch.ones(4), "m/obj": {"a": 1}})
        out = await ts.get_batch({"m/y": None, "m/obj": None})
        assert out["m/y"].eq(1).all() and out["m/obj"] == {"a": 1}
        assert sorted(await ts.keys("m")) == ["m/obj", "m/x", "m/y"]
        await ts.delete("m/x")
        assert not await ts.exists("m/x")
    finally:
        await ts.shutdown()


async def test_get_batch_plan_cache():
    """Steady-state loops reuse the fetch plan (one verify RPC instead of
    locate + planning); any layout change invalidates it."""

    async def body():
        from torchstore_amd.types import LocalShard, TensorSlice

        a = torch.randn(64, 8)
        b = torch.randn(32)
        await ts.put_batch({"pc/a": a, "pc/b": b})
        da, db = torch.zeros_like(a), torch.zeros_like(b)
        c = ts.client()
        fetches = {"pc/a": da, "pc/b": db}
        out = await c.get_batch(fetches)
        assert torch.equal(da, a) and torch.equal(db, b)
        assert len(c._plan_cache) == 1
        # same dests again: cached plan path
        a2 = torch.randn(64, 8)
        await ts.put("pc/a", a2)  # same layout -> fingerprint unchanged
        out = await c.get_batch(fetches)
        assert torch.equal(da, a2)
        assert len(c._plan_cache) == 1
        # different dest objects: plan must NOT be reused
        da3 = torch.zeros_like(a)
        await c.get_batch({"pc/a": da3, "pc/b": db})
        assert torch.equal(da3, a2)
        # layout change (plain tensor -> sharded) invalidates the fingerprint
        shard0 = LocalShard(
            tensor=a2[:32].clone(),
            slice=TensorSlice((0, 0), (32, 8), (64, 8), (0,), (2,)),
        )
        shard1 = LocalShard(
            tensor=a2[32:].clone(),
            slice=TensorSlice((32, 0), (32, 8), (64, 8), (1,), (2,)),
        )
        await ts.put("pc/a", shard0)
        await ts.put("pc/a", shard1)
        da.zero_()
        await c.get_batch(fetches)  # replans, still correct
        assert torch.equal(da, a2)
        # delete invalidates too: cached entry must not mask the KeyError
        await ts.delete("pc/b")
        with pytest.raises(KeyError):
            await c.get_batch(fetches)

    await _with_store(TransportType.RPC, body)


async def test_keys_dotted_prefix_api():
    """API-level dotted-prefix filtering: keys('sd/model') matches dotted
    module paths by component (reference StringTrie('.') semantics)."""

    async def body():
        await ts.put_state_dict(
            {"model": {"layers": {"0": torch.ones(2), "1": torch.ones(2)}},
             "opt": {"lr": 0.1}},
            "sd",
        )
        ks = await ts.keys("sd/model.layers")
        assert sorted(ks) == ["sd/model.layers.0", "sd/model.layers.1"]
        ks2 = await ts.keys("sd/model")
        assert sorted(ks2) == ["sd/model.layers.0", "sd/model.layers.1"]
        # component boundary: "sd/mod" matches nothing
        assert await ts.keys("sd/mod") == []

    await _with_store(TransportType.RPC, body)


async def test_reput_across_volumes_serves_newest(monkeypatch):
    """A re-put routed to a DIFFERENT volume (client identity changed)
    must win over the stale copy on the old volume — the controller's
    write sequence decides, never locality/lexical order."""
    from torchstore_amd.strategy import LocalRankStrategy

    await ts.initialize(
        num_storage_volumes=2,
        strategy=LocalRankStrategy(),
        storage_device="cpu",
    )
    try:
        monkeypatch.setenv("RANK", "0")
        await ts.put("x", torch.zeros(32))          # volume 0
        monkeypatch.setenv("RANK", "1")
        await ts.put("x", torch.ones(32))           # volume 1 (newer)
        for r in ("0", "1"):
            monkeypatch.setenv("RANK", r)
            out = await ts.get("x")
            assert out.eq(1.0).all(), f"stale copy served for reader rank {r}"
        # objects too
        monkeypatch.setenv("RANK", "1")
        await ts.put("o", {"v": 1})
        monkeypatch.setenv("RANK", "0")
        await ts.put("o", {"v": 2})
        assert (await ts.get("o"))["v"] == 2
        # keys()/exists see the newest kind: overwrite a half-committed
        # shard set with a plain tensor -> readable immediately
        from torchstore_amd.types import LocalShard, TensorSlice

        monkeypatch.setenv("RANK", "0")
        await ts.put("h", LocalShard(
            tensor=torch.zeros(8, 4),
            slice=TensorSlice((0, 0), (8, 4), (16, 4), (0,), (2,)),
        ))  # half-committed: coordinate (1,) missing
        assert not await ts.exists("h")
        monkeypatch.setenv("RANK", "1")
        await ts.put("h", torch.full((16, 4), 7.0))  # plain overwrite
        assert await ts.exists("h")
        assert (await ts.get("h")).eq(7.0).all()
    finally:
        await ts.shutdown()


async def test_noncontiguous_put(transport):
    """Non-contiguous values (transposes, slices) store their materialized
    content (reference test_store.py:556)."""

    async def body():
        base = torch.randn(32, 48)
        views = {
            "nc/t": base.t(),              # stride-swapped
            "nc/slice": base[::2, 1:17],   # offset + both dims strided
            "nc/chan": base.unsqueeze(0).expand(3, 32, 48)[1],
        }
        await ts.put_batch(views)
        for k, v in views.items():
            out = await ts.get(k)
            assert torch.equal(out, v.contiguous()), k
            dest = torch.zeros(v.shape)
            await ts.get(k, dest)
            assert torch.equal(dest, v), k

    await _with_store(transport, body)


async def test_get_batch_all_or_nothing(transport):
    """A missing key fails the whole get_batch (reference locate raises
    before any transport work begins — no partial results)."""

    async def body():
        await ts.put_batch({"ab/x": torch.ones(4), "ab/y": torch.ones(4)})
        with pytest.raises(KeyError, match="does not exist"):
            await ts.get_batch({"ab/x": None, "ab/missing": None})
        # the present keys remain readable afterwards
        out = await ts.get_batch({"ab/x": None, "ab/y": None})
        assert out["ab/x"].eq(1).all() and out["ab/y"].eq(1).all()

    await _with_store(transport, body)
