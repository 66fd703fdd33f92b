"""state_dict exchange: flatten/unflatten, commit marker, dtype cast,
strictness, and the multi-process DTensor path."""

import asyncio
import tempfile
import uuid

import pytest
import torch

import torchstore_amd as ts
from torchstore_amd.runtime import spawn_actors, close_connections
from torchstore_amd.strategy import LocalRankStrategy, SingletonStrategy
from tests.utils import DTensorWorker


async def _with_store(body, **kw):
    await ts.initialize(
        num_storage_volumes=kw.pop("num_volumes", 1),
        strategy=kw.pop("strategy", SingletonStrategy()),
        storage_device="cpu",
    )
    try:
        await body()
    finally:
        await ts.shutdown()


async def test_nested_roundtrip():
    async def body():
        sd = {
            "model": {
                "layers": [
                    {"w": torch.randn(8, 8), "b": torch.randn(8)},
                    {"w": torch.randn(8, 8), "b": torch.randn(8)},
                ],
                "norm": torch.randn(8),
            },
            "step": 17,
            "lr": 0.125,
        }
        await ts.put_state_dict(sd, "ck")
        out = await ts.get_state_dict("ck")
        assert torch.equal(out["model"]["layers"][0]["w"], sd["model"]["layers"][0]["w"])
        assert torch.equal(out["model"]["norm"], sd["model"]["norm"])
        assert out["step"] == 17 and out["lr"] == 0.125

    await _with_store(body)


async def test_inplace_with_user_state_dict():
    async def body():
        sd = {"a": torch.randn(16), "b": {"c": torch.randn(4, 4)}}
        await ts.put_state_dict(sd, "ck")
        dest = {"a": torch.zeros(16), "b": {"c": torch.zeros(4, 4)}}
        out = await ts.get_state_dict("ck", dest)
        assert torch.equal(out["a"], sd["a"])
        assert torch.equal(out["b"]["c"], sd["b"]["c"])
        # landed in place
        assert torch.equal(dest["a"], sd["a"])

    await _with_store(body)


async def test_transfer_dtype_cast():
    async def body():
        sd = {"w": torch.randn(64, 64, dtype=torch.float32), "n": 3}
        await ts.put_state_dict(sd, "ck", transfer_dtype=torch.bfloat16)
        out = await ts.get_state_dict("ck")
        assert out["w"].dtype == torch.bfloat16
        assert torch.equal(out["w"], sd["w"].to(torch.bfloat16))
        assert out["n"] == 3

    await _with_store(body)


async def test_missing_push_raises():
    async def body():
        with pytest.raises(RuntimeError, match="no state_dict was pushed"):
            await ts.get_state_dict("never")

    await _with_store(body)


async def test_partial_push_invisible():
    """Entries without the commit marker must look like 'no push'."""

    async def body():
        await ts.put("ck/w", torch.randn(4))  # entry but no MAPPING
        with pytest.raises(RuntimeError, match="no state_dict was pushed"):
            await ts.get_state_dict("ck")

    await _with_store(body)


async def test_strict_mismatch():
    async def body():
        await ts.put_state_dict({"a": torch.randn(4)}, "ck")
        # user-only entry: strict rejects
        with pytest.raises(KeyError, match="mapping mismatch"):
            await ts.get_state_dict("ck", {"zz": torch.zeros(4)})
        # stored-only entry (user dict is a subset): strict rejects too —
        # the reference asserts full mapping EQUALITY both ways
        await ts.put_state_dict({"a": torch.randn(4), "b": torch.randn(4)}, "ck2")
        with pytest.raises(KeyError, match="mapping mismatch"):
            await ts.get_state_dict("ck2", {"a": torch.zeros(4)})
        # non-strict: extra stored entries are fine, missing ones error later
        out = await ts.get_state_dict("ck", {"a": torch.zeros(4)}, strict=False)
        assert out["a"].abs().sum() > 0

    await _with_store(body)


async def test_overwrite_push():
    async def body():
        await ts.put_state_dict({"w": torch.ones(8)}, "ck")
        await ts.put_state_dict({"w": torch.full((8,), 5.0)}, "ck")
        out = await ts.get_state_dict("ck")
        assert out["w"].eq(5).all()

    await _with_store(body)


async def test_dtensor_state_dict_across_worlds():
    """2-rank world pushes a sharded state_dict; 2-rank world pulls with a
    different placement — through the real worker actors."""
    controller = await ts.initialize(
        num_storage_volumes=2,
        strategy=LocalRankStrategy(),
        storage_device="cpu",
    )
    put_m = get_m = None
    try:
        pg1 = tempfile.mktemp(prefix=f"sd-pg-{uuid.uuid4().hex[:6]}")
        pg2 = tempfile.mktemp(prefix=f"sd-pg-{uuid.uuid4().hex[:6]}")
        put_m = await asyncio.to_thread(
            spawn_actors, 2, DTensorWorker, "sd-put", 2, pg1, controller
        )
        get_m = await asyncio.to_thread(
            spawn_actors, 2, DTensorWorker, "sd-get", 2, pg2, controller
        )
        res = await put_m.put_state_dict.call("ck", (2,), ["0"], (16, 16))
        assert res == ["ok", "ok"]
        res = await get_m.get_state_dict.call("ck", (2,), ["1"], (16, 16))
        assert res == ["ok", "ok"]
    finally:
        for m in (put_m, get_m):
            if m is not None:
                await m.stop()
        await ts.shutdown()
        await close_connections()
