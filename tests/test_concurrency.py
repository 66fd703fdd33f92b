"""Concurrent multi-client traffic against one store: interleaved puts,
gets, overwrites and deletes must stay linearizable per key."""

import asyncio
import os
import tempfile

import torch

import torchstore_amd as ts
from torchstore_amd import api
from torchstore_amd.runtime import (
    Actor,
    actor_context,
    close_connections,
    endpoint,
    spawn_actors,
)
from torchstore_amd.strategy import SingletonStrategy


class Hammer(Actor):
    def __init__(self, controller):
        self.rank = actor_context().rank
        api.attach(controller, SingletonStrategy())

    @endpoint
    async def run(self, rounds: int):
        # own keyspace: full rounds of put/get/overwrite/delete
        for i in range(rounds):
            key = f"h{self.rank}/k{i % 4}"
            val = torch.full((256,), float(self.rank * 1000 + i))
            await api.put(key, val)
            out = await api.get(key)
            if not torch.equal(out, val):
                raise AssertionError(f"{key}: read back wrong value")
            if i % 3 == 2:
                await api.delete(key)
        # shared key: last-writer-wins, value always self-consistent
        for i in range(rounds):
            stamp = float(self.rank * 10000 + i)
            await api.put("shared", torch.full((64,), stamp))
            out = await api.get("shared")
            u = out.unique()
            if u.numel() != 1:
                raise AssertionError(f"torn read on shared key: {u}")
        return "ok"


async def test_concurrent_clients_one_volume():
    controller = await ts.initialize(
        num_storage_volumes=1,
        strategy=SingletonStrategy(),
        storage_device="cpu",
    )
    mesh = None
    try:
        mesh = await asyncio.to_thread(
            spawn_actors, 3, Hammer, "hammer", controller
        )
        res = await mesh.run.call(12)
        assert res == ["ok", "ok", "ok"]
        # store still healthy afterwards
        await ts.put("after", torch.ones(4))
        assert (await ts.get("after")).eq(1).all()
    finally:
        if mesh is not None:
            await mesh.stop()
        await ts.shutdown()
        await close_connections()
