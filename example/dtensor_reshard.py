"""DTensor resharding through the store: a 4-rank world pushes Shard(0)
shards, then the SAME ranks read the key back as Shard(1) — the store
computes the slice intersections and reassembles each rank's new shard.

Run:  python example/dtensor_reshard.py
(it spawns its own 4-process gloo world on CPU; on a GPU node swap the
mesh device for "cuda" and the volumes become HBM-resident)
"""

import asyncio
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

import torchstore_amd as ts
from torchstore_amd import api
from torchstore_amd.runtime import (
    Actor,
    actor_context,
    close_connections,
    endpoint,
    spawn_actors,
)
from torchstore_amd.strategy import LocalRankStrategy

WORLD = 4


class Rank(Actor):
    def __init__(self, pg_file, controller):
        rank = actor_context().rank
        os.environ["RANK"] = str(rank)
        dist.init_process_group(
            "gloo", init_method=f"file://{pg_file}", rank=rank, world_size=WORLD
        )
        api.attach(controller, LocalRankStrategy())

    @endpoint
    async def push_and_reshard(self):
        from torch.distributed.device_mesh import init_device_mesh
        from torch.distributed.tensor import Shard, distribute_tensor

        mesh = init_device_mesh("cpu", (WORLD,))
        full = torch.arange(64 * 64, dtype=torch.float32).reshape(64, 64)

        row_sharded = distribute_tensor(full, mesh, [Shard(0)])
        await api.put("w", row_sharded)

        # the commit gate makes "w" readable only once EVERY coordinate has
        # stored its shard — barrier before reading, like any SPMD step
        dist.barrier()

        col_dest = distribute_tensor(torch.zeros_like(full), mesh, [Shard(1)])
        out = await api.get("w", col_dest)

        expect = distribute_tensor(full, mesh, [Shard(1)])
        assert torch.equal(out.to_local(), expect.to_local())
        return f"rank {dist.get_rank()}: reshard Shard(0)->Shard(1) ok"

    def teardown_local(self):
        dist.destroy_process_group()


async def main():
    controller = await ts.initialize(
        num_storage_volumes=WORLD,
        strategy=LocalRankStrategy(),
        storage_device="cpu",
    )
    pg_file = tempfile.mktemp(prefix="example-pg")
    mesh = await asyncio.to_thread(
        spawn_actors, WORLD, Rank, "rank", pg_file, controller
    )
    try:
        for line in await mesh.push_and_reshard.call():
            print(line)
    finally:
        await mesh.stop()
        await ts.shutdown()
        await close_connections()


if __name__ == "__main__":
    asyncio.run(main())
