"""CPU rehearsal of the multi-rank bench flow (what the driver runs at
N=2..8 on GPUs): sharded Llama state_dict push (FSDP layout) + barrier +
resharding pull (TP layout) across 2 ranks over gloo."""

import asyncio
import os
import tempfile
import uuid

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


class BenchWorker(Actor):
    """One bench rank: builds src (fsdp) and dst (tp) shards like bench.py."""

    def __init__(self, world, pg_file, controller):
        self.rank = actor_context().rank
        self.world = world
        os.environ["RANK"] = str(self.rank)
        dist.init_process_group(
            "gloo", init_method=f"file://{pg_file}",
            rank=self.rank, world_size=world,
        )
        api.attach(controller, LocalRankStrategy())
        from torch.distributed.device_mesh import init_device_mesh
        from torchstore_amd.models import llama

        self.mesh = init_device_mesh("cpu", (world,))
        self.src = llama.make_sharded_state_dict(
            self.mesh, llama.fsdp_placement, device="cpu", layers=1,
            seed=100 + self.rank, dtype=torch.float32, scale=8,
        )
        self.dst = llama.make_sharded_state_dict(
            self.mesh, llama.tp_placement, device="cpu", layers=1, zero=True,
            dtype=torch.float32, scale=8,
        )

    @endpoint
    async def step(self):
        await api.put_state_dict(self.src, "bench")
        dist.barrier()
        await api.get_state_dict("bench", self.dst)
        return "ok"

    @endpoint
    def verify(self):
        """Reassemble both layouts' full tensors and compare."""
        for name in self.src:
            s = self.src[name]
            d = self.dst[name]
            sf = s.full_tensor() if hasattr(s, "full_tensor") else s
            df = d.full_tensor() if hasattr(d, "full_tensor") else d
            if not torch.equal(sf, df):
                raise AssertionError(f"rank {self.rank}: mismatch in {name}")
        return "ok"

    def teardown_local(self):
        if dist.is_initialized():
            dist.destroy_process_group()


async def test_bench_flow_2rank_cpu():
    controller = await ts.initialize(
        num_storage_volumes=2,
        strategy=LocalRankStrategy(),
        storage_device="cpu",
    )
    mesh = None
    try:
        pg_file = tempfile.mktemp(prefix=f"bench-pg-{uuid.uuid4().hex[:6]}")
        mesh = await asyncio.to_thread(
            spawn_actors, 2, BenchWorker, "benchw", 2, pg_file, controller,
            timeout=240,
        )
        for _ in range(2):  # warm + steady step (exercises overwrite reuse)
            res = await mesh.step.call()
            assert res == ["ok", "ok"]
        res = await mesh.verify.call()
        assert res == ["ok", "ok"]
    finally:
        if mesh is not None:
            await mesh.stop()
        await ts.shutdown()
        await close_connections()
