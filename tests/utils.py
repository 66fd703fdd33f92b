"""Shared test helpers: DTensor worker actors for multi-process resharding
tests on CPU (gloo over a file store), mirroring the reference's test
strategy: a "put world" mesh and an independent "get world"
mesh emulate two jobs (trainer / inference fleet) sharing one store.
"""

from __future__ import annotations

import os
from typing import List, Optional, Sequence, Tuple

import torch
import torch.distributed as dist

from torchstore_amd import api
from torchstore_amd.runtime import Actor, ActorHandle, endpoint, actor_context
from torchstore_amd.strategy import LocalRankStrategy


def make_full_tensor(shape=(16, 16), dtype=torch.float32) -> torch.Tensor:
    """Deterministic asymmetric tensor every process can recompute."""
    numel = 1
    for s in shape:
        numel *= s
    return torch.arange(numel, dtype=dtype).reshape(shape) / 7.0


def _placements(specs: Sequence[str]):
    from torch.distributed.tensor import Replicate, Shard

    out = []
    for s in specs:
        if s == "r":
            out.append(Replicate())
        else:
            out.append(Shard(int(s)))
    return out


class DTensorWorker(Actor):
    """One rank of a DTensor world; forms a gloo PG over a file store."""

    def __init__(
        self,
        world_size: int,
        pg_file: str,
        controller: ActorHandle,
        transport=None,
    ):
        self.rank = actor_context().rank
        self.world_size = world_size
        os.environ["RANK"] = str(self.rank)
        os.environ["LOCAL_RANK"] = str(self.rank)
        dist.init_process_group(
            "gloo",
            init_method=f"file://{pg_file}",
            rank=self.rank,
            world_size=world_size,
        )
        api.attach(controller, LocalRankStrategy(transport=transport))

    def _mesh(self, mesh_shape: Tuple[int, ...]):
        from torch.distributed.device_mesh import init_device_mesh

        return init_device_mesh("cpu", tuple(mesh_shape))

    def _dtensor(self, full, mesh_shape, placements):
        from torch.distributed.tensor import distribute_tensor

        mesh = self._mesh(mesh_shape)
        return distribute_tensor(full, mesh, _placements(placements))

    @endpoint
    async def put_dtensor(
        self,
        key: str,
        shape: Tuple[int, ...],
        mesh_shape: Tuple[int, ...],
        placements: Sequence[str],
        skip: bool = False,
        value_scale: float = 1.0,
    ):
        full = make_full_tensor(shape) * value_scale
        dt = self._dtensor(full, mesh_shape, placements)
        if skip:
            return "skipped"
        await api.put(key, dt)
        return "ok"

    @endpoint
    async def get_dtensor(
        self,
        key: str,
        shape: Tuple[int, ...],
        mesh_shape: Tuple[int, ...],
        placements: Sequence[str],
    ):
        """Fetch into a zeroed DTensor of this layout; verify vs recomputed truth."""
        full = make_full_tensor(shape)
        expected = self._dtensor(full, mesh_shape, placements)
        dest = self._dtensor(torch.zeros_like(full), mesh_shape, placements)
        got = await api.get(key, dest)
        if not torch.equal(got.to_local(), expected.to_local()):
            raise AssertionError(
                f"rank {self.rank}: reshard mismatch for {key}: "
                f"{got.to_local()} vs {expected.to_local()}"
            )
        return "ok"

    @endpoint
    async def put_plain(self, key: str, value):
        await api.put(key, value)
        return "ok"

    @endpoint
    async def get_full(self, key: str, shape: Tuple[int, ...]):
        out = await api.get(key)
        full = make_full_tensor(shape)
        if not torch.equal(out, full):
            raise AssertionError(f"full get mismatch: {out} vs {full}")
        return "ok"

    @endpoint
    async def put_state_dict(self, sd_key: str, mesh_shape, placements, shape):
        full = make_full_tensor(shape)
        dt = self._dtensor(full, mesh_shape, placements)
        sd = {"model": {"weight": dt, "step": 7}}
        await api.put_state_dict(sd, sd_key)
        return "ok"

    @endpoint
    async def get_state_dict(self, sd_key: str, mesh_shape, placements, shape):
        full = make_full_tensor(shape)
        expected = self._dtensor(full, mesh_shape, placements)
        dest = self._dtensor(torch.zeros_like(full), mesh_shape, placements)
        sd = {"model": {"weight": dest, "step": 0}}
        out = await api.get_state_dict(sd_key, sd)
        got = out["model"]["weight"]
        if not torch.equal(got.to_local(), expected.to_local()):
            raise AssertionError("state_dict reshard mismatch")
        if out["model"]["step"] != 7:
            raise AssertionError(f"object entry mismatch: {out['model']['step']}")
        return "ok"

    @endpoint
    async def dcp_save_and_push(
        self, ckpt_path: str, sd_key: str, shape, mesh_shape, placements
    ):
        import torch.distributed.checkpoint as dcp

        full = make_full_tensor(shape)
        dt = self._dtensor(full, mesh_shape, placements)
        sd = {"w": dt, "meta": 42}
        dcp.save({"w": dt}, checkpoint_id=ckpt_path)
        await api.put_state_dict(sd, sd_key)
        return "ok"

    @endpoint
    async def dcp_load_and_compare(
        self, ckpt_path: str, sd_key: str, shape, mesh_shape, placements
    ):
        """Ground truth: DCP's own resharding load must equal our pull."""
        import torch.distributed.checkpoint as dcp

        full = make_full_tensor(shape)
        dcp_dest = self._dtensor(torch.zeros_like(full), mesh_shape, placements)
        dcp_sd = {"w": dcp_dest}
        dcp.load(dcp_sd, checkpoint_id=ckpt_path)

        ts_dest = self._dtensor(torch.zeros_like(full), mesh_shape, placements)
        out = await api.get_state_dict(sd_key, {"w": ts_dest, "meta": 0})
        if not torch.equal(out["w"].to_local(), dcp_sd["w"].to_local()):
            raise AssertionError(
                f"rank {self.rank}: torchstore reshard != DCP reshard"
            )
        if out["meta"] != 42:
            raise AssertionError("object entry lost")
        return "ok"

    def teardown_local(self):
        if dist.is_initialized():
            dist.destroy_process_group()
