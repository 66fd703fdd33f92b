"""SPMD bootstrap: bring the store up inside a torchrun job.

Reference semantics (torchstore ``spmd.py``): every rank calls
:func:`initialize_spmd`; rank 0 spawns the storage volumes + controller and
publishes the controller handle through a TCPStore rendezvous; the other
ranks attach.  ``shutdown`` tears the store down on rank 0 and broadcasts
the outcome so peers surface a failed teardown (``spmd.py:155-203``).

Env contract (same vars torchrun sets): RANK, LOCAL_RANK, WORLD_SIZE,
LOCAL_WORLD_SIZE, MASTER_ADDR, MASTER_PORT.  The rendezvous TCPStore binds
MASTER_PORT+1 so it never collides with torch.distributed's own store.
"""

from __future__ import annotations

import os
import pickle
from dataclasses import dataclass
from typing import Dict, Optional

from torchstore_amd import api
from torchstore_amd.runtime import ActorHandle
from torchstore_amd.strategy import LocalRankStrategy, PlacementStrategy
from torchstore_amd.utils.logging import get_logger

logger = get_logger("torchstore_amd.spmd")

_RDV_PORT_OFFSET = 1


@dataclass
class SPMDEnv:
    rank: int
    local_rank: int
    world_size: int
    local_world_size: int
    master_addr: str
    master_port: int

    @classmethod
    def from_env(cls, env: Optional[Dict[str, str]] = None) -> "SPMDEnv":
        e = env if env is not None else os.environ
        try:
            return cls(
                rank=int(e["RANK"]),
                local_rank=int(e.get("LOCAL_RANK", e["RANK"])),
                world_size=int(e["WORLD_SIZE"]),
                local_world_size=int(e.get("LOCAL_WORLD_SIZE", e["WORLD_SIZE"])),
                master_addr=e.get("MASTER_ADDR", "127.0.0.1"),
                master_port=int(e.get("MASTER_PORT", "29500")),
            )
        except KeyError as exc:
            raise RuntimeError(
                f"initialize_spmd needs torchrun-style env vars; missing {exc}"
            ) from exc

    @property
    def num_hosts(self) -> int:
        return max(1, self.world_size // max(1, self.local_world_size))


class _SPMDSession:
    def __init__(self, env: SPMDEnv, store, store_name: str, local_mesh=None):
        self.env = env
        self.store = store
        self.store_name = store_name
        # non-rank-0 hosts: the volume mesh this host's local-rank-0 spawned
        self.local_mesh = local_mesh

    async def shutdown(self) -> None:
        import asyncio

        env = self.env
        # arrival barrier: nobody tears down until every rank is done with
        # its in-flight store operations
        arrived_key = f"torchstore_amd/{self.store_name}/shutdown_arrived"
        self.store.add(arrived_key, 1)
        while int(self.store.add(arrived_key, 0)) < env.world_size:
            await asyncio.sleep(0.02)
        done_key = f"torchstore_amd/{self.store_name}/shutdown_done"
        if env.rank == 0:
            status = "ok"
            try:
                await api.shutdown(self.store_name)
            except Exception as exc:  # noqa: BLE001 — broadcast to peers
                status = f"error: {exc!r}"
            self.store.set(f"torchstore_amd/{self.store_name}/shutdown", status)
            # the TCPStore master lives in this process: stay alive until
            # every peer has read the status
            self.store.add(done_key, 1)
            while int(self.store.add(done_key, 0)) < env.world_size:
                await asyncio.sleep(0.02)
            if status != "ok":
                raise RuntimeError(f"store teardown failed: {status}")
        else:
            api.reset_client(self.store_name)
            api._sessions.pop(self.store_name, None)
            status = self.store.get(
                f"torchstore_amd/{self.store_name}/shutdown"
            ).decode()
            if self.local_mesh is not None:
                # this host's volumes, spawned by its local-rank-0 — stopped
                # after rank 0's teardown (which resets them over RPC) so
                # the reset never races the process exit
                await self.local_mesh.stop()
            self.store.add(done_key, 1)
            if status != "ok":
                raise RuntimeError(f"rank 0 store teardown failed: {status}")


_spmd_sessions: Dict[str, _SPMDSession] = {}


async def initialize_spmd(
    strategy: Optional[PlacementStrategy] = None,
    env: Optional[SPMDEnv] = None,
    store_name: str = api.DEFAULT_STORE,
    storage_device: str = "auto",
    storage_capacity_gb: float = None,
    timeout_s: float = 300.0,
) -> ActorHandle:
    """Collective store bring-up across a torchrun world.

    Returns the controller handle on every rank.
    """
    from datetime import timedelta

    from torch.distributed import TCPStore

    if env is None:
        env = SPMDEnv.from_env()
    if strategy is None:
        strategy = LocalRankStrategy()

    store = TCPStore(
        env.master_addr,
        env.master_port + _RDV_PORT_OFFSET,
        env.world_size,
        is_master=env.rank == 0,
        timeout=timedelta(seconds=timeout_s),
    )
    key = f"torchstore_amd/{store_name}/controller"
    total_volumes = strategy.num_volumes_for(env.world_size, env.num_hosts)
    per_host = max(1, total_volumes // env.num_hosts)
    host_index = env.rank // max(1, env.local_world_size)
    count_key = f"torchstore_amd/{store_name}/volumes_registered"
    local_mesh = None
    if env.rank == 0:
        # rank 0 spawns the controller + ITS OWN host's volumes only; the
        # other hosts' local-rank-0 processes spawn theirs below (the
        # reference places volumes on every host via Monarch host meshes,
        # torchstore spmd.py:317-326 — here each host self-spawns and
        # registers with the controller)
        controller = await api.initialize(
            num_storage_volumes=per_host,
            strategy=strategy,
            store_name=store_name,
            storage_device=storage_device,
            storage_capacity_gb=storage_capacity_gb,
        )
        store.set(key, pickle.dumps(controller))
        store.add(count_key, per_host)
    else:
        controller = pickle.loads(store.get(key))
        api.attach(controller, strategy, store_name)
        if env.local_rank == 0 and host_index > 0:
            local_mesh, infos = await _spawn_host_volumes(
                strategy, store_name, storage_device, env, host_index,
                per_host, storage_capacity_gb,
            )
            await controller.register_volumes.call_one(infos)
            store.add(count_key, len(infos))
    # everyone waits until every host's volumes are registered — a client
    # resolving volumes before that would cache an incomplete list
    while int(store.add(count_key, 0)) < total_volumes:
        import asyncio

        await asyncio.sleep(0.02)
    _spmd_sessions[store_name] = _SPMDSession(env, store, store_name, local_mesh)
    logger.info(
        "spmd store %s up: rank %d/%d attached (%d volumes, %d hosts)",
        store_name, env.rank, env.world_size, total_volumes, env.num_hosts,
    )
    return controller


async def _spawn_host_volumes(
    strategy: PlacementStrategy,
    store_name: str,
    storage_device: str,
    env: SPMDEnv,
    host_index: int,
    per_host: int,
    storage_capacity_gb: float = None,
):
    """Spawn this host's volume processes (called on local-rank-0 of every
    non-zero host) and build their registration infos."""
    import asyncio

    from torchstore_amd.controller import VolumeInfo
    from torchstore_amd.runtime import spawn_actors

    seed = strategy.volume_id_seed
    if seed == "rank":
        seed = f"rank_offset:{host_index * env.local_world_size}"
    mesh = await asyncio.to_thread(
        spawn_actors,
        per_host,
        _volume_cls(),
        f"{store_name}-volume-h{host_index}",
        volume_id_seed=seed,
        device=storage_device,
        capacity_gb=storage_capacity_gb,
        timeout=240.0,
    )
    ids = await mesh.get_id.call()
    infos = [
        VolumeInfo(volume_id=vid, hostname=host, device=dev, handle=h)
        for (vid, host, dev), h in zip(ids, mesh.handles)
    ]
    return mesh, infos


def _volume_cls():
    from torchstore_amd.storage import StorageVolume

    return StorageVolume


async def shutdown_spmd(store_name: str = api.DEFAULT_STORE) -> None:
    session = _spmd_sessions.pop(store_name, None)
    if session is None:
        await api.shutdown(store_name)
        return
    await session.shutdown()
