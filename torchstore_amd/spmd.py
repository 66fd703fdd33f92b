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
    def __init__(self, env: SPMDEnv, store, store_name: str):
        self.env = env
        self.store = store
        self.store_name = store_name

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
            self.store.add(done_key, 1)
            if status != "ok":
                raise RuntimeError(f"rank 0 store teardown failed: {status}")


_spmd_sessions: Dict[str, _SPMDSession] = {}


async def initialize_spmd(
    strategy: Optional[PlacementStrategy] = None,
    env: Optional[SPMDEnv] = None,
    store_name: str = api.DEFAULT_STORE,
    storage_device: str = "auto",
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
    if env.rank == 0:
        num_volumes = strategy.num_volumes_for(env.world_size, env.num_hosts)
        controller = await api.initialize(
            num_storage_volumes=num_volumes,
            strategy=strategy,
            store_name=store_name,
            storage_device=storage_device,
        )
        store.set(key, pickle.dumps(controller))
    else:
        controller = pickle.loads(store.get(key))
        api.attach(controller, strategy, store_name)
    _spmd_sessions[store_name] = _SPMDSession(env, store, store_name)
    logger.info(
        "spmd store %s up: rank %d/%d attached", store_name, env.rank,
        env.world_size,
    )
    return controller


async def shutdown_spmd(store_name: str = api.DEFAULT_STORE) -> None:
    session = _spmd_sessions.pop(store_name, None)
    if session is None:
        await api.shutdown(store_name)
        return
    await session.shutdown()
