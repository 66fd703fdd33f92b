"""Module-level async API — the user-facing façade.

Same surface as the reference (torchstore ``api.py``): ``initialize``,
``put``/``get``/``put_batch``/``get_batch``, ``delete``/``delete_batch``,
``keys``/``exists``, ``put_state_dict``/``get_state_dict``, ``shutdown``.
State is kept per store name: one spawned controller + volume mesh + a
cached LocalClient per process.
"""

from __future__ import annotations

import asyncio
from dataclasses import dataclass
from typing import Any, Dict, List, Optional, Sequence

from torchstore_amd.client import LocalClient
from torchstore_amd.controller import Controller, VolumeInfo
from torchstore_amd.runtime import (
    ActorHandle,
    ActorMesh,
    close_connections,
    spawn_actor,
    spawn_actors,
    stop_actor,
)
from torchstore_amd.storage import StorageVolume
from torchstore_amd.strategy import PlacementStrategy, SingletonStrategy
from torchstore_amd.utils.logging import get_logger

logger = get_logger("torchstore_amd.api")

DEFAULT_STORE = "default"


@dataclass
class StoreSession:
    name: str
    controller: ActorHandle
    strategy: PlacementStrategy
    volume_mesh: Optional[ActorMesh] = None  # only on the spawning process
    owns_processes: bool = True


_sessions: Dict[str, StoreSession] = {}
_clients: Dict[str, LocalClient] = {}


async def initialize(
    num_storage_volumes: Optional[int] = None,
    strategy: Optional[PlacementStrategy] = None,
    store_name: str = DEFAULT_STORE,
    storage_device: str = "auto",
    storage_capacity_gb: Optional[float] = None,
) -> ActorHandle:
    """Spawn volumes + controller; returns the controller handle.

    The handle is picklable — other processes can join the same store with
    :func:`attach`.
    """
    if store_name in _sessions:
        raise RuntimeError(f"store {store_name!r} already initialized")
    if strategy is None:
        strategy = SingletonStrategy()
    if num_storage_volumes is None:
        num_storage_volumes = 1

    mesh, controller = await asyncio.gather(
        asyncio.to_thread(
            spawn_actors,
            num_storage_volumes,
            StorageVolume,
            f"{store_name}-volume",
            volume_id_seed=strategy.volume_id_seed,
            device=storage_device,
            capacity_gb=storage_capacity_gb,
            # a node bringing up many ranks + volumes at once can take a
            # while per torch import; don't flake on a loaded box
            timeout=240.0,
        ),
        asyncio.to_thread(
            spawn_actor, Controller, f"{store_name}-controller", store_name,
            # same generous timeout as the volumes: a loaded box paging
            # torch into a fresh process can take >60 s
            timeout=240.0,
        ),
    )
    ids = await mesh.get_id.call()
    infos = [
        VolumeInfo(volume_id=vid, hostname=host, device=dev, handle=h)
        for (vid, host, dev), h in zip(ids, mesh.handles)
    ]
    await controller.register_volumes.call_one(infos, strategy.spec())
    _sessions[store_name] = StoreSession(
        name=store_name,
        controller=controller,
        strategy=strategy,
        volume_mesh=mesh,
    )
    logger.info("store %s initialized: %d volumes", store_name, len(infos))
    return controller


def attach(
    controller: ActorHandle,
    strategy: Optional[PlacementStrategy] = None,
    store_name: str = DEFAULT_STORE,
) -> None:
    """Join an existing store from another process (SPMD ranks != 0)."""
    _sessions[store_name] = StoreSession(
        name=store_name,
        controller=controller,
        strategy=strategy,
        volume_mesh=None,
        owns_processes=False,
    )


def client(store_name: str = DEFAULT_STORE) -> LocalClient:
    c = _clients.get(store_name)
    if c is None:
        session = _sessions.get(store_name)
        if session is None:
            raise RuntimeError(
                f"store {store_name!r} is not initialized in this process; "
                "call initialize() or attach() first"
            )
        c = LocalClient(session.controller, session.strategy)
        _clients[store_name] = c
    return c


async def shutdown(store_name: str = DEFAULT_STORE) -> None:
    # SPMD jobs route through the collective session (rank-0 teardown +
    # status broadcast) when one exists
    from torchstore_amd import spmd as _spmd

    spmd_session = _spmd._spmd_sessions.pop(store_name, None)
    if spmd_session is not None:
        await spmd_session.shutdown()
        return
    session = _sessions.pop(store_name, None)
    c = _clients.pop(store_name, None)
    if c is not None:
        c.close()
    if session is None:
        return
    if session.owns_processes:
        try:
            await session.controller.teardown.call_one()
        except (ConnectionError, OSError):
            pass
        if session.volume_mesh is not None:
            await session.volume_mesh.stop()
        await stop_actor(session.controller)
    await close_connections()


def reset_client(store_name: str = DEFAULT_STORE) -> None:
    c = _clients.pop(store_name, None)
    if c is not None:
        c.close()


# -- data ops -----------------------------------------------------------


async def put(key: str, value: Any, store_name: str = DEFAULT_STORE) -> None:
    await client(store_name).put(key, value)


async def get(key: str, like: Any = None, store_name: str = DEFAULT_STORE) -> Any:
    return await client(store_name).get(key, like)


async def put_batch(items: Dict[str, Any], store_name: str = DEFAULT_STORE) -> None:
    await client(store_name).put_batch(items)


async def get_batch(
    fetches: Dict[str, Any], store_name: str = DEFAULT_STORE
) -> Dict[str, Any]:
    return await client(store_name).get_batch(fetches)


async def delete(
    key: str, missing_ok: bool = False, store_name: str = DEFAULT_STORE
) -> None:
    await client(store_name).delete(key, missing_ok)


async def delete_batch(
    keys_: Sequence[str], missing_ok: bool = True, store_name: str = DEFAULT_STORE
) -> None:
    await client(store_name).delete_batch(keys_, missing_ok)


async def keys(
    prefix: Optional[str] = None, store_name: str = DEFAULT_STORE
) -> List[str]:
    return await client(store_name).keys(prefix)


async def exists(key: str, store_name: str = DEFAULT_STORE) -> bool:
    return await client(store_name).exists(key)


async def stats(store_name: str = DEFAULT_STORE) -> Dict[str, Any]:
    """Whole-store observability: the controller's index stats plus every
    volume's entry/byte/tier counters (beyond reference parity)."""
    c = client(store_name)
    volumes = await c._ensure_volumes()
    ctrl, *vols = await asyncio.gather(
        c._controller.stats.call_one(),
        *(v.handle.stats.call_one() for v in volumes.values()),
    )
    return {"controller": ctrl, "volumes": list(vols)}


async def put_state_dict(
    state_dict: Dict[str, Any], key: str, store_name: str = DEFAULT_STORE, **kw
) -> None:
    from torchstore_amd.state_dict import put_state_dict as _impl

    await _impl(client(store_name), state_dict, key, **kw)


async def get_state_dict(
    key: str,
    user_state_dict: Optional[Dict[str, Any]] = None,
    store_name: str = DEFAULT_STORE,
    **kw,
) -> Dict[str, Any]:
    from torchstore_amd.state_dict import get_state_dict as _impl

    return await _impl(client(store_name), key, user_state_dict, **kw)
