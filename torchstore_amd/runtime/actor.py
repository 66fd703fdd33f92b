"""Actors, endpoints, handles and meshes — the Monarch-actor replacement.

Semantics kept from the reference runtime:

* subclass :class:`Actor`, mark remote-callable methods with ``@endpoint``;
* an :class:`ActorHandle` is a picklable address — any process holding one
  can ``await handle.method.call_one(...)``;
* an :class:`ActorMesh` is an N-d arrangement of handles —
  ``await mesh.method.call(...)`` fans out to every actor and gathers results
  in mesh order, ``mesh.slice(coord)`` selects one member;
* serialization of arguments honours ``__getstate__`` (transport buffers
  strip their local tensors exactly like the reference's do).

Connections are cached per event loop so repeated calls reuse one TCP
stream; concurrent calls multiplex on it.
"""

from __future__ import annotations

import asyncio
import weakref
from dataclasses import dataclass, field
from typing import Any, List, Sequence, Tuple

from torchstore_amd.runtime.rpc import (
    SHUTDOWN_METHOD,
    HEALTH_METHOD,
    RpcConnection,
)


def endpoint(fn):
    """Mark a method remotely callable."""
    fn._is_endpoint = True
    return fn


class Actor:
    """Base class for remote actors. Construction happens in the actor process."""

    async def setup(self) -> None:  # optional async init after ctor
        return None


# ---------------------------------------------------------------------------
# connection pool (per event loop)
# ---------------------------------------------------------------------------

_pools: "weakref.WeakKeyDictionary[asyncio.AbstractEventLoop, Dict[Tuple[str, int], RpcConnection]]" = (
    weakref.WeakKeyDictionary()
)
_pool_locks: "weakref.WeakKeyDictionary[asyncio.AbstractEventLoop, asyncio.Lock]" = (
    weakref.WeakKeyDictionary()
)


async def get_connection(host: str, port: int) -> RpcConnection:
    loop = asyncio.get_running_loop()
    pool = _pools.setdefault(loop, {})
    lock = _pool_locks.setdefault(loop, asyncio.Lock())
    async with lock:
        conn = pool.get((host, port))
        if conn is not None and conn.is_open:
            return conn
        conn = RpcConnection(host, port)
        await conn.connect()
        pool[(host, port)] = conn
        return conn


async def close_connections() -> None:
    loop = asyncio.get_running_loop()
    pool = _pools.pop(loop, {})
    for conn in pool.values():
        await conn.close()


# ---------------------------------------------------------------------------
# handles
# ---------------------------------------------------------------------------


def _rpc_timeout() -> float:
    """Opt-in per-call RPC timeout (seconds; 0 = wait forever).

    Off by default: windowed multi-GB transfers legitimately hold RPCs
    open for seconds.  Production deployments that prefer failing fast on
    a hung volume set TORCHSTORE_AMD_RPC_TIMEOUT comfortably above their
    largest transfer (e.g. 120)."""
    import os

    try:
        return float(os.environ.get("TORCHSTORE_AMD_RPC_TIMEOUT", "0") or 0)
    except ValueError:
        return 0.0


class _Endpoint:
    __slots__ = ("_handle", "_name")

    def __init__(self, handle: "ActorHandle", name: str):
        self._handle = handle
        self._name = name

    async def call_one(self, *args, **kwargs) -> Any:
        conn = await get_connection(self._handle.host, self._handle.port)
        t = _rpc_timeout()
        if t > 0:
            return await asyncio.wait_for(
                conn.call(self._name, *args, **kwargs), timeout=t
            )
        return await conn.call(self._name, *args, **kwargs)

    # alias so a single handle can stand in where mesh semantics are expected
    async def call(self, *args, **kwargs) -> List[Any]:
        return [await self.call_one(*args, **kwargs)]


@dataclass(frozen=True)
class ActorHandle:
    """Picklable address of a remote actor."""

    host: str
    port: int
    name: str = ""
    rank: int = 0

    def __getattr__(self, item: str) -> _Endpoint:
        if item.startswith("_"):
            raise AttributeError(item)
        return _Endpoint(self, item)

    async def stop(self) -> None:
        try:
            conn = await get_connection(self.host, self.port)
            await asyncio.wait_for(conn.call(SHUTDOWN_METHOD), timeout=10)
        except (ConnectionError, OSError, asyncio.TimeoutError):
            pass

    async def health(self) -> bool:
        try:
            conn = await get_connection(self.host, self.port)
            return await conn.call(HEALTH_METHOD) == "ok"
        except (ConnectionError, OSError):
            return False


class _MeshEndpoint:
    __slots__ = ("_mesh", "_name")

    def __init__(self, mesh: "ActorMesh", name: str):
        self._mesh = mesh
        self._name = name

    async def call(self, *args, **kwargs) -> List[Any]:
        """Fan out to every actor; results in mesh order."""
        return list(
            await asyncio.gather(
                *(
                    _Endpoint(h, self._name).call_one(*args, **kwargs)
                    for h in self._mesh.handles
                )
            )
        )

    async def call_one(self, *args, **kwargs) -> Any:
        if len(self._mesh.handles) != 1:
            raise RuntimeError(
                f"call_one on mesh of {len(self._mesh.handles)} actors"
            )
        return await _Endpoint(self._mesh.handles[0], self._name).call_one(
            *args, **kwargs
        )


@dataclass
class ActorMesh:
    """N-d arrangement of actor handles (row-major over mesh_shape)."""

    handles: List[ActorHandle]
    mesh_shape: Tuple[int, ...] = ()
    _procs: list = field(default_factory=list, repr=False, compare=False)

    def __post_init__(self):
        if not self.mesh_shape:
            self.mesh_shape = (len(self.handles),)

    def __len__(self) -> int:
        return len(self.handles)

    def __getattr__(self, item: str) -> _MeshEndpoint:
        if item.startswith("_"):
            raise AttributeError(item)
        return _MeshEndpoint(self, item)

    def __getstate__(self):
        # child processes receiving a mesh must not inherit process objects
        return {
            "handles": self.handles,
            "mesh_shape": self.mesh_shape,
            "_procs": [],
        }

    def __setstate__(self, state):
        self.__dict__.update(state)

    def flat_index(self, coord: Sequence[int]) -> int:
        idx = 0
        for c, s in zip(coord, self.mesh_shape):
            if not (0 <= c < s):
                raise IndexError(f"coordinate {tuple(coord)} outside mesh {self.mesh_shape}")
            idx = idx * s + c
        return idx

    def slice(self, coord: Sequence[int]) -> ActorHandle:
        return self.handles[self.flat_index(coord)]

    def single(self, index: int) -> "ActorMesh":
        return ActorMesh(handles=[self.handles[index]], mesh_shape=(1,))

    async def stop(self, join_timeout: float = 10.0) -> None:
        await asyncio.gather(
            *(h.stop() for h in self.handles), return_exceptions=True
        )
        for p in self._procs:
            p.join(timeout=join_timeout)
        for p in self._procs:
            if p.is_alive():
                p.terminate()
                p.join(timeout=2)
        self._procs = []
