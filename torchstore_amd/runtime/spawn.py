"""Actor process spawning.

Replaces Monarch's ``this_host().spawn_procs`` (reference call site
torchstore ``utils.py:128-139``): each actor gets its own OS process
(spawn start-method — no inherited HIP/torch state, which matters because a
forked HIP context is unusable), runs an asyncio loop with an
:class:`~torchstore_amd.runtime.rpc.RpcServer`, and reports its port back
through a pipe.  The spawner gets an :class:`ActorMesh` of handles.

Inside an actor process, :func:`actor_context` exposes the spawn-time rank /
world size / mesh shape (what Monarch's ``current_rank()`` provided — used by
placement strategies to derive volume ids).
"""

from __future__ import annotations

import asyncio
import multiprocessing as mp
import os
from dataclasses import dataclass
from typing import Dict, Optional, Sequence, Tuple, Type

from torchstore_amd.runtime.actor import Actor, ActorHandle, ActorMesh
from torchstore_amd.runtime.rpc import RpcServer
from torchstore_amd.utils.logging import get_logger

logger = get_logger("torchstore_amd.spawn")


@dataclass
class ActorContext:
    rank: int = 0
    world_size: int = 1
    mesh_shape: Tuple[int, ...] = (1,)
    name: str = ""


_context = ActorContext()


def actor_context() -> ActorContext:
    return _context


def _set_context(ctx: ActorContext) -> None:
    global _context
    _context = ctx


async def _actor_serve(
    actor_cls: Type[Actor], args, kwargs, report, ctx: ActorContext
) -> None:
    try:
        actor = actor_cls(*args, **kwargs)
        await actor.setup()
        server = RpcServer(actor)
        host, port = await server.start()
        report.send(("ok", host, port))
    except Exception as exc:  # noqa: BLE001 — reported to spawner
        import traceback

        report.send(("err", f"{exc}\n{traceback.format_exc()}", None))
        return
    finally:
        report.close()
    await server.serve_until_stopped()
    teardown = getattr(actor, "teardown_local", None)
    if teardown is not None:
        res = teardown()
        if asyncio.iscoroutine(res):
            await res


def _actor_main(actor_cls, args, kwargs, report, ctx: ActorContext, env: Dict[str, str]):
    os.environ.update(env)
    _set_context(ctx)
    from torchstore_amd.utils.logging import init_logging

    init_logging()
    asyncio.run(_actor_serve(actor_cls, args, kwargs, report, ctx))


def spawn_actors(
    num: int,
    actor_cls: Type[Actor],
    name: str,
    *args,
    mesh_shape: Optional[Sequence[int]] = None,
    env_per_rank: Optional[Dict[int, Dict[str, str]]] = None,
    # a loaded box paging torch into a fresh spawn-context process can
    # take >60 s (measured mid-suite on the GPU pool) — be generous
    timeout: float = 240.0,
    **kwargs,
) -> ActorMesh:
    """Spawn ``num`` actor processes and return their mesh (blocking)."""
    shape = tuple(mesh_shape) if mesh_shape else (num,)
    total = 1
    for s in shape:
        total *= s
    if total != num:
        raise ValueError(f"mesh_shape {shape} != num {num}")

    ctx_mp = mp.get_context("spawn")
    procs = []
    pipes = []
    for rank in range(num):
        parent, child = ctx_mp.Pipe()
        actx = ActorContext(rank=rank, world_size=num, mesh_shape=shape, name=name)
        env = dict(env_per_rank.get(rank, {})) if env_per_rank else {}
        p = ctx_mp.Process(
            target=_actor_main,
            args=(actor_cls, args, kwargs, child, actx, env),
            name=f"{name}-{rank}",
            daemon=True,
        )
        p.start()
        child.close()
        procs.append(p)
        pipes.append(parent)

    handles = []
    try:
        for rank, pipe in enumerate(pipes):
            if not pipe.poll(timeout):
                raise TimeoutError(f"actor {name}-{rank} did not start in {timeout}s")
            status, a, b = pipe.recv()
            if status != "ok":
                raise RuntimeError(f"actor {name}-{rank} failed to start:\n{a}")
            handles.append(ActorHandle(host=a, port=b, name=f"{name}-{rank}", rank=rank))
    except Exception:
        for p in procs:
            p.terminate()
        raise
    finally:
        for pipe in pipes:
            pipe.close()

    mesh = ActorMesh(handles=handles, mesh_shape=shape)
    mesh._procs = procs
    return mesh


def spawn_actor(actor_cls: Type[Actor], name: str, *args, **kwargs) -> ActorHandle:
    """Spawn one actor process; returns its handle (mesh kept for teardown)."""
    mesh = spawn_actors(1, actor_cls, name, *args, **kwargs)
    handle = mesh.handles[0]
    _singleton_meshes[handle] = mesh
    return handle


# keeps process objects alive for actors spawned via spawn_actor
_singleton_meshes: Dict[ActorHandle, ActorMesh] = {}


async def stop_actor(handle: ActorHandle) -> None:
    mesh = _singleton_meshes.pop(handle, None)
    if mesh is not None:
        await mesh.stop()
    else:
        await handle.stop()
