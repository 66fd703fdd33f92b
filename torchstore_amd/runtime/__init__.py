from torchstore_amd.runtime.actor import (
    Actor,
    ActorHandle,
    ActorMesh,
    close_connections,
    endpoint,
)
from torchstore_amd.runtime.rpc import RemoteError
from torchstore_amd.runtime.spawn import (
    actor_context,
    spawn_actor,
    spawn_actors,
    stop_actor,
)

__all__ = [
    "Actor",
    "ActorHandle",
    "ActorMesh",
    "RemoteError",
    "actor_context",
    "close_connections",
    "endpoint",
    "spawn_actor",
    "spawn_actors",
    "stop_actor",
]
