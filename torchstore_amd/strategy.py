"""Placement strategies: which volume a client writes to / reads from.

Mirrors the reference strategies (torchstore ``strategy.py``):

* :class:`LocalRankStrategy` — volume per rank; a client writes to the
  volume whose id equals its own rank (``RANK``/``LOCAL_RANK`` env).
* :class:`HostStrategy` — volume per host; clients write to their host's
  volume.
* :class:`SingletonStrategy` — one volume for everything (the reference's
  deprecated ControllerStorageVolumes shape).

A strategy may force a transport (``transport=TransportType.X``), otherwise
the factory auto-selects per (client, volume) pair.
"""

from __future__ import annotations

import os
import socket
from dataclasses import dataclass
from typing import List, Optional

from torchstore_amd.runtime import ActorHandle
from torchstore_amd.transport.base import TransportContext, TransportType


@dataclass
class StorageVolumeRef:
    """A client's view of one volume: handle + locality info + caches."""

    volume: ActorHandle
    volume_id: str
    hostname: str
    device: str
    transport_context: TransportContext
    transport_type: Optional[TransportType] = None

    @property
    def is_local(self) -> bool:
        myhost = os.environ.get("HOSTNAME") or socket.gethostname()
        return myhost == self.hostname


@dataclass
class PlacementStrategy:
    """Base: deterministic client→volume mapping over registered volumes."""

    transport: Optional[TransportType] = None
    # how volumes derive their id at spawn ("rank" | "host")
    volume_id_seed: str = "rank"

    def client_id(self) -> str:
        raise NotImplementedError

    def select_volume_id(self, volume_ids: List[str]) -> str:
        raise NotImplementedError

    def num_volumes_for(self, world_size: int, hosts: int = 1) -> int:
        raise NotImplementedError

    def spec(self) -> dict:
        return {
            "kind": type(self).__name__,
            "transport": self.transport.value if self.transport else None,
        }


@dataclass
class LocalRankStrategy(PlacementStrategy):
    volume_id_seed: str = "rank"

    def client_id(self) -> str:
        return os.environ.get("RANK", os.environ.get("LOCAL_RANK", "0"))

    def select_volume_id(self, volume_ids: List[str]) -> str:
        cid = self.client_id()
        if cid in volume_ids:
            return cid
        # more clients than volumes: deterministic modulo placement
        ordered = sorted(volume_ids, key=lambda v: (len(v), v))
        return ordered[int(cid) % len(ordered)]

    def num_volumes_for(self, world_size: int, hosts: int = 1) -> int:
        return world_size


@dataclass
class HostStrategy(PlacementStrategy):
    volume_id_seed: str = "host"

    def client_id(self) -> str:
        return os.environ.get("HOSTNAME") or socket.gethostname()

    def select_volume_id(self, volume_ids: List[str]) -> str:
        cid = self.client_id()
        if cid in volume_ids:
            return cid
        return sorted(volume_ids)[0]

    def num_volumes_for(self, world_size: int, hosts: int = 1) -> int:
        return hosts


@dataclass
class SingletonStrategy(PlacementStrategy):
    volume_id_seed: str = "0"

    def client_id(self) -> str:
        return "0"

    def select_volume_id(self, volume_ids: List[str]) -> str:
        return sorted(volume_ids)[0]

    def num_volumes_for(self, world_size: int, hosts: int = 1) -> int:
        return 1


def strategy_from_spec(spec: Optional[dict]) -> PlacementStrategy:
    if spec is None:
        return SingletonStrategy()
    kinds = {
        "LocalRankStrategy": LocalRankStrategy,
        "HostStrategy": HostStrategy,
        "SingletonStrategy": SingletonStrategy,
    }
    cls = kinds.get(spec.get("kind"), SingletonStrategy)
    transport = spec.get("transport")
    return cls(transport=TransportType(transport) if transport else None)
