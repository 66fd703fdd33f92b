"""Control plane: the global key index and DTensor commit gate.

The controller only ever sees metadata (``Request.meta_only()`` copies);
bulk bytes never touch it.  Semantics match the reference Controller
(torchstore ``controller.py``):

* index: prefix-trie of ``key → {volume_id: StorageInfo}``;
* commit gate: a sharded key is invisible to readers until **every** mesh
  coordinate (cartesian product of the mesh shape) has been stored
  (``controller.py:66-104``) — reads of a half-committed DTensor raise
  ``KeyError`` mentioning "partially committed";
* delete protocol: the client notifies the controller *before* deleting on
  the volume so the index never points at freed data.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from enum import Enum
from typing import Dict, List, Optional, Sequence, Set, Tuple

from torchstore_amd.runtime import Actor, ActorHandle, endpoint
from torchstore_amd.types import Request, TensorSlice
from torchstore_amd.utils.logging import get_logger
from torchstore_amd.utils.trie import Trie

logger = get_logger("torchstore_amd.controller")


class ObjectType(Enum):
    OBJECT = "object"
    TENSOR = "tensor"
    TENSOR_SLICE = "tensor_slice"


@dataclass
class StorageInfo:
    object_type: ObjectType
    tensor_slices: Set[TensorSlice] = field(default_factory=set)
    # controller-assigned write sequence: readers of a key stored on
    # several volumes (e.g. a re-put routed to a different volume after a
    # strategy/client change) must serve the NEWEST copy, not whichever
    # volume sorts first — the stale-copy gap the reference leaves open
    seq: int = 0

    def merge(self, other: "StorageInfo") -> None:
        self.seq = max(self.seq, other.seq)
        if other.object_type != self.object_type:
            # last write wins on type change (e.g. object overwritten by tensor)
            self.object_type = other.object_type
            self.tensor_slices = set(other.tensor_slices)
            return
        if other.tensor_slices and self.tensor_slices:
            mine = next(iter(self.tensor_slices))
            new = next(iter(other.tensor_slices))
            if (
                mine.mesh_shape != new.mesh_shape
                or mine.global_shape != new.global_shape
            ):
                # new sharding epoch replaces the old layout
                self.tensor_slices = set(other.tensor_slices)
                return
        self.tensor_slices |= other.tensor_slices


@dataclass
class VolumeInfo:
    volume_id: str
    hostname: str
    device: str
    handle: ActorHandle


def layout_fingerprint(locations: Dict[str, StorageInfo]) -> str:
    """Deterministic digest of a key's placement (volumes + slices).

    Clients cache fetch plans keyed on this: one cheap ``verify_layouts``
    RPC replaces locate + replanning in steady-state sync loops; any
    layout change (re-shard, delete, volume move) mismatches and forces a
    replan.  Built from sorted reprs — never the salted builtin hash.
    """
    import hashlib

    parts = []
    for vid in sorted(locations):
        info = locations[vid]
        slices = ";".join(sorted(repr(s) for s in info.tensor_slices))
        parts.append(f"{vid}|{info.object_type.value}|{slices}")
    return hashlib.md5("\n".join(parts).encode()).hexdigest()


def _info_from_request(request: Request) -> StorageInfo:
    if request.is_object:
        return StorageInfo(ObjectType.OBJECT)
    if request.tensor_slice is not None:
        return StorageInfo(
            ObjectType.TENSOR_SLICE, {request.tensor_slice}
        )
    return StorageInfo(ObjectType.TENSOR)


class Controller(Actor):
    def __init__(self, store_name: str = "default"):
        self.store_name = store_name
        self.index: Trie = Trie()
        self.volumes: Dict[str, VolumeInfo] = {}
        self.strategy_spec: Optional[dict] = None
        self._write_seq = 0

    # -- bring-up ---------------------------------------------------------
    @endpoint
    def register_volumes(
        self, volumes: Sequence[VolumeInfo], strategy_spec: Optional[dict] = None
    ) -> None:
        for v in volumes:
            self.volumes[v.volume_id] = v
        if strategy_spec is not None:
            self.strategy_spec = strategy_spec
        logger.info(
            "controller %s: %d volumes registered", self.store_name, len(self.volumes)
        )

    @endpoint
    def get_volumes(self) -> Tuple[List[VolumeInfo], Optional[dict]]:
        return list(self.volumes.values()), self.strategy_spec

    # -- commit tracking --------------------------------------------------
    @staticmethod
    def _is_fully_committed(locations: Dict[str, StorageInfo]) -> bool:
        # the NEWEST write decides the key's current kind: a stale plain-
        # tensor copy left by an earlier epoch must not make a half-
        # committed shard set readable (and vice versa)
        newest = max(locations.values(), key=lambda i: i.seq)
        if newest.object_type != ObjectType.TENSOR_SLICE:
            return True  # whole tensors/objects commit atomically
        slices: Set[TensorSlice] = set()
        for info in locations.values():
            if info.object_type == ObjectType.TENSOR_SLICE:
                slices |= info.tensor_slices
        if not slices:
            return False
        mesh_shape = next(iter(slices)).mesh_shape
        have = {s.coordinates for s in slices}
        for coord in TensorSlice.expected_coordinates(mesh_shape):
            if tuple(coord) not in have:
                return False
        return True

    @endpoint
    def notify_put_batch(
        self, requests: Sequence[Request], volume_id: str
    ) -> None:
        for r in requests:
            if r.has_payload:
                raise AssertionError(
                    "controller must only receive meta-only requests"
                )
            info = _info_from_request(r)
            self._write_seq += 1
            info.seq = self._write_seq
            locations: Dict[str, StorageInfo] = self.index.get(r.key)
            if locations is None:
                locations = {}
                self.index[r.key] = locations
            if info.object_type == ObjectType.TENSOR_SLICE and locations:
                new_slice = next(iter(info.tensor_slices))
                stale = [
                    vid for vid, old in locations.items()
                    if old.object_type == ObjectType.TENSOR_SLICE
                    and old.tensor_slices
                    and (
                        next(iter(old.tensor_slices)).mesh_shape
                        != new_slice.mesh_shape
                        or next(iter(old.tensor_slices)).global_shape
                        != new_slice.global_shape
                    )
                ]
                # a different mesh/global shape starts a new sharding epoch:
                # drop every volume's old-layout entries for the key
                for vid in stale:
                    del locations[vid]
            if volume_id in locations:
                locations[volume_id].merge(info)
            else:
                locations[volume_id] = info

    @endpoint
    def locate(
        self, keys: Sequence[str], missing_ok: bool = False
    ) -> Dict[str, Dict[str, StorageInfo]]:
        """Per key: which volumes hold it (and which slices each holds)."""
        out: Dict[str, Dict[str, StorageInfo]] = {}
        for key in keys:
            locations = self.index.get(key)
            if locations is None:
                if missing_ok:
                    continue
                raise KeyError(f"key {key!r} does not exist in store")
            if not self._is_fully_committed(locations):
                if missing_ok:
                    continue
                raise KeyError(
                    f"key {key!r} is partially committed: not every mesh "
                    "coordinate has stored its shard yet"
                )
            out[key] = locations
        return out

    @endpoint
    def verify_layouts(self, fingerprints: Dict[str, str]) -> bool:
        """True iff every key still has exactly the fingerprinted layout."""
        for key, fp in fingerprints.items():
            locations = self.index.get(key)
            if locations is None or not self._is_fully_committed(locations):
                return False
            if layout_fingerprint(locations) != fp:
                return False
        return True

    @endpoint
    def notify_delete(self, key: str, missing_ok: bool = False) -> List[str]:
        """Remove from index; returns volume ids that held the key."""
        locations = self.index.pop(key, None)
        if locations is None:
            if missing_ok:
                return []
            raise KeyError(f"key {key!r} does not exist in store")
        return list(locations.keys())

    @endpoint
    def notify_delete_batch(
        self, keys: Sequence[str], missing_ok: bool = True
    ) -> Dict[str, List[str]]:
        out = {}
        for key in keys:
            locations = self.index.pop(key, None)
            if locations is None:
                if not missing_ok:
                    raise KeyError(f"key {key!r} does not exist in store")
                continue
            out[key] = list(locations.keys())
        return out

    @endpoint
    def list_keys(self, prefix: Optional[str] = None) -> List[str]:
        committed = []
        for key in self.index.keys_with_prefix(prefix):
            if self._is_fully_committed(self.index[key]):
                committed.append(key)
        return committed

    @endpoint
    def key_exists(self, key: str) -> bool:
        locations = self.index.get(key)
        return locations is not None and self._is_fully_committed(locations)

    @endpoint
    def stats(self) -> Dict[str, int]:
        """Index-level observability: key/volume/shard counts."""
        n_keys = 0
        n_sharded = 0
        n_shards = 0
        for key in self.index:
            n_keys += 1
            for info in self.index[key].values():
                if info.object_type == ObjectType.TENSOR_SLICE:
                    n_sharded += 1
                    n_shards += len(info.tensor_slices)
        return {
            "keys": n_keys,
            "volumes": len(self.volumes),
            "sharded_entries": n_sharded,
            "total_shards": n_shards,
        }

    @endpoint
    async def teardown(self) -> None:
        """Reset every volume's storage (volume processes stay up)."""
        for v in self.volumes.values():
            try:
                await v.handle.reset.call_one()
            except (ConnectionError, OSError):
                pass
        self.index = Trie()
