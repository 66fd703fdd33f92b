"""Storage plane: volume actors and the pluggable in-memory store.

Semantics follow the reference's StorageVolume/StorageImpl/InMemoryStore
(torchstore ``storage_volume.py``) with one MI355X-first change: a volume can
keep its tensors **GPU-resident** (``device="auto"`` picks the HBM of the GPU
assigned to the volume process — 288 GB per MI355X means an 8-volume node
holds >2 TB of hot state without touching host DRAM).  Slice extraction on a
GPU-resident store runs the CDNA4 gather kernel (K1) instead of torch
narrow+clone.

Storage shapes per key (same three as the reference):
  * object        — any pickled python object
  * full tensor   — a plain tensor
  * shard dict    — ``{mesh_coords: (TensorSlice, tensor)}`` for DTensor keys
"""

from __future__ import annotations

import os
import socket
import time
from dataclasses import dataclass
from typing import Any, Dict, List, Optional, Sequence, Tuple, Union

import torch

from torchstore_amd.ops.slicing import extract_region
from torchstore_amd.runtime import Actor, endpoint, actor_context
from torchstore_amd.transport.base import TransportBuffer, TransportContext
from torchstore_amd.types import Request, TensorSlice
from torchstore_amd.utils.logging import get_logger

logger = get_logger("torchstore_amd.storage")

OBJ_SENTINEL = "<obj>"


@dataclass
class TensorMeta:
    shape: Tuple[int, ...]
    dtype: torch.dtype
    device: str  # "cpu" | "cuda"


class StorageImpl:
    """Backend interface; a tiered/persistent store plugs in here."""

    def put(self, request: Request, value: Any) -> None:
        raise NotImplementedError

    def find_existing(self, request: Request) -> Optional[torch.Tensor]:
        raise NotImplementedError

    def fetch(self, request: Request) -> Any:
        raise NotImplementedError

    def meta(self, request: Request) -> Union[TensorMeta, str]:
        raise NotImplementedError

    def delete(self, key: str, missing_ok: bool = False) -> None:
        raise NotImplementedError

    def keys(self) -> List[str]:
        raise NotImplementedError

    def reset(self) -> None:
        raise NotImplementedError


class _ObjEntry:
    __slots__ = ("obj",)

    def __init__(self, obj):
        self.obj = obj


class _TensorEntry:
    __slots__ = ("tensor",)

    def __init__(self, tensor):
        self.tensor = tensor


class _ShardEntry:
    __slots__ = ("shards",)

    def __init__(self):
        # mesh coords -> (TensorSlice, tensor)
        self.shards: Dict[Tuple[int, ...], Tuple[TensorSlice, torch.Tensor]] = {}


class InMemoryStore(StorageImpl):
    """Dict-backed store; ``device`` decides where tensor bytes live."""

    def __init__(self, device: Union[str, torch.device] = "cpu"):
        self.device = torch.device(device)
        self.kv: Dict[str, Union[_ObjEntry, _TensorEntry, _ShardEntry]] = {}

    # -- placement hooks (overridden by TieredStore) ----------------------
    def _place(self, tensor: torch.Tensor) -> torch.Tensor:
        """Decide where a stored tensor's bytes live."""
        if tensor.device != self.device:
            tensor = tensor.to(self.device)
        return tensor

    def _release(self, tensor: torch.Tensor) -> None:
        """Accounting hook: ``tensor`` is leaving the store."""
        return None

    def _release_entry(self, entry) -> None:
        if isinstance(entry, _TensorEntry):
            self._release(entry.tensor)
        elif isinstance(entry, _ShardEntry):
            for _s, t in entry.shards.values():
                self._release(t)

    # -- write -----------------------------------------------------------
    def put(self, request: Request, value: Any) -> None:
        key = request.key
        old = self.kv.get(key)
        if request.is_object:
            if old is not None:
                self._release_entry(old)
            self.kv[key] = _ObjEntry(value)
            return
        tensor = value
        if not isinstance(tensor, torch.Tensor):
            raise TypeError(f"put of non-tensor {type(tensor)} without is_object")
        if request.tensor_slice is None:
            if old is not None:
                self._release_entry(old)
            self.kv[key] = _TensorEntry(self._place(tensor))
            return
        entry = old if isinstance(old, _ShardEntry) else None
        if entry is None:
            if old is not None:
                self._release_entry(old)
            entry = _ShardEntry()
            self.kv[key] = entry
        ts = request.tensor_slice
        if entry.shards:
            any_slice, _t = next(iter(entry.shards.values()))
            if (
                any_slice.mesh_shape != ts.mesh_shape
                or any_slice.global_shape != ts.global_shape
            ):
                # a re-push under a different sharding: stale shards of the
                # old layout could otherwise serve wrong regions
                for _s, t in entry.shards.values():
                    self._release(t)
                entry.shards.clear()
        prev = entry.shards.get(ts.coordinates)
        if prev is not None:
            self._release(prev[1])
        entry.shards[ts.coordinates] = (ts, self._place(tensor))

    def find_existing(self, request: Request) -> Optional[torch.Tensor]:
        entry = self.kv.get(request.key)
        if entry is None or request.is_object:
            return None
        if isinstance(entry, _TensorEntry) and request.tensor_slice is None:
            return entry.tensor
        if isinstance(entry, _ShardEntry) and request.tensor_slice is not None:
            hit = entry.shards.get(request.tensor_slice.coordinates)
            if hit is not None and hit[0] == request.tensor_slice:
                return hit[1]
        return None

    # -- read ------------------------------------------------------------
    def _entry(self, key: str):
        entry = self.kv.get(key)
        if entry is None:
            raise KeyError(f"key {key!r} not found in storage volume")
        return entry

    def fetch(self, request: Request) -> Any:
        entry = self._entry(request.key)
        if isinstance(entry, _ObjEntry):
            return entry.obj
        ts = request.tensor_slice
        if isinstance(entry, _TensorEntry):
            if ts is None:
                return entry.tensor
            full_off = (0,) * entry.tensor.dim()
            return extract_region(entry.tensor, full_off, ts.offsets, ts.local_shape)
        # shard entry: find a stored shard fully containing the region
        assert isinstance(entry, _ShardEntry)
        if ts is None:
            raise KeyError(
                f"key {request.key!r} is sharded; a slice request is required"
            )
        for stored_slice, tensor in entry.shards.values():
            inter = stored_slice.intersect(ts)
            if inter is not None and inter.local_shape == ts.local_shape:
                return extract_region(
                    tensor, stored_slice.offsets, ts.offsets, ts.local_shape
                )
        raise KeyError(
            f"no stored shard of {request.key!r} contains region "
            f"{ts.offsets}+{ts.local_shape}"
        )

    def meta(self, request: Request) -> Union[TensorMeta, str]:
        entry = self._entry(request.key)
        if isinstance(entry, _ObjEntry):
            return OBJ_SENTINEL
        ts = request.tensor_slice
        if isinstance(entry, _TensorEntry):
            t = entry.tensor
            shape = ts.local_shape if ts is not None else tuple(t.shape)
            return TensorMeta(shape=shape, dtype=t.dtype, device=t.device.type)
        assert isinstance(entry, _ShardEntry)
        any_slice, any_tensor = next(iter(entry.shards.values()))
        shape = ts.local_shape if ts is not None else any_slice.global_shape
        return TensorMeta(shape=shape, dtype=any_tensor.dtype, device=any_tensor.device.type)

    # -- admin -----------------------------------------------------------
    def delete(self, key: str, missing_ok: bool = False) -> None:
        entry = self.kv.pop(key, None)
        if entry is not None:
            self._release_entry(entry)
        elif not missing_ok:
            raise KeyError(key)

    def keys(self) -> List[str]:
        return list(self.kv.keys())

    def reset(self) -> None:
        for entry in self.kv.values():
            self._release_entry(entry)
        self.kv.clear()


class TieredStore(InMemoryStore):
    """HBM-primary store with a host-memory overflow tier.

    Fills the ``StorageImpl`` seam the reference leaves open
    (torchstore ``storage_volume.py:102-143``: "a second backend, e.g.
    GPU-resident or tiered, is an explicit extension point"): tensors
    live in the volume's HBM until ``capacity_bytes`` of primary
    residency, then overflow to ``spill_device`` (pageable host memory).
    Fetches/slices serve from either tier transparently — transports
    route CPU-resident results over the inline/SHM paths.  Placement is
    decided at write time; freeing primary bytes (delete/overwrite/
    reset) makes room for later puts.  No eviction of already-resident
    entries — the store is not a cache, readers hold zero-copy views.
    """

    def __init__(
        self,
        device: Union[str, torch.device],
        capacity_bytes: int,
        spill_device: Union[str, torch.device] = "cpu",
    ):
        super().__init__(device)
        self.capacity_bytes = int(capacity_bytes)
        self.spill_device = torch.device(spill_device)
        self.primary_used = 0
        self._resident: set = set()  # id() of primary-resident tensors

    def _place(self, tensor: torch.Tensor) -> torch.Tensor:
        nb = tensor.numel() * tensor.element_size()
        if self.primary_used + nb <= self.capacity_bytes:
            if tensor.device != self.device:
                tensor = tensor.to(self.device)
            self.primary_used += nb
            self._resident.add(id(tensor))
            return tensor
        logger.info(
            "tiered store: spilling %d bytes of %r to %s "
            "(primary %d/%d used)",
            nb, tensor.shape, self.spill_device, self.primary_used,
            self.capacity_bytes,
        )
        if tensor.device != self.spill_device:
            if self.spill_device.type == "cpu" and torch.cuda.is_available():
                try:
                    # PINNED spill: D2H lands at DMA rate and later reads
                    # stage back through HBM at ~50 GB/s instead of the
                    # ~13 GB/s pageable path
                    pinned = torch.empty(
                        tensor.shape, dtype=tensor.dtype, pin_memory=True
                    )
                    pinned.copy_(tensor)
                    return pinned
                except RuntimeError:  # pinned allocation failed: fall back
                    pass
            tensor = tensor.to(self.spill_device)
        return tensor

    def _release(self, tensor: torch.Tensor) -> None:
        if id(tensor) in self._resident:
            self._resident.discard(id(tensor))
            self.primary_used -= tensor.numel() * tensor.element_size()

    def reset(self) -> None:
        super().reset()
        self.primary_used = 0
        self._resident.clear()


def _resolve_device(device: str) -> torch.device:
    """``auto`` → this volume's GPU when one is visible, else CPU."""
    if device != "auto":
        return torch.device(device)
    if torch.cuda.is_available():
        n = torch.cuda.device_count()
        idx = actor_context().rank % max(n, 1)
        return torch.device("cuda", idx)
    return torch.device("cpu")


class StorageVolume(Actor):
    """A storage actor process; holds one StorageImpl + a transport context."""

    def __init__(
        self,
        volume_id_seed: str = "rank",
        device: str = "auto",
        capacity_gb: Optional[float] = None,
        spill_device: str = "cpu",
    ):
        self.ctx = TransportContext()
        self.device = _resolve_device(device)
        if self.device.type == "cuda":
            torch.cuda.set_device(self.device)
        if capacity_gb is not None:
            self.store: StorageImpl = TieredStore(
                self.device, int(capacity_gb * 1e9), spill_device
            )
        else:
            self.store = InMemoryStore(self.device)
        self.volume_id = self._make_volume_id(volume_id_seed)
        self.hostname = os.environ.get("HOSTNAME") or socket.gethostname()
        logger.info(
            "storage volume %s up on %s device=%s",
            self.volume_id, self.hostname, self.device,
        )

    @staticmethod
    def _make_volume_id(seed: str) -> str:
        if seed == "rank":
            return str(actor_context().rank)
        if seed.startswith("rank_offset:"):
            # multi-host spawning: host h's local spawn mesh restarts ranks
            # at 0, the offset makes volume ids global (h * local_world)
            return str(actor_context().rank + int(seed.split(":", 1)[1]))
        if seed == "host":
            return os.environ.get("HOSTNAME") or socket.gethostname()
        return seed

    # -- identity ---------------------------------------------------------
    @endpoint
    def get_id(self) -> Tuple[str, str, str]:
        return self.volume_id, self.hostname, str(self.device)

    # -- data plane -------------------------------------------------------
    @endpoint
    async def handshake(
        self, buffer: TransportBuffer, requests: Sequence[Request], phase: str
    ):
        buffer.attach_volume(self.ctx)
        return buffer.recv_handshake(requests, phase, self)

    @endpoint
    async def put(self, buffer: TransportBuffer, requests: Sequence[Request]):
        t0 = time.perf_counter()
        buffer.attach_volume(self.ctx)
        existing = [self.store.find_existing(r) for r in requests]
        t1 = time.perf_counter()
        values = await buffer.volume_receive(requests, existing, self.device)
        t2 = time.perf_counter()
        for r, v in zip(requests, values):
            self.store.put(r, v)
        logger.info(
            "volume.put n=%d find=%.1fms recv=%.1fms store=%.1fms",
            len(requests), (t1 - t0) * 1e3, (t2 - t1) * 1e3,
            (time.perf_counter() - t2) * 1e3,
        )

    @endpoint
    async def get(self, buffer: TransportBuffer, requests: Sequence[Request]):
        t0 = time.perf_counter()
        buffer.attach_volume(self.ctx)
        values = [self.store.fetch(r) for r in requests]
        t1 = time.perf_counter()
        out = await buffer.volume_send(requests, values)
        logger.info(
            "volume.get n=%d fetch=%.1fms send=%.1fms",
            len(requests), (t1 - t0) * 1e3, (time.perf_counter() - t1) * 1e3,
        )
        return out

    @endpoint
    def get_meta(self, requests: Sequence[Request]):
        return [self.store.meta(r) for r in requests]

    # -- admin ------------------------------------------------------------
    @endpoint
    def delete(self, key: str, missing_ok: bool = False) -> None:
        self.store.delete(key, missing_ok=missing_ok)
        self.ctx.drop_key(key)

    @endpoint
    def delete_batch(self, keys: Sequence[str], missing_ok: bool = True) -> None:
        for k in keys:
            self.store.delete(k, missing_ok=missing_ok)
            self.ctx.drop_key(k)

    @endpoint
    def reset(self) -> None:
        self.store.reset()
        self.ctx.close()
        self.ctx = TransportContext()

    @endpoint
    def stored_keys(self) -> List[str]:
        return self.store.keys()

    @endpoint
    def stats(self) -> Dict[str, Any]:
        """Per-volume storage observability: entry counts and resident
        bytes by kind (plus tier occupancy for TieredStore volumes)."""
        out: Dict[str, Any] = {
            "volume_id": self.volume_id,
            "device": str(self.device),
            "entries": 0,
            "tensor_entries": 0,
            "shard_entries": 0,
            "object_entries": 0,
            "tensor_bytes": 0,
        }
        kv = getattr(self.store, "kv", {})
        for entry in kv.values():
            out["entries"] += 1
            if isinstance(entry, _TensorEntry):
                out["tensor_entries"] += 1
                t = entry.tensor
                out["tensor_bytes"] += t.numel() * t.element_size()
            elif isinstance(entry, _ShardEntry):
                out["shard_entries"] += 1
                for _s, t in entry.shards.values():
                    out["tensor_bytes"] += t.numel() * t.element_size()
            else:
                out["object_entries"] += 1
        if isinstance(self.store, TieredStore):
            out["tier_primary_used"] = self.store.primary_used
            out["tier_capacity"] = self.store.capacity_bytes
        return out

    def teardown_local(self):
        self.ctx.close()
