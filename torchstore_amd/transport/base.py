"""Transport abstraction: how tensor bytes move client↔volume.

The control plane (actor RPC) never carries bulk data except on the fallback
path; each :class:`TransportBuffer` implements the two-sided lifecycle the
reference establishes (torchstore ``transport/buffers.py:79-155``):

client side                                 volume side (same object,
                                            reconstructed from the RPC frame)
-----------                                 ------------------------------
put:  [handshake]                           recv_handshake
      client_stage_put   (stage/export)
      volume.put  ───────────────────────►  volume_receive  (materialise)
      on_put_success / drop

get:  fetch metas, allocate dests
      [handshake]
      client_stage_get   (register dests)
      volume.get  ───────────────────────►  volume_send    (emit/transfer)
      client_complete_get ◄────────────────   reply
      drop

Transports on MI355X (auto-selected per (client, volume) pair):

=============  ================================  =========================
type           when                              mechanism
=============  ================================  =========================
HIP_IPC        same node, both sides have GPUs   hipIpcMemHandle export +
                                                 hipMemcpyPeerAsync on HIP
                                                 streams over xGMI
SHARED_MEMORY  same host                         POSIX SHM segments, pinned
                                                 via hipHostRegister, HIP
                                                 copy streams for D2H/H2D
RCCL           cross-host, GPU tensors           2-rank RCCL communicator
GLOO           cross-host, CPU tensors           2-rank gloo process group
RPC            always                            payload inside the RPC
                                                 frame (out-of-band bufs)
=============  ================================  =========================
"""

from __future__ import annotations

import os
from abc import ABC, abstractmethod
from enum import Enum
from typing import Any, Dict, List, Optional, Sequence, Type

import torch

from torchstore_amd.types import Request
from torchstore_amd.utils.logging import get_logger

logger = get_logger("torchstore_amd.transport")


class TransportType(Enum):
    HIP_IPC = "hip_ipc"
    SHARED_MEMORY = "shared_memory"
    RCCL = "rccl"
    GLOO = "gloo"
    RPC = "rpc"


class TransportCache(ABC):
    """Long-lived per-process state a transport keeps between operations."""

    @abstractmethod
    def drop_key(self, key: str) -> None: ...

    def close(self) -> None:
        return None


class TransportContext:
    """Type-keyed registry of transport caches, lazily created.

    Both the client process and every volume process own one; buffers look
    their side's caches up here (reference: ``buffers.py:39-69``).
    """

    def __init__(self):
        import uuid

        self._caches: Dict[type, TransportCache] = {}
        # stable per-context identity (e.g. volume-side caches key client
        # resources by it so concurrent clients never share buffers)
        self.uid = uuid.uuid4().hex

    def cache(self, cls: Type[TransportCache]) -> TransportCache:
        inst = self._caches.get(cls)
        if inst is None:
            inst = cls()
            self._caches[cls] = inst
        return inst

    def drop_key(self, key: str) -> None:
        for cache in self._caches.values():
            cache.drop_key(key)

    def close(self) -> None:
        for cache in self._caches.values():
            cache.close()
        self._caches.clear()


class TransportBuffer(ABC):
    """One put/get operation's worth of transport state.

    The object itself crosses the RPC boundary; ``__getstate__`` of concrete
    subclasses strips client-local tensors/handles that must not travel.
    """

    transport_type: TransportType = TransportType.RPC
    requires_handshake: bool = False

    def __init__(self):
        self._volume_ref = None       # client-side only
        self._client_ctx: Optional[TransportContext] = None
        self._volume_ctx: Optional[TransportContext] = None

    # -- wiring ----------------------------------------------------------
    def bind_client(self, volume_ref, ctx: TransportContext) -> None:
        self._volume_ref = volume_ref
        self._client_ctx = ctx

    def attach_volume(self, ctx: TransportContext) -> None:
        """Called volume-side before volume_receive/volume_send."""
        self._volume_ctx = ctx

    def __getstate__(self):
        state = self.__dict__.copy()
        state["_volume_ref"] = None
        state["_client_ctx"] = None
        state["_volume_ctx"] = None
        return state

    # -- client-side hooks ----------------------------------------------
    async def client_stage_put(self, requests: Sequence[Request]) -> None:
        return None

    async def client_stage_get(self, requests: Sequence[Request]) -> None:
        return None

    def on_handshake_reply(self, reply: Any, phase: str) -> None:
        return None

    def on_success(self) -> None:
        """Publish handshake-private resources into long-lived caches."""
        return None

    async def drop(self) -> None:
        """Release per-op resources (always runs, even on failure)."""
        return None

    # -- volume-side hooks -----------------------------------------------
    def recv_handshake(self, requests: Sequence[Request], phase: str, store) -> Any:
        return None

    @abstractmethod
    async def volume_receive(
        self,
        requests: Sequence[Request],
        existing: Sequence[Optional[torch.Tensor]],
        device: torch.device,
    ) -> List[Any]:
        """Materialise each request's payload on the volume (tensor or object).

        ``existing[i]`` is a same-shape/dtype tensor already stored under the
        key (in-place overwrite fast path) or None.  ``device`` is where the
        store keeps tensors.
        """

    @abstractmethod
    async def volume_send(
        self, requests: Sequence[Request], values: Sequence[Any]
    ) -> Any:
        """Move fetched values toward the client; return the RPC reply."""

    # -- client-side completion ------------------------------------------
    @abstractmethod
    def client_complete_get(
        self, requests: Sequence[Request], reply: Any
    ) -> List[Any]:
        """Produce the final per-request values on the client."""

    # -- orchestration (shared by all transports) -------------------------
    async def put(self, requests: Sequence[Request]) -> None:
        volume = self._volume_ref.volume
        metas = [r.meta_only() for r in requests]
        try:
            if self.requires_handshake:
                reply = await volume.handshake.call_one(self, metas, "put")
                self.on_handshake_reply(reply, "put")
            await self.client_stage_put(requests)
            await volume.put.call_one(self, metas)
            self.on_success()
        finally:
            await self.drop()

    async def get(self, requests: Sequence[Request]) -> List[Any]:
        volume = self._volume_ref.volume
        metas = [r.meta_only() for r in requests]
        try:
            if self.requires_handshake:
                reply = await volume.handshake.call_one(self, metas, "get")
                self.on_handshake_reply(reply, "get")
            await self.client_stage_get(requests)
            reply = await volume.get.call_one(self, metas)
            out = self.client_complete_get(requests, reply)
            self.on_success()
            return out
        finally:
            await self.drop()


def _env_on(name: str, default: str = "1") -> bool:
    return os.environ.get(name, default) not in ("0", "false", "False", "")
