"""Transport factory: auto-selects how bytes move per (client, volume) pair.

Priority on MI355X:
  1. HIP_IPC       — same node and both ends GPU-capable: hipIpcMemHandle +
                     hipMemcpyPeerAsync over xGMI (one-sided, zero staging)
  2. SHARED_MEMORY — same host: POSIX SHM + pinned HIP copy streams
  3. RCCL          — cross-host GPU tensors (2-rank communicator)
  4. GLOO          — cross-host CPU tensors
  5. RPC           — always works: payload inside the RPC frame

Env gates (all default-enabled where applicable):
  TORCHSTORE_AMD_IPC_ENABLED, TORCHSTORE_AMD_SHM_ENABLED,
  TORCHSTORE_AMD_RCCL_ENABLED, TORCHSTORE_AMD_GLOO_ENABLED
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed

from torchstore_amd.transport.base import (
    TransportBuffer,
    TransportContext,
    TransportType,
    _env_on,
)
from torchstore_amd.transport.rpc import RpcTransportBuffer
from torchstore_amd.utils.logging import get_logger

logger = get_logger("torchstore_amd.transport")

__all__ = [
    "TransportBuffer",
    "TransportContext",
    "TransportType",
    "create_transport",
    "resolve_transport_type",
]


def _registry():
    from torchstore_amd.transport.shm import ShmTransportBuffer
    from torchstore_amd.transport.hip_ipc import HipIpcTransportBuffer
    from torchstore_amd.transport.pg import (
        GlooTransportBuffer,
        RcclTransportBuffer,
    )
    return {
        TransportType.RPC: RpcTransportBuffer,
        TransportType.SHARED_MEMORY: ShmTransportBuffer,
        TransportType.HIP_IPC: HipIpcTransportBuffer,
        TransportType.GLOO: GlooTransportBuffer,
        TransportType.RCCL: RcclTransportBuffer,
    }


def _ipc_available(volume_ref) -> bool:
    if not _env_on("TORCHSTORE_AMD_IPC_ENABLED"):
        return False
    if not volume_ref.is_local:
        return False
    if not torch.cuda.is_available() or not volume_ref.device.startswith("cuda"):
        return False
    from torchstore_amd.ops import gpu

    return gpu.extension_available()


def _shm_available(volume_ref) -> bool:
    return _env_on("TORCHSTORE_AMD_SHM_ENABLED") and volume_ref.is_local


def resolve_transport_type(volume_ref) -> TransportType:
    if volume_ref.transport_type is not None:
        return volume_ref.transport_type
    if _ipc_available(volume_ref):
        return TransportType.HIP_IPC
    if _shm_available(volume_ref):
        return TransportType.SHARED_MEMORY
    if not volume_ref.is_local:
        if (
            _env_on("TORCHSTORE_AMD_RCCL_ENABLED")
            and torch.cuda.is_available()
            and volume_ref.device.startswith("cuda")
        ):
            return TransportType.RCCL
        if _env_on("TORCHSTORE_AMD_GLOO_ENABLED") and torch.distributed.is_gloo_available():
            return TransportType.GLOO
    return TransportType.RPC


_logged_resolutions = set()


def create_transport(volume_ref, ctx: Optional[TransportContext] = None) -> TransportBuffer:
    ttype = resolve_transport_type(volume_ref)
    if (volume_ref.volume_id, ttype) not in _logged_resolutions:
        _logged_resolutions.add((volume_ref.volume_id, ttype))
        logger.info(
            "transport to volume %s (%s, %s): %s",
            volume_ref.volume_id, volume_ref.hostname, volume_ref.device,
            ttype.value,
        )
    reg = _registry()
    cls = reg.get(ttype)
    if cls is None:
        raise NotImplementedError(f"transport {ttype} not implemented yet")
    buf = cls()
    buf.bind_client(volume_ref, ctx or volume_ref.transport_context)
    return buf
