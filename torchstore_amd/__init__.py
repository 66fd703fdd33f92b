"""torchstore_amd — an MI355X-native distributed tensor store.

Async key→tensor/object store for PyTorch-ROCm with DTensor-aware
resharding, state_dict exchange and zero-copy direct weight sync, built on:

* an asyncio actor/RPC runtime (64-bit frames, out-of-band tensor buffers);
* auto-selected transports: HIP IPC over xGMI (same node, GPU↔GPU),
  POSIX shared memory with pinned HIP copy streams (same host), RCCL
  send/recv (cross-host GPU), payload-in-RPC fallback;
* hand-written CDNA4 (gfx950) HIP kernels for slice gather / scatter
  assembly / fused dtype cast;
* GPU-resident storage volumes sized for 288 GB HBM3E per MI355X.

Capability parity target: meta-pytorch/torchstore.
"""

from torchstore_amd.api import (
    attach,
    client,
    delete,
    delete_batch,
    exists,
    get,
    get_batch,
    get_state_dict,
    initialize,
    keys,
    put,
    put_batch,
    put_state_dict,
    reset_client,
    shutdown,
    stats,
)
from torchstore_amd.spmd import SPMDEnv, initialize_spmd, shutdown_spmd
from torchstore_amd.strategy import (
    HostStrategy,
    LocalRankStrategy,
    PlacementStrategy,
    SingletonStrategy,
)
from torchstore_amd.transport import TransportType
from torchstore_amd.types import LocalShard, Request, TensorSlice
from torchstore_amd.utils.logging import init_logging

init_logging()

__version__ = "0.1.0"

__all__ = [
    "attach",
    "init_logging",
    "client",
    "delete",
    "delete_batch",
    "exists",
    "get",
    "get_batch",
    "get_state_dict",
    "initialize",
    "keys",
    "put",
    "put_batch",
    "put_state_dict",
    "reset_client",
    "shutdown",
    "stats",
    "SPMDEnv",
    "initialize_spmd",
    "shutdown_spmd",
    "HostStrategy",
    "LocalRankStrategy",
    "PlacementStrategy",
    "SingletonStrategy",
    "TransportType",
    "LocalShard",
    "Request",
    "TensorSlice",
]
