"""Universal fallback transport: payload rides inside the RPC frame.

The runtime's serializer already moves tensors as out-of-band zero-copy
buffers (GPU tensors staged through CPU), so this is bandwidth-reasonable on
loopback but never the fast path — it exists so every (client, volume) pair
works with zero native I/O, mirroring the reference's Monarch-RPC tier
(torchstore ``transport/monarch_rpc.py``).
"""

from __future__ import annotations

from typing import Any, List, Optional, Sequence

import torch

from torchstore_amd.transport.base import TransportBuffer, TransportType
from torchstore_amd.types import Request


class RpcTransportBuffer(TransportBuffer):
    transport_type = TransportType.RPC
    requires_handshake = False

    def __init__(self):
        super().__init__()
        # payload travelling client→volume: list aligned with requests
        self.data: Optional[List[Any]] = None

    async def client_stage_put(self, requests: Sequence[Request]) -> None:
        payload = []
        for r in requests:
            if r.is_object:
                payload.append(("obj", r.objects))
            else:
                payload.append(("tensor", r.tensor_val))
        self.data = payload

    async def volume_receive(self, requests, existing, device):
        assert self.data is not None, "put arrived without payload"
        out = []
        for (kind, value), prior in zip(self.data, existing):
            if kind == "obj":
                out.append(value)
                continue
            t = value
            if prior is not None and prior.shape == t.shape and prior.dtype == t.dtype:
                prior.copy_(t)
                out.append(prior)
            else:
                out.append(t.to(device))
        return out

    async def volume_send(self, requests, values):
        # serializer stages GPU-resident stored tensors through CPU
        return list(values)

    def client_complete_get(self, requests, reply) -> List[Any]:
        out = []
        for r, value in zip(requests, reply):
            if r.is_object or not isinstance(value, torch.Tensor):
                out.append(value)
                continue
            if r.tensor_val is not None:
                r.tensor_val.copy_(value)
                out.append(r.tensor_val)
            else:
                out.append(value)
        return out
