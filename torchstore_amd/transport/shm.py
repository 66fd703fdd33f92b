"""Same-host transport over POSIX shared memory.

MI355X adaptation of the reference SHM transport (torchstore
``transport/shared_memory.py``):

* segments are torch filename-shared CPU storages
  (``UntypedStorage._new_using_filename_cpu``), attached on the peer by
  ``_new_shared_filename_cpu`` — zero serialization of payload bytes;
* GPU↔segment copies run on **dedicated per-device HIP copy streams** and
  synchronize only those streams (warm puts never stall unrelated streams —
  the reference's stream-isolation invariant, ``test_shared_memory.py:1034``);
* segments touched by GPU copies are page-pinned via ``hipHostRegister``
  (``torch.cuda.cudart()`` maps to HIP on ROCm), fail-open with a
  once-per-error warning;
* both sides keep caches: the client reuses put segments per key and attach
  mappings per segment name; the volume reuses attach mappings, records
  which stored tensors *are* segments (so warm gets of SHM-stored keys are
  volume-side zero-copy), and reuses response segments per (key, region).

Unlike the reference there is no handshake RPC: descriptors ride the data
RPC and both sides' caches make reuse decisions locally — one round trip
fewer per operation.
"""

from __future__ import annotations

import warnings
from dataclasses import dataclass
from typing import Any, Dict, List, Optional, Sequence, Tuple

import torch

from torchstore_amd.ops.slicing import byte_view
from torchstore_amd.transport.base import (
    TransportBuffer,
    TransportCache,
    TransportType,
)
from torchstore_amd.types import Request
from torchstore_amd.utils.logging import get_logger

logger = get_logger("torchstore_amd.shm")

_PIN_WARNED: set = set()
_HIP_HOST_REGISTER_PORTABLE = 1


@dataclass(frozen=True)
class ShmDescriptor:
    manager: bytes
    name: bytes
    nbytes: int
    dtype: torch.dtype
    shape: Tuple[int, ...]

    @property
    def seg_key(self) -> Tuple[bytes, bytes]:
        return (self.manager, self.name)


def _allocate_segment(nbytes: int) -> torch.Tensor:
    storage = torch.UntypedStorage._new_using_filename_cpu(nbytes)
    return torch.empty(0, dtype=torch.uint8).set_(storage)


def _segment_handle(seg: torch.Tensor) -> Tuple[bytes, bytes, int]:
    manager, name, size = seg.untyped_storage()._share_filename_cpu_()
    return manager, name, size


def _attach_segment(desc: ShmDescriptor) -> torch.Tensor:
    storage = torch.UntypedStorage._new_shared_filename_cpu(
        desc.manager, desc.name, desc.nbytes
    )
    return torch.empty(0, dtype=torch.uint8).set_(storage)


def _typed_view(seg_u8: torch.Tensor, desc: ShmDescriptor) -> torch.Tensor:
    numel = 1
    for s in desc.shape:
        numel *= s
    if numel == 0:
        return torch.empty(desc.shape, dtype=desc.dtype)
    return seg_u8[: numel * torch._utils._element_size(desc.dtype)].view(
        desc.dtype
    ).reshape(desc.shape)


def _try_pin(seg_u8: torch.Tensor, pinned: Dict[int, int]) -> None:
    """Pin a segment's pages for DMA; fail-open (copies still work unpinned)."""
    if not torch.cuda.is_available():
        return
    ptr = seg_u8.untyped_storage().data_ptr()
    if ptr in pinned:
        return
    nbytes = seg_u8.untyped_storage().nbytes()
    try:
        rc = torch.cuda.cudart().cudaHostRegister(
            ptr, nbytes, _HIP_HOST_REGISTER_PORTABLE
        )
        code = int(rc) if not hasattr(rc, "value") else int(rc.value)
    except Exception as exc:  # noqa: BLE001
        code = -1
        rc = exc
    if code == 0:
        pinned[ptr] = nbytes
    elif code not in _PIN_WARNED:
        _PIN_WARNED.add(code)
        warnings.warn(
            f"hipHostRegister failed (code {rc}); shared-memory GPU copies "
            "will run unpinned (slower). This warning is shown once per code."
        )


def _unpin_all(pinned: Dict[int, int]) -> None:
    if not pinned or not torch.cuda.is_available():
        pinned.clear()
        return
    for ptr in list(pinned):
        try:
            torch.cuda.cudart().cudaHostUnregister(ptr)
        except Exception:  # noqa: BLE001
            pass
    pinned.clear()


class _CopyStreams:
    """One HIP stream per device, used ONLY for SHM staging copies."""

    def __init__(self):
        self._streams: Dict[int, torch.cuda.Stream] = {}

    def get(self, device_index: int) -> torch.cuda.Stream:
        s = self._streams.get(device_index)
        if s is None:
            s = torch.cuda.Stream(device=device_index)
            self._streams[device_index] = s
        return s


def _copy_bytes(dst_u8: torch.Tensor, src: torch.Tensor, streams: _CopyStreams) -> Optional[torch.cuda.Stream]:
    """Copy ``src`` (any device/dtype, contiguous) into a CPU byte span.

    Returns the stream used (caller synchronizes it) or None for CPU→CPU.
    """
    src_c = src.contiguous()
    src_u8 = byte_view(src_c)
    if src.device.type == "cuda":
        stream = streams.get(src.device.index)
        # order after pending default-stream work that may produce src
        stream.wait_stream(torch.cuda.current_stream(src.device))
        with torch.cuda.stream(stream):
            dst_u8[: src_u8.numel()].copy_(src_u8, non_blocking=True)
            # keep a possibly-temporary contiguous src alive for the copy
            src_c.record_stream(stream)
        return stream
    dst_u8[: src_u8.numel()].copy_(src_u8)
    return None


class ShmClientCache(TransportCache):
    def __init__(self):
        self.put_segments: Dict[str, Tuple[ShmDescriptor, torch.Tensor]] = {}
        self.attached: Dict[Tuple[bytes, bytes], torch.Tensor] = {}
        self.streams = _CopyStreams()
        self.pinned: Dict[int, int] = {}

    def drop_key(self, key: str) -> None:
        self.put_segments.pop(key, None)

    def close(self) -> None:
        _unpin_all(self.pinned)
        self.put_segments.clear()
        self.attached.clear()


class ShmVolumeCache(TransportCache):
    def __init__(self):
        self.attached: Dict[Tuple[bytes, bytes], torch.Tensor] = {}
        self.desc_by_storage: Dict[int, ShmDescriptor] = {}
        self.get_segments: Dict[Any, Tuple[ShmDescriptor, torch.Tensor]] = {}
        self.streams = _CopyStreams()
        self.pinned: Dict[int, int] = {}

    def drop_key(self, key: str) -> None:
        for k in [k for k in self.get_segments if k[0] == key]:
            del self.get_segments[k]

    def close(self) -> None:
        _unpin_all(self.pinned)
        self.attached.clear()
        self.desc_by_storage.clear()
        self.get_segments.clear()


class ShmTransportBuffer(TransportBuffer):
    transport_type = TransportType.SHARED_MEMORY
    requires_handshake = False

    def __init__(self):
        super().__init__()
        # aligned with requests: ("obj", value) | ("shm", ShmDescriptor)
        self.payload: Optional[List[Tuple[str, Any]]] = None

    # -- client put -------------------------------------------------------
    async def client_stage_put(self, requests: Sequence[Request]) -> None:
        cache: ShmClientCache = self._client_ctx.cache(ShmClientCache)
        payload: List[Tuple[str, Any]] = []
        streams_used = []
        for r in requests:
            if r.is_object:
                payload.append(("obj", r.objects))
                continue
            t = r.tensor_val
            nbytes = t.numel() * t.element_size()
            entry = cache.put_segments.get(r.key)
            if entry is None or entry[0].nbytes < nbytes:
                seg = _allocate_segment(max(nbytes, 1))
                manager, name, size = _segment_handle(seg)
                desc = ShmDescriptor(manager, name, size, t.dtype, tuple(t.shape))
                cache.put_segments[r.key] = (desc, seg)
            else:
                desc, seg = entry
                desc = ShmDescriptor(
                    desc.manager, desc.name, desc.nbytes, t.dtype, tuple(t.shape)
                )
                cache.put_segments[r.key] = (desc, seg)
            if t.device.type == "cuda":
                _try_pin(seg, cache.pinned)
            stream = _copy_bytes(seg, t, cache.streams)
            if stream is not None:
                streams_used.append(stream)
            payload.append(("shm", desc))
        for s in set(streams_used):
            s.synchronize()
        self.payload = payload

    # -- volume put -------------------------------------------------------
    async def volume_receive(self, requests, existing, device):
        cache: ShmVolumeCache = self._volume_ctx.cache(ShmVolumeCache)
        out: List[Any] = []
        for (kind, value), prior in zip(self.payload, existing):
            if kind == "obj":
                out.append(value)
                continue
            desc: ShmDescriptor = value
            seg = cache.attached.get(desc.seg_key)
            if seg is None:
                seg = _attach_segment(desc)
                cache.attached[desc.seg_key] = seg
                cache.desc_by_storage[seg.untyped_storage().data_ptr()] = desc
            typed = _typed_view(seg, desc)
            if device.type == "cuda":
                _try_pin(seg, cache.pinned)
                if (
                    prior is not None
                    and prior.shape == typed.shape
                    and prior.dtype == typed.dtype
                    and prior.device == device
                ):
                    prior.copy_(typed, non_blocking=True)
                    torch.cuda.synchronize(device)
                    out.append(prior)
                else:
                    gpu_t = typed.to(device)
                    out.append(gpu_t)
            else:
                # CPU store adopts the segment — zero-copy warm path
                out.append(typed)
        return out

    # -- volume get -------------------------------------------------------
    async def volume_send(self, requests, values):
        cache: ShmVolumeCache = self._volume_ctx.cache(ShmVolumeCache)
        reply: List[Tuple[str, Any]] = []
        streams_used = []
        for r, v in zip(requests, values):
            if not isinstance(v, torch.Tensor):
                reply.append(("obj", v))
                continue
            nbytes = v.numel() * v.element_size()
            # zero-copy: the stored tensor IS a full attached segment
            if v.device.type == "cpu" and v.is_contiguous() and v.storage_offset() == 0:
                known = cache.desc_by_storage.get(v.untyped_storage().data_ptr())
                if known is not None and known.nbytes >= nbytes:
                    reply.append(
                        ("shm", ShmDescriptor(
                            known.manager, known.name, known.nbytes,
                            v.dtype, tuple(v.shape),
                        ))
                    )
                    continue
            region = (
                r.key,
                r.tensor_slice.offsets if r.tensor_slice else None,
                tuple(v.shape),
            )
            entry = cache.get_segments.get(region)
            if entry is None or entry[0].nbytes < nbytes:
                seg = _allocate_segment(max(nbytes, 1))
                manager, name, size = _segment_handle(seg)
                desc = ShmDescriptor(manager, name, size, v.dtype, tuple(v.shape))
                cache.get_segments[region] = (desc, seg)
            else:
                desc, seg = entry
                desc = ShmDescriptor(
                    desc.manager, desc.name, desc.nbytes, v.dtype, tuple(v.shape)
                )
            if v.device.type == "cuda":
                _try_pin(seg, cache.pinned)
            stream = _copy_bytes(seg, v, cache.streams)
            if stream is not None:
                streams_used.append(stream)
            reply.append(("shm", desc))
        for s in set(streams_used):
            s.synchronize()
        return reply

    # -- client get completion -------------------------------------------
    def client_complete_get(self, requests, reply) -> List[Any]:
        cache: ShmClientCache = self._client_ctx.cache(ShmClientCache)
        out: List[Any] = []
        streams_used = []
        for r, (kind, value) in zip(requests, reply):
            if kind == "obj":
                out.append(value)
                continue
            desc: ShmDescriptor = value
            seg = cache.attached.get(desc.seg_key)
            if seg is None:
                seg = _attach_segment(desc)
                cache.attached[desc.seg_key] = seg
            typed = _typed_view(seg, desc)
            dest = r.tensor_val
            if dest is None:
                out.append(typed.clone())
                continue
            if dest.device.type == "cuda":
                _try_pin(seg, cache.pinned)
                stream = cache.streams.get(dest.device.index)
                stream.wait_stream(torch.cuda.current_stream(dest.device))
                with torch.cuda.stream(stream):
                    dest.copy_(typed, non_blocking=True)
                streams_used.append(stream)
            else:
                dest.copy_(typed)
            out.append(dest)
        for s in set(streams_used):
            s.synchronize()
        return out
