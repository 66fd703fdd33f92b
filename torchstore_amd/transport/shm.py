"""Same-host transport over POSIX shared memory.

MI355X adaptation of the reference SHM transport (torchstore
``transport/shared_memory.py``) with one deliberate ownership inversion:
**every segment is allocated by the (long-lived) storage volume**; clients
only attach.  torch's filename-SHM manager makes the *allocating* process
block on exit while peers hold attachments (~30 s), so short-lived client
processes must never own segments.  A put therefore does a handshake RPC in
which the volume allocates (or reuses, keyed by key) the segments and
returns descriptors; the client attaches and copies in; the data RPC then
just tells the volume to adopt the bytes.

Other properties:

* GPU↔segment copies run on dedicated per-device HIP copy streams and
  synchronize only those streams (warm puts never stall unrelated streams —
  the reference's stream-isolation invariant, ``test_shared_memory.py:1034``);
* segments touched by GPU copies are page-pinned via ``hipHostRegister``
  (``torch.cuda.cudart()`` maps to HIP on ROCm), fail-open with a
  once-per-error warning;
* a CPU volume *adopts* the typed view of its own segment as storage, so a
  warm get of an SHM-stored key returns the descriptor with **zero**
  volume-side copies;
* both sides cache attach mappings by segment name; the volume reuses
  response segments per (key, region).
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

# opt-in zero-copy gets: return the LIVE typed view of the volume's segment
# instead of a clone (the reference's TORCHSTORE_MUTABLE_SHM semantics —
# later puts to the key become visible through previously returned tensors)
def _mutable_shm() -> bool:
    import os

    return os.environ.get("TORCHSTORE_AMD_MUTABLE_SHM", "0") == "1"
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

    def with_layout(self, dtype: torch.dtype, shape) -> "ShmDescriptor":
        return ShmDescriptor(self.manager, self.name, self.nbytes, dtype, tuple(shape))


def _allocate_segment(nbytes: int) -> torch.Tensor:
    storage = torch.UntypedStorage._new_using_filename_cpu(nbytes)
    return torch.empty(0, dtype=torch.uint8).set_(storage)


def _segment_descriptor(seg: torch.Tensor) -> ShmDescriptor:
    manager, name, size = seg.untyped_storage()._share_filename_cpu_()
    return ShmDescriptor(manager, name, size, torch.uint8, (size,))


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
    esize = torch._utils._element_size(desc.dtype)
    return seg_u8[: numel * esize].view(desc.dtype).reshape(desc.shape)


def _pin_enabled() -> bool:
    import os

    # reference parity: TORCHSTORE_PIN_SHM opt-out (shared_memory.py:50-53)
    return os.environ.get("TORCHSTORE_AMD_PIN_SHM", "1") != "0"


def _try_pin(seg_u8: torch.Tensor, pinned: Dict[int, int]) -> None:
    """Pin a segment's pages for DMA; fail-open (copies still work unpinned)."""
    if not torch.cuda.is_available() or not _pin_enabled():
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


def _copy_into_u8(
    dst_u8: torch.Tensor, src: torch.Tensor, streams: _CopyStreams
) -> Optional[torch.cuda.Stream]:
    """Copy ``src`` (any device, made contiguous) into a CPU byte span."""
    src_c = src.contiguous()
    src_u8 = byte_view(src_c)
    if src.device.type == "cuda":
        from torchstore_amd.ops import gpu as gpu_ops

        try:
            ext = gpu_ops.ext()
        except RuntimeError:
            ext = None
        if ext is not None:
            # direct SDMA into the (registered) segment: ~2x torch's copy_
            # into externally pinned host memory
            torch.cuda.current_stream(src.device).synchronize()
            ext.sdma_copy(
                dst_u8.data_ptr(), src_u8.data_ptr(), src_u8.numel(),
                src.device.index,
            )
            return None
        stream = streams.get(src.device.index)
        stream.wait_stream(torch.cuda.current_stream(src.device))
        with torch.cuda.stream(stream):
            dst_u8[: src_u8.numel()].copy_(src_u8, non_blocking=True)
            src_c.record_stream(stream)
        return stream
    dst_u8[: src_u8.numel()].copy_(src_u8)
    return None


def _copy_from_u8(
    dest: torch.Tensor, typed_src: torch.Tensor, streams: _CopyStreams
) -> Optional[torch.cuda.Stream]:
    """Copy a typed CPU view into dest (any device, may be strided)."""
    if dest.device.type == "cuda":
        stream = streams.get(dest.device.index)
        stream.wait_stream(torch.cuda.current_stream(dest.device))
        with torch.cuda.stream(stream):
            dest.copy_(typed_src, non_blocking=True)
        return stream
    dest.copy_(typed_src)
    return None


class ShmClientCache(TransportCache):
    def __init__(self):
        self.attached: Dict[Tuple[bytes, bytes], torch.Tensor] = {}
        self.streams = _CopyStreams()
        self.pinned: Dict[int, int] = {}

    def attach(self, desc: ShmDescriptor) -> torch.Tensor:
        seg = self.attached.get(desc.seg_key)
        if seg is None:
            seg = _attach_segment(desc)
            self.attached[desc.seg_key] = seg
        return seg

    def drop_key(self, key: str) -> None:
        return None  # attachments are segment-level

    def close(self) -> None:
        _unpin_all(self.pinned)
        self.attached.clear()


class ShmVolumeCache(TransportCache):
    """Volume side owns every segment."""

    def __init__(self):
        # key -> (descriptor, segment) for put targets
        self.put_segments: Dict[str, Tuple[ShmDescriptor, torch.Tensor]] = {}
        # (key, region) -> (descriptor, segment) for get responses
        self.get_segments: Dict[Any, Tuple[ShmDescriptor, torch.Tensor]] = {}
        # storage ptr -> descriptor (zero-copy warm gets of adopted entries)
        self.desc_by_storage: Dict[int, ShmDescriptor] = {}
        self.streams = _CopyStreams()
        self.pinned: Dict[int, int] = {}

    def obtain(
        self, table: Dict, cache_key, nbytes: int
    ) -> Tuple[ShmDescriptor, torch.Tensor]:
        entry = table.get(cache_key)
        if entry is None or entry[0].nbytes < nbytes:
            seg = _allocate_segment(max(nbytes, 1))
            desc = _segment_descriptor(seg)
            table[cache_key] = (desc, seg)
            self.desc_by_storage[seg.untyped_storage().data_ptr()] = desc
            return desc, seg
        return entry

    def drop_key(self, key: str) -> None:
        for k in [k for k in self.put_segments if k[1] == key]:
            desc, seg = self.put_segments.pop(k)
            self.desc_by_storage.pop(seg.untyped_storage().data_ptr(), None)
        for k in [k for k in self.get_segments if k[1] == key]:
            desc, seg = self.get_segments.pop(k)
            self.desc_by_storage.pop(seg.untyped_storage().data_ptr(), None)

    def close(self) -> None:
        _unpin_all(self.pinned)
        self.put_segments.clear()
        self.get_segments.clear()
        self.desc_by_storage.clear()


class ShmTransportBuffer(TransportBuffer):
    transport_type = TransportType.SHARED_MEMORY
    requires_handshake = False  # put() below runs its own handshake phase

    def __init__(self):
        super().__init__()
        # put: aligned with requests, ("seg", ShmDescriptor) | ("obj", value)
        # (descriptors are produced volume-side in the handshake)
        self.payload: Optional[List[Tuple[str, Any]]] = None
        # put handshake: per-request nbytes to allocate (None for objects)
        self.alloc_sizes: Optional[List[Optional[int]]] = None
        # client identity: volume-side put segments are keyed (client, key)
        # so concurrent clients writing one key never share a segment
        self.client_uid: str = ""

    # -- handshake (volume allocates put segments) ------------------------
    def recv_handshake(self, requests: Sequence[Request], phase: str, volume):
        if phase != "put":
            return None
        cache: ShmVolumeCache = self._volume_ctx.cache(ShmVolumeCache)
        out: List[Optional[ShmDescriptor]] = []
        for r, nbytes in zip(requests, self.alloc_sizes):
            if nbytes is None:
                out.append(None)
                continue
            desc, _seg = cache.obtain(
                cache.put_segments, (self.client_uid, r.key), nbytes
            )
            out.append(desc)
        return out

    # -- client put -------------------------------------------------------
    async def put(self, requests: Sequence[Request]) -> None:
        volume = self._volume_ref.volume
        metas = [r.meta_only() for r in requests]
        self.alloc_sizes = [
            None if r.is_object else r.nbytes() for r in requests
        ]
        self.client_uid = self._client_ctx.uid
        try:
            reply = await volume.handshake.call_one(self, metas, "put")
            cache: ShmClientCache = self._client_ctx.cache(ShmClientCache)
            payload: List[Tuple[str, Any]] = []
            streams_used = []
            for r, desc in zip(requests, reply):
                if r.is_object or desc is None:
                    payload.append(("obj", r.objects))
                    continue
                t = r.tensor_val
                seg = cache.attach(desc)
                if t.device.type == "cuda":
                    _try_pin(seg, cache.pinned)
                stream = _copy_into_u8(seg, t, cache.streams)
                if stream is not None:
                    streams_used.append(stream)
                payload.append(
                    ("seg", desc.with_layout(t.dtype, t.shape))
                )
            for s in set(streams_used):
                s.synchronize()
            self.payload = payload
            await volume.put.call_one(self, [r.meta_only() for r in requests])
        finally:
            await self.drop()

    # -- volume put -------------------------------------------------------
    async def volume_receive(self, requests, existing, device):
        cache: ShmVolumeCache = self._volume_ctx.cache(ShmVolumeCache)
        out: List[Any] = []
        streams_used = []
        for r, (kind, value), prior in zip(requests, self.payload, existing):
            if kind == "obj":
                out.append(value)
                continue
            desc: ShmDescriptor = value
            cached = cache.put_segments.get((self.client_uid, r.key))
            if cached is not None and cached[0].seg_key == desc.seg_key:
                seg = cached[1]
            else:
                # shouldn't happen (handshake allocated it) — re-attach safely
                seg = _attach_segment(desc)
            typed = _typed_view(seg, desc)
            if device.type == "cuda":
                _try_pin(seg, cache.pinned)
                if (
                    prior is not None
                    and prior.shape == typed.shape
                    and prior.dtype == typed.dtype
                    and prior.device == device
                ):
                    stream = _copy_from_u8(prior, typed, cache.streams)
                    if stream is not None:
                        streams_used.append(stream)
                    out.append(prior)
                else:
                    out.append(typed.to(device))
            else:
                # CPU store adopts the volume-owned segment — zero-copy
                out.append(typed)
        for s in set(streams_used):
            s.synchronize()
        return out

    # -- client get (stage): stamp the client identity ---------------------
    async def client_stage_get(self, requests: Sequence[Request]) -> None:
        # get-response segments are keyed per client: a get concurrent with
        # another client's get of the same key must never share a response
        # segment (torn values for the slower reader otherwise)
        self.client_uid = self._client_ctx.uid

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
            # zero-copy: the stored tensor IS (a prefix of) an owned segment
            if (
                v.device.type == "cpu"
                and v.is_contiguous()
                and v.storage_offset() == 0
            ):
                known = cache.desc_by_storage.get(v.untyped_storage().data_ptr())
                if known is not None and known.nbytes >= nbytes:
                    reply.append(("seg", known.with_layout(v.dtype, v.shape)))
                    continue
            region = (
                self.client_uid,
                r.key,
                r.tensor_slice.offsets if r.tensor_slice else None,
                tuple(v.shape),
            )
            desc, seg = cache.obtain(cache.get_segments, region, nbytes)
            if v.device.type == "cuda":
                _try_pin(seg, cache.pinned)
            stream = _copy_into_u8(seg, v, cache.streams)
            if stream is not None:
                streams_used.append(stream)
            reply.append(("seg", desc.with_layout(v.dtype, v.shape)))
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
            seg = cache.attach(desc)
            typed = _typed_view(seg, desc)
            dest = r.tensor_val
            if dest is None or (r.dest_owned and _mutable_shm()):
                out.append(typed if _mutable_shm() else typed.clone())
                continue
            if dest.device.type == "cuda":
                _try_pin(seg, cache.pinned)
            stream = _copy_from_u8(dest, typed, cache.streams)
            if stream is not None:
                streams_used.append(stream)
            out.append(dest)
        for s in set(streams_used):
            s.synchronize()
        return out
