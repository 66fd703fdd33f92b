"""Tensor-aware RPC serialization with out-of-band (zero-copy) payloads.

Replaces Monarch's Rust codec (the reference bumps
``HYPERACTOR_CODEC_MAX_FRAME_LENGTH`` to ~900 GB so RPC frames can carry
tensors, torchstore ``__init__.py:37-44``).  Here frames have 64-bit lengths
from the start and tensor payloads travel as separate buffers:

    header  = pickle protocol 5 stream (PickleBuffer placeholders inside)
    buffers = one raw byte span per (CPU, contiguous view of a) tensor

* Serialize: ``dumps(obj)`` → ``(header: bytes, buffers: list[memoryview])``.
  No tensor bytes are copied for CPU contiguous tensors — the memoryview
  aliases the tensor storage, and the socket writer sends it directly.
* Deserialize: the reader allocates one writable ``bytearray`` per buffer,
  reads straight off the socket into it, and ``loads`` rebuilds tensors as
  zero-copy ``torch.frombuffer`` views of those bytearrays.

GPU tensors are staged to CPU here (this is the *fallback* payload path —
bulk GPU traffic rides the SHM / HIP-IPC transports instead), and rebuilt on
CPU with their original device recorded in ``tensor.__dict__`` untouched.
``__getstate__``/``__setstate__`` of arbitrary objects (e.g. transport
buffers stripping local state) work exactly as with plain pickle.
"""

from __future__ import annotations

import io
import pickle
from typing import Any, List, Sequence, Tuple

import torch


def _rebuild_tensor(dtype: torch.dtype, shape: Tuple[int, ...], buf) -> torch.Tensor:
    numel = 1
    for s in shape:
        numel *= s
    if numel == 0:
        return torch.empty(shape, dtype=dtype)
    if isinstance(buf, pickle.PickleBuffer):
        buf = buf.raw()
    t = torch.frombuffer(buf, dtype=dtype, count=numel)
    return t.reshape(shape)


class _TensorPickler(pickle.Pickler):
    def __init__(self, file, buffer_callback):
        super().__init__(file, protocol=5, buffer_callback=buffer_callback)

    def reducer_override(self, obj):
        if isinstance(obj, torch.Tensor):
            t = obj.detach()
            if t.device.type != "cpu":
                t = t.cpu()
            if not t.is_contiguous():
                t = t.contiguous()
            if t.numel() == 0:
                return _rebuild_tensor, (t.dtype, tuple(t.shape), b"")
            flat = t.reshape(-1).view(torch.uint8)
            pb = pickle.PickleBuffer(flat.numpy())
            return _rebuild_tensor, (t.dtype, tuple(t.shape), pb)
        return NotImplemented


def dumps(obj: Any) -> Tuple[bytes, List[memoryview]]:
    """Serialize to (header, out-of-band buffers)."""
    out: List[memoryview] = []

    def collect(pb: pickle.PickleBuffer):
        out.append(pb.raw())

    bio = io.BytesIO()
    _TensorPickler(bio, collect).dump(obj)
    return bio.getvalue(), out


def loads(header: bytes, buffers: Sequence[Any]) -> Any:
    """Deserialize from a header plus the raw buffers read off the wire."""
    return pickle.loads(header, buffers=buffers)
