"""Loader + python surface for the native HIP extension (`_hipstore`).

The extension is built in-tree for gfx950 (``python setup.py build_ext
--inplace``) and provides:

* K1 slice-gather / K2 batched multi-slice-scatter / K3 fused cast kernels;
* HIP IPC: export/open ``hipIpcMemHandle_t``, batched peer copies on
  dedicated streams, event completion;
* host page pinning (``hipHostRegister``) for the SHM transport.

Policy: on a machine **with** a GPU the extension is required — ops raise
instead of silently falling back to eager torch (the bench must never
measure a python fallback).  On CPU-only machines everything falls back to
torch so tests run anywhere.
"""

from __future__ import annotations

import os
from typing import Optional

import torch

_ext = None
_ext_err: Optional[str] = None


def _load():
    global _ext, _ext_err
    if _ext is not None or _ext_err is not None:
        return _ext
    try:
        from torchstore_amd import _hipstore  # built in-tree

        _ext = _hipstore
    except ImportError as exc:
        _ext_err = str(exc)
        if torch.cuda.is_available() and os.environ.get(
            "TORCHSTORE_AMD_ALLOW_NO_EXT", "0"
        ) != "1":
            raise RuntimeError(
                "GPU present but the _hipstore HIP extension is not built: "
                f"{exc}\nBuild it with `python setup.py build_ext --inplace` "
                "(PYTORCH_ROCM_ARCH=gfx950). Set TORCHSTORE_AMD_ALLOW_NO_EXT=1 "
                "to force the (slow, non-benchmarkable) eager fallback."
            ) from exc
        _ext = None
    return _ext


def extension_available() -> bool:
    try:
        return _load() is not None
    except RuntimeError:
        return True  # GPU + unbuilt extension: selectable, will raise loudly on use


def ext():
    e = _load()
    if e is None:
        raise RuntimeError(f"_hipstore extension unavailable: {_ext_err}")
    return e


# ---------------------------------------------------------------------------
# typed wrappers (tensor → raw pointer marshalling)
# ---------------------------------------------------------------------------

_DTYPE_CODES = {torch.float32: 0, torch.float16: 1, torch.bfloat16: 2}


def _stream(device: torch.device) -> int:
    return torch.cuda.current_stream(device).cuda_stream


def cast_copy(src: torch.Tensor, out: torch.Tensor) -> None:
    """K3: fused cast+pack on the caller's current stream (async)."""
    if src.numel() != out.numel():
        raise ValueError("cast_copy numel mismatch")
    if not (src.is_contiguous() and out.is_contiguous()):
        raise ValueError("cast_copy requires contiguous tensors")
    if src.dtype == out.dtype:
        out.copy_(src)
        return
    sc = _DTYPE_CODES.get(src.dtype)
    dc = _DTYPE_CODES.get(out.dtype)
    if (
        sc is None
        or dc is None
        or src.data_ptr() % 16
        or out.data_ptr() % 16
    ):
        # unusual dtype pair / misaligned view: defer to torch
        # (K3 covers the hot float paths on allocator-aligned tensors)
        out.copy_(src)
        return
    ext().cast_copy(
        src.data_ptr(), sc, out.data_ptr(), dc,
        src.numel(), src.device.index, _stream(src.device),
    )


def _slice_desc(src: torch.Tensor, dst: torch.Tensor):
    """Build a copy_slices descriptor; None when the pair doesn't qualify."""
    if src.shape != dst.shape or src.dtype != dst.dtype:
        return None
    if src.numel() == 0:
        return ()
    es = src.element_size()
    if src.dim() == 0:
        return (src.data_ptr(), dst.data_ptr(), es, [], [], [])
    if (
        src.is_contiguous()
        and dst.is_contiguous()
        and src.numel() * es < (1 << 32)
    ):
        return (src.data_ptr(), dst.data_ptr(), src.numel() * es, [], [], [])
    if src.stride(-1) != 1 or dst.stride(-1) != 1:
        return None
    row_bytes = src.shape[-1] * es
    outer = list(src.shape[:-1])
    sst = [s * es for s in src.stride()[:-1]]
    dstst = [s * es for s in dst.stride()[:-1]]
    return (src.data_ptr(), dst.data_ptr(), row_bytes, outer, sst, dstst)


# large 2-D copies route to SDMA pitched engines (hipMemcpy2D moves a
# strided pack at ~2.7 TB/s payload vs ~1.5 for the kernel); below this the
# ~20us per-enqueue host cost makes the single batched kernel launch win
_SDMA_2D_MIN_BYTES = 32 << 20


def _pitched_params(t: torch.Tensor):
    """(pitch_bytes, width_bytes, height) for a <=2-D unit-inner-stride view."""
    es = t.element_size()
    if t.dim() == 2 and t.stride(1) == 1:
        return (t.stride(0) * es, t.shape[1] * es, t.shape[0])
    if t.dim() == 1 and t.stride(0) == 1:
        return (t.numel() * es, t.numel() * es, 1)
    return None


def copy_pairs(
    pairs, device: torch.device, blocking: bool = True
) -> None:
    """K1/K2: batched strided copies ``[(src_view, dst_view), ...]``.

    All tensors must be on ``device``.  Large 2-D pairs go to the SDMA
    pitched engines; the rest batch into one kernel launch; pairs neither
    can express fall back to ``copy_``.
    """
    descs = []
    sdma2d = []
    for src, dst in pairs:
        nbytes = src.numel() * src.element_size()
        if nbytes >= _SDMA_2D_MIN_BYTES:
            sp = _pitched_params(src)
            dp = _pitched_params(dst)
            if sp and dp and sp[1] == dp[1] and sp[2] == dp[2]:
                sdma2d.append(
                    (dst.data_ptr(), device.index, dp[0],
                     src.data_ptr(), device.index, sp[0], sp[1], sp[2])
                )
                continue
        d = _slice_desc(src, dst)
        if d is None:
            dst.copy_(src)
        elif d != ():
            descs.append(d)
    if descs:
        ext().copy_slices(descs, device.index, _stream(device), blocking)
    if sdma2d:
        ext().copy_batch_2d(sdma2d)  # synchronizes internally


def _desc_view_to_ptr(src: torch.Tensor, dst_ptr: int):
    """Descriptor: strided src view → contiguous bytes at a raw pointer.

    Fuses K1 (gather) with the transfer itself: the kernel reads the
    stored view and writes straight into the (same-device) peer-mapped
    destination — no packed intermediate, half the HBM traffic.
    """
    if src.numel() == 0:
        return ()
    es = src.element_size()
    nbytes = src.numel() * es
    if src.dim() == 0:
        return (src.data_ptr(), dst_ptr, es, [], [], [])
    if src.is_contiguous():
        # flat descriptor: full 16 KiB tiles, no per-row offset math
        return (src.data_ptr(), dst_ptr, nbytes, [], [], [])
    if src.stride(-1) != 1:
        return None
    row_bytes = src.shape[-1] * es
    outer = list(src.shape[:-1])
    sst = [s * es for s in src.stride()[:-1]]
    # contiguous destination strides for src.shape
    dstst = []
    acc = row_bytes
    for d in reversed(outer):
        dstst.append(acc)
        acc *= d
    dstst = list(reversed(dstst))
    return (src.data_ptr(), dst_ptr, row_bytes, outer, sst, dstst)


def copy_views_to_ptrs(items, device: torch.device, blocking: bool = True):
    """Batched fused gather+write: ``[(src_view, dst_ptr), ...]``.

    Large strided 2-D sources use SDMA pitched copies; the rest batch into
    one kernel launch.  Returns the items neither could express (caller
    falls back to pack+copy for those).
    """
    descs = []
    sdma2d = []
    rejects = []
    for src, dst_ptr in items:
        nbytes = src.numel() * src.element_size()
        if nbytes >= _SDMA_2D_MIN_BYTES and not src.is_contiguous():
            sp = _pitched_params(src)
            if sp:
                sdma2d.append(
                    (dst_ptr, device.index, sp[1],
                     src.data_ptr(), device.index, sp[0], sp[1], sp[2])
                )
                continue
        d = _desc_view_to_ptr(src, dst_ptr)
        if d is None:
            rejects.append((src, dst_ptr))
        elif d != ():
            descs.append(d)
    if descs:
        ext().copy_slices(descs, device.index, _stream(device), blocking)
    if sdma2d:
        ext().copy_batch_2d(sdma2d)  # synchronizes internally
    return rejects


def pack_region(src_view: torch.Tensor) -> torch.Tensor:
    """K1: strided region → freshly-allocated contiguous tensor."""
    if src_view.is_contiguous():
        return src_view
    out = torch.empty(
        src_view.shape, dtype=src_view.dtype, device=src_view.device
    )
    copy_pairs([(src_view, out)], src_view.device, blocking=False)
    return out


def alloc_generation(device_index: int) -> int:
    """Caching-allocator generation for the export-handle cache.

    ``segment.all.freed`` increments exactly when an allocator block is
    returned to the OS (``empty_cache`` / OOM-retry release) — the only
    event after which a cached ``hipIpcMemHandle_t`` for a re-allocated
    base address would be stale.  Batch call sites compute this once per
    batch (``torch.cuda.memory_stats`` is ~100 µs, too slow per tensor).
    """
    try:
        return int(
            torch.cuda.memory_stats(device_index).get("segment.all.freed", 0)
        )
    except Exception:  # noqa: BLE001 — stats unavailable: disable caching
        return -1


def ipc_export(ptr: int, device_index: int, generation: Optional[int] = None):
    if generation is None:
        generation = alloc_generation(device_index)
    return ext().ipc_export(ptr, device_index, generation)


def ipc_open(handle: bytes, local_device: int, src_device: int) -> int:
    return ext().ipc_open(handle, local_device, src_device)


def ipc_close(base: int, local_device: int) -> None:
    ext().ipc_close(base, local_device)


def copy_batch(copies) -> None:
    ext().copy_batch(copies)
