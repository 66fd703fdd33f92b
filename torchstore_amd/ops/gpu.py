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
