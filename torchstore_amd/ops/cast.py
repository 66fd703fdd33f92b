"""Dtype cast for transfer staging (K3).

On a GPU tensor this calls the fused cast+pack CDNA4 kernel from the native
extension (one read + one write of HBM, vectorized 16 B/lane, also packing
strided inputs); on CPU it falls back to ``tensor.to(dtype)``.
Reference call sites: torchstore ``state_dict_utils.py:177-189`` (per-sync
fp32→bf16 of every floating param) and ``direct_weight_sync.py:133``.
"""

from __future__ import annotations

import torch


def cast_tensor(t: torch.Tensor, dtype: torch.dtype) -> torch.Tensor:
    if t.dtype == dtype:
        return t
    if type(t) is not torch.Tensor:
        # tensor subclasses (DTensor etc.) go through their own dispatch
        return t.to(dtype)
    if t.device.type == "cuda":
        from torchstore_amd.ops import gpu

        out = torch.empty(t.shape, dtype=dtype, device=t.device)
        gpu.cast_copy(t.contiguous(), out)
        return out
    return t.to(dtype)


def cast_into(t: torch.Tensor, out: torch.Tensor) -> None:
    """Cast ``t`` into a preallocated ``out`` (staging-buffer refresh path)."""
    if t.device.type == "cuda" and out.device.type == "cuda":
        from torchstore_amd.ops import gpu

        gpu.cast_copy(t.contiguous(), out)
        return
    out.copy_(t)
