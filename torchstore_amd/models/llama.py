"""Synthetic Llama-3-8B state_dict (random init, no network access needed).

Used by the bench + real-model tests: exact parameter shapes of
meta-llama/Meta-Llama-3-8B (~8.03B params, ~16.06 GB bf16), materialized
shard-by-shard so each rank only allocates its own slice.
"""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch

# (name template, shape) — 32 transformer layers
HIDDEN = 4096
INTER = 14336
KV = 1024
VOCAB = 128256
LAYERS = 32


def llama3_8b_shapes(
    layers: int = LAYERS, scale: int = 1
) -> Dict[str, Tuple[int, ...]]:
    """Exact Llama-3-8B shapes; ``scale`` > 1 shrinks every dim (tests)."""
    v, h, kv, it = VOCAB // scale, HIDDEN // scale, KV // scale, INTER // scale
    shapes: Dict[str, Tuple[int, ...]] = {
        "model.embed_tokens.weight": (v, h),
    }
    for i in range(layers):
        p = f"model.layers.{i}."
        shapes[p + "self_attn.q_proj.weight"] = (h, h)
        shapes[p + "self_attn.k_proj.weight"] = (kv, h)
        shapes[p + "self_attn.v_proj.weight"] = (kv, h)
        shapes[p + "self_attn.o_proj.weight"] = (h, h)
        shapes[p + "mlp.gate_proj.weight"] = (it, h)
        shapes[p + "mlp.up_proj.weight"] = (it, h)
        shapes[p + "mlp.down_proj.weight"] = (h, it)
        shapes[p + "input_layernorm.weight"] = (h,)
        shapes[p + "post_attention_layernorm.weight"] = (h,)
    shapes["model.norm.weight"] = (h,)
    shapes["lm_head.weight"] = (v, h)
    return shapes


def total_bytes(
    shapes: Dict[str, Tuple[int, ...]], dtype: torch.dtype = torch.bfloat16
) -> int:
    esize = torch._utils._element_size(dtype)
    return sum(
        esize * int(torch.tensor(s).prod()) if s else esize
        for s in shapes.values()
    )


def fsdp_placement(shape: Tuple[int, ...], world: int) -> Optional[int]:
    """FSDP-style: Shard(0) when divisible, else replicate (None)."""
    if len(shape) >= 1 and shape[0] % world == 0:
        return 0
    return None


def tp_placement(shape: Tuple[int, ...], world: int) -> Optional[int]:
    """TP-style: prefer Shard(1) on 2-D weights, else Shard(0), else replicate."""
    if len(shape) == 2 and shape[1] % world == 0:
        return 1
    if len(shape) >= 1 and shape[0] % world == 0:
        return 0
    return None


_PATTERN_COEFS = (7, 13, 29, 31)


def _pattern_block(t: torch.Tensor, offsets) -> None:
    idx = None
    for d in range(t.dim()):
        ar = (
            torch.arange(
                offsets[d], offsets[d] + t.shape[d],
                device=t.device, dtype=torch.int64,
            )
            * _PATTERN_COEFS[d]
        )
        shape = [1] * t.dim()
        shape[d] = -1
        ar = ar.view(shape)
        idx = ar if idx is None else idx + ar
    vals = ((idx % 61) - 30).to(torch.float32) * 0.01
    t.copy_(vals.to(t.dtype).expand_as(t))


def pattern_fill(t: torch.Tensor, offsets) -> torch.Tensor:
    """Fill a shard with values determined by GLOBAL position.

    Any process can recompute any region independently, so cross-GPU
    reshard paths can be verified without exchanging data:
    ``v[i0,i1,..] = (((Σ coef_d * global_i_d) mod 61) - 30) * 0.01``.

    Computed in row chunks: the int64 index grid of a big shard would
    otherwise allocate a multi-GB transient whose cached allocator block
    later hosts small tensors — pushing them over the ≥2 GiB IPC-mapping
    limit and onto the slow windowed path.
    """
    max_elems = 16 << 20  # ≈128 MB of int64 transients per block
    if t.dim() == 0 or t.numel() <= max_elems:
        _pattern_block(t, tuple(offsets))
        return t
    row_elems = max(1, t[0].numel())
    rows_per = max(1, max_elems // row_elems)
    for r0 in range(0, t.shape[0], rows_per):
        sub = t[r0 : r0 + rows_per]
        _pattern_block(sub, (offsets[0] + r0,) + tuple(offsets[1:]))
    return t


def expected_pattern(shape, offsets, dtype, device) -> torch.Tensor:
    out = torch.empty(shape, dtype=dtype, device=device)
    return pattern_fill(out, offsets)


def make_local_shard_state_dict(
    rank: int,
    world: int,
    shard_fn,
    dtype: torch.dtype = torch.bfloat16,
    device: str = "cuda",
    layers: int = LAYERS,
    zero: bool = True,
    scale: int = 1,
    pattern: bool = False,
):
    """DTensor-free sharded state_dict: {name: LocalShard | tensor}.

    For processes outside any torch.distributed world (serving fleets).
    Shard dims chosen by ``shard_fn`` are always divisible by ``world``
    for these shapes, so the even split matches DTensor chunking exactly.
    """
    from torchstore_amd.types import LocalShard, TensorSlice

    shapes = llama3_8b_shapes(layers, scale)
    out = {}
    for name, shape in shapes.items():
        if world == 1:
            t = torch.empty(shape, dtype=dtype, device=device)
            if pattern:
                pattern_fill(t, (0,) * len(shape))
            elif not zero:
                t.normal_(0, 0.02)
            out[name] = t
            continue
        dim = shard_fn(shape, world)
        if dim is None:
            local_shape = shape
            offsets = (0,) * len(shape)
        else:
            assert shape[dim] % world == 0
            local_shape = list(shape)
            local_shape[dim] = shape[dim] // world
            local_shape = tuple(local_shape)
            offsets = tuple(
                rank * (shape[d] // world) if d == dim else 0
                for d in range(len(shape))
            )
        t = torch.empty(local_shape, dtype=dtype, device=device)
        if pattern:
            pattern_fill(t, offsets)
        elif not zero:
            t.normal_(0, 0.02)
        out[name] = LocalShard(
            tensor=t,
            slice=TensorSlice(
                offsets=offsets, local_shape=local_shape, global_shape=shape,
                coordinates=(rank,), mesh_shape=(world,),
            ),
        )
    return out


def make_sharded_state_dict(
    mesh,
    shard_fn,
    dtype: torch.dtype = torch.bfloat16,
    device: str = "cuda",
    layers: int = LAYERS,
    zero: bool = False,
    seed: Optional[int] = None,
    scale: int = 1,
) -> Dict[str, torch.Tensor]:
    """Build {name: DTensor|tensor} where each rank materializes only its shard.

    ``mesh`` is a 1-D DeviceMesh (or None for single-process: plain tensors).
    ``shard_fn(shape, world) -> dim|None`` picks the shard dim per param.
    """
    from torch.distributed.tensor import DTensor, Replicate, Shard
    from torch.distributed.tensor._utils import (
        compute_local_shape_and_global_offset,
    )

    shapes = llama3_8b_shapes(layers, scale)
    out: Dict[str, torch.Tensor] = {}
    world = mesh.size() if mesh is not None else 1
    gen = None
    if seed is not None:
        gen = torch.Generator(device=device)
        gen.manual_seed(seed)
    for name, shape in shapes.items():
        if mesh is None or world == 1:
            t = torch.empty(shape, dtype=dtype, device=device)
            if not zero:
                t.normal_(0, 0.02, generator=gen)
            out[name] = t
            continue
        dim = shard_fn(shape, world)
        placements = [Replicate() if dim is None else Shard(dim)]
        local_shape, _off = compute_local_shape_and_global_offset(
            shape, mesh, placements
        )
        local = torch.empty(local_shape, dtype=dtype, device=device)
        if not zero:
            local.normal_(0, 0.02, generator=gen)
        out[name] = DTensor.from_local(
            local, mesh, placements, run_check=False,
            shape=torch.Size(shape),
            stride=torch.empty(shape, device="meta").stride(),
        )
    return out
