"""state_dict exchange: flatten → batched put, with a commit marker.

Reference semantics (torchstore ``state_dict_utils.py``):

* ``put_state_dict`` flattens the nested dict (torch DCP's util), optionally
  casts floating tensors to a transfer dtype, batch-puts every entry under
  ``"{key}/{flat_key}"`` and writes ``"{key}/<MAPPING>"`` **last** — readers
  fetch the mapping first, so a partially-pushed state_dict is invisible
  (the commit-marker protocol);
* ``get_state_dict`` fetches the mapping (missing ⇒ "no matching push"),
  batch-gets all entries (in place into the user's state_dict tensors when
  given — DTensor entries reshard per-entry), and unflattens.

On GPU the dtype cast uses the fused cast+pack HIP kernel (K3) so the
staging copy and the cast are one kernel pass.
"""

from __future__ import annotations

import asyncio
import os
from typing import Any, Dict, List, Optional

import torch

from torchstore_amd.client import LocalClient
from torchstore_amd.utils.logging import LatencyTracker, get_logger

logger = get_logger("torchstore_amd.state_dict")

MAPPING_KEY = "<MAPPING>"

# pipeline factor: big state_dicts move as N concurrent sub-batches so RPC
# framing/parsing overlaps the volume's bulk copies
_PIPELINE = max(1, int(os.environ.get("TORCHSTORE_AMD_SD_PIPELINE", "8")))


async def _get_pipeline(client: LocalClient) -> int:
    """Gets already fan out one RPC per involved volume; extra client-side
    splitting only multiplies per-RPC launch/sync overhead once the volume
    count itself provides the concurrency."""
    volumes = await client._ensure_volumes()
    return max(1, _PIPELINE // max(1, len(volumes)))


def _split(d: Dict[str, Any], n: int) -> List[Dict[str, Any]]:
    if n <= 1 or len(d) <= 8:
        return [d]
    items = list(d.items())
    size = (len(items) + n - 1) // n
    return [dict(items[i : i + size]) for i in range(0, len(items), size)]


def _flatten(state_dict: Dict[str, Any]):
    from torch.distributed.checkpoint._nested_dict import flatten_state_dict

    return flatten_state_dict(state_dict)


def _unflatten(flat: Dict[str, Any], mapping):
    from torch.distributed.checkpoint._nested_dict import unflatten_state_dict

    return unflatten_state_dict(flat, mapping)


def _cast_floating(
    flat: Dict[str, Any], dtype: Optional[torch.dtype]
) -> Dict[str, Any]:
    if dtype is None:
        return flat
    from torchstore_amd.ops.cast import cast_tensor

    out = {}
    for k, v in flat.items():
        if isinstance(v, torch.Tensor) and v.is_floating_point():
            out[k] = cast_tensor(v, dtype)
        else:
            out[k] = v
    return out


def _nbytes(flat: Dict[str, Any]) -> int:
    total = 0
    for v in flat.values():
        if isinstance(v, torch.Tensor):
            total += v.numel() * v.element_size()
    return total


# per-(client, key) direct-sync endpoints, built lazily on first use
_direct_sources: Dict[tuple, Any] = {}
_direct_dests: Dict[tuple, Any] = {}


async def put_state_dict(
    client: LocalClient,
    state_dict: Dict[str, Any],
    key: str,
    transfer_dtype: Optional[torch.dtype] = None,
    direct: bool = False,
    rank: Optional[int] = None,
    world_size: Optional[int] = None,
    direct_rdma: bool = False,  # reference-compat alias for ``direct``
) -> None:
    direct = direct or direct_rdma
    if direct:
        from torchstore_amd.weight_sync import DirectWeightSyncSource

        ck = (id(client), key)
        source = _direct_sources.get(ck)
        if source is None:
            source = DirectWeightSyncSource(
                client, key, transfer_dtype, rank=rank, world_size=world_size
            )
            _direct_sources[ck] = source
        await source.push(state_dict)
        return
    tracker = LatencyTracker(f"put_state_dict[{key}]")
    flat, mapping = _flatten(state_dict)
    tracker.step("flatten")
    flat = _cast_floating(flat, transfer_dtype)
    tracker.step("cast")
    prefixed = {f"{key}/{k}": v for k, v in flat.items()}
    await asyncio.gather(
        *(client.put_batch(chunk) for chunk in _split(prefixed, _PIPELINE))
    )
    tracker.step("put_batch", _nbytes(flat))
    # commit marker: written last, fetched first by readers
    await client.put(f"{key}/{MAPPING_KEY}", mapping)
    tracker.step("commit")
    tracker.e2e(_nbytes(flat))


async def get_state_dict(
    client: LocalClient,
    key: str,
    user_state_dict: Optional[Dict[str, Any]] = None,
    strict: bool = True,
    direct: bool = False,
    direct_rdma: bool = False,  # reference-compat alias for ``direct``
) -> Dict[str, Any]:
    direct = direct or direct_rdma
    if direct:
        from torchstore_amd.weight_sync import DirectWeightSyncDest

        if user_state_dict is None:
            raise ValueError("direct get_state_dict needs a destination state_dict")
        ck = (id(client), key)
        dest = _direct_dests.get(ck)
        if dest is None:
            dest = DirectWeightSyncDest(client, key)
            _direct_dests[ck] = dest
        await dest.pull(user_state_dict)
        return user_state_dict
    tracker = LatencyTracker(f"get_state_dict[{key}]")
    try:
        mapping = await client.get(f"{key}/{MAPPING_KEY}")
    except KeyError as exc:
        raise RuntimeError(
            f"no state_dict was pushed under {key!r} (missing commit marker)"
        ) from exc
    tracker.step("mapping")

    if user_state_dict is not None:
        user_flat, user_mapping = _flatten(user_state_dict)
        if strict:
            # reference asserts MAPPING EQUALITY, both directions
            # (torchstore state_dict_utils.py:146-152): entries the user
            # has but the store doesn't AND entries the store has but the
            # user doesn't are both strict-mode errors
            stored_keys = {str(k) for k in mapping}
            missing = set(user_flat.keys()) - stored_keys
            extra = stored_keys - set(user_flat.keys())
            if missing or extra:
                raise KeyError(
                    f"strict get_state_dict mapping mismatch: "
                    f"user-only entries {sorted(missing)[:5]}, "
                    f"stored-only entries {sorted(extra)[:5]}"
                )
        fetches = {f"{key}/{k}": v for k, v in user_flat.items()}
        chunks = await asyncio.gather(
            *(
                client.get_batch(c)
                for c in _split(fetches, await _get_pipeline(client))
            )
        )
        results = {}
        for c in chunks:
            results.update(c)
        tracker.step("get_batch", _nbytes(user_flat))
        flat = {k: results[f"{key}/{k}"] for k in user_flat}
        out = _unflatten(flat, user_mapping)
        tracker.e2e(_nbytes(user_flat))
        return out

    flat_keys = list(mapping.keys())
    fetches = {f"{key}/{k}": None for k in flat_keys}
    chunks = await asyncio.gather(
        *(
            client.get_batch(c)
            for c in _split(fetches, await _get_pipeline(client))
        )
    )
    results = {}
    for c in chunks:
        results.update(c)
    tracker.step("get_batch")
    flat = {k: results[f"{key}/{k}"] for k in flat_keys}
    out = _unflatten(flat, mapping)
    tracker.e2e(_nbytes(flat))
    return out
