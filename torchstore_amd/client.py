"""LocalClient — data-plane orchestration in the user's process.

Responsibilities (reference: torchstore ``client.py``):
* build Requests from user values (DTensor → shard + TensorSlice);
* pick the target volume per batch via the placement strategy;
* expand a get into per-volume slice sub-requests (intersection of stored
  vs requested shards), deduplicating replicated shards;
* run all involved volumes' transports concurrently (asyncio.gather);
* land bytes in-place when a destination tensor was provided, else
  assemble fetched parts into the requested region;
* keep the controller's index in sync (notify after put, before delete).
"""

from __future__ import annotations

import asyncio
from dataclasses import dataclass
from typing import Any, Dict, List, Optional, Sequence, Tuple

import torch

from torchstore_amd.controller import ObjectType, StorageInfo, VolumeInfo
from torchstore_amd.ops.slicing import assemble, region_view, same_memory
from torchstore_amd.runtime import ActorHandle
from torchstore_amd.storage import OBJ_SENTINEL, TensorMeta
from torchstore_amd.strategy import (
    PlacementStrategy,
    StorageVolumeRef,
    strategy_from_spec,
)
from torchstore_amd.transport import TransportContext, create_transport
from torchstore_amd.types import Request, TensorSlice
from torchstore_amd.utils.logging import LatencyTracker, get_logger

logger = get_logger("torchstore_amd.client")


@dataclass
class _SubFetch:
    """One per-volume piece of a get."""

    key: str
    request: Request
    # global region this piece covers (None for objects / whole tensors)
    region_offsets: Optional[Tuple[int, ...]] = None
    result: Any = None


@dataclass
class _CachedFetchPlan:
    """A reusable get_batch plan for steady-state sync loops.

    Valid while (a) the caller passes the SAME destination objects (checked
    by identity against held strong references) and (b) the controller
    confirms every key's layout fingerprint is unchanged — one tiny RPC
    instead of locate + slice planning (~2 ms/step for a Llama-8B dict).
    """

    likes: Dict[str, Any]
    fingerprints: Dict[str, str]
    plans: Dict[str, List[Tuple[str, "_SubFetch"]]]
    by_volume: Dict[str, List["_SubFetch"]]
    busy: bool = False  # a concurrent identical call must not share subs


def _plan_row_split(
    shape: Tuple[int, ...], elem_size: int, limit: int = 1 << 31,
    piece_bytes: int = 1 << 30,
) -> Optional[Tuple[int, int]]:
    """(rows_per_piece, n_pieces) for splitting an oversized tensor along
    dim 0, or None when splitting is impossible/unneeded (pure math —
    unit-tested on CPU)."""
    if not shape:
        return None
    numel = 1
    for d in shape:
        numel *= d
    nbytes = numel * elem_size
    if nbytes < limit:
        return None
    rows = shape[0]
    row_bytes = nbytes // max(rows, 1)
    if rows < 2 or row_bytes == 0 or row_bytes >= limit:
        return None
    rows_per_piece = max(1, piece_bytes // row_bytes)
    k = (rows + rows_per_piece - 1) // rows_per_piece
    if k < 2:
        return None
    return rows_per_piece, k


def _split_huge_tensors(
    requests: List[Request], ref: StorageVolumeRef
) -> List[Request]:
    """Split GPU-resident plain tensors whose allocator block is too large
    to IPC-export (>=2 GiB, the platform dmabuf-import limit) into
    row-range shards with synthetic TensorSlices.

    The pieces are ordinary shards: they store, commit (all coordinates
    land in one notify batch), reshard and reassemble through the existing
    machinery — but each piece moves over the DIRECT one-sided push/pull
    paths instead of the double-staged windowed protocol (measured ~870
    GB/s windowed vs ~2.3 TB/s direct at these sizes).  Only plain
    tensors split; user DTensor shards keep their real coordinates and
    use the windowed fallback when oversized.
    """
    from torchstore_amd.transport.hip_ipc import IPC_BLOCK_LIMIT

    if not str(ref.device).startswith("cuda"):
        return requests
    out: List[Request] = []
    for r in requests:
        t = r.tensor_val
        if (
            r.is_object
            or t is None
            or r.tensor_slice is not None
            or t.device.type != "cuda"
            or t.dim() == 0
            or t.numel() * t.element_size() < IPC_BLOCK_LIMIT
        ):
            out.append(r)
            continue
        tc = t.contiguous()
        plan = _plan_row_split(
            tuple(tc.shape), tc.element_size(), IPC_BLOCK_LIMIT
        )
        if plan is None:
            out.append(r)  # single-row giants keep the windowed path
            continue
        rows_per_piece, k = plan
        rows = tc.shape[0]
        gshape = tuple(tc.shape)
        for j in range(k):
            r0 = j * rows_per_piece
            nr = min(rows_per_piece, rows - r0)
            out.append(
                Request(
                    key=r.key,
                    tensor_val=tc[r0 : r0 + nr],
                    tensor_slice=TensorSlice(
                        offsets=(r0,) + (0,) * (tc.dim() - 1),
                        local_shape=(nr,) + gshape[1:],
                        global_shape=gshape,
                        coordinates=(j,),
                        mesh_shape=(k,),
                    ),
                    inplace=r.inplace,
                )
            )
    return out


def _full_region_slice(global_shape: Sequence[int]) -> TensorSlice:
    shape = tuple(global_shape)
    return TensorSlice(
        offsets=(0,) * len(shape),
        local_shape=shape,
        global_shape=shape,
        coordinates=(),
        mesh_shape=(),
    )


class LocalClient:
    def __init__(
        self,
        controller: ActorHandle,
        strategy: Optional[PlacementStrategy] = None,
    ):
        self._controller = controller
        self._strategy = strategy
        self._ctx = TransportContext()
        self._volumes: Optional[Dict[str, VolumeInfo]] = None
        self._plan_cache: Dict[Tuple[str, ...], _CachedFetchPlan] = {}

    # -- bring-up ---------------------------------------------------------
    async def _ensure_volumes(self) -> Dict[str, VolumeInfo]:
        if self._volumes is None:
            infos, spec = await self._controller.get_volumes.call_one()
            self._volumes = {v.volume_id: v for v in infos}
            if self._strategy is None:
                self._strategy = strategy_from_spec(spec)
        return self._volumes

    def _volume_ref(self, volume_id: str) -> StorageVolumeRef:
        v = self._volumes[volume_id]
        return StorageVolumeRef(
            volume=v.handle,
            volume_id=v.volume_id,
            hostname=v.hostname,
            device=v.device,
            transport_context=self._ctx,
            transport_type=self._strategy.transport if self._strategy else None,
        )

    # -- put --------------------------------------------------------------
    async def put(self, key: str, value: Any) -> None:
        await self.put_batch({key: value})

    async def put_batch(self, items: Dict[str, Any]) -> None:
        if not items:
            return
        volumes = await self._ensure_volumes()
        requests = [Request.from_any(k, v) for k, v in items.items()]
        volume_id = self._strategy.select_volume_id(list(volumes.keys()))
        ref = self._volume_ref(volume_id)
        requests = _split_huge_tensors(requests, ref)
        tracker = LatencyTracker(f"put_batch[{len(requests)}]")
        from torchstore_amd.utils.logging import roctx_range

        with roctx_range(f"ts::put_batch[{len(requests)}]"):
            buffer = create_transport(ref)
            await buffer.put(requests)
            tracker.step("transport", sum(r.nbytes() for r in requests))
            await self._controller.notify_put_batch.call_one(
                [r.meta_only() for r in requests], volume_id
            )
            tracker.step("notify")

    # -- get --------------------------------------------------------------
    async def get(self, key: str, like: Any = None) -> Any:
        result = await self.get_batch({key: like})
        return result[key]

    async def get_batch(self, fetches: Dict[str, Any]) -> Dict[str, Any]:
        if not fetches:
            return {}
        tracker = LatencyTracker(f"get_batch[{len(fetches)}]")
        await self._ensure_volumes()
        cache_key = tuple(sorted(fetches.keys()))
        cached = self._plan_cache.get(cache_key)
        if cached is not None and not cached.busy:
            if all(cached.likes[k] is fetches[k] for k in fetches):
                ok = await self._controller.verify_layouts.call_one(
                    cached.fingerprints
                )
                if ok:
                    tracker.step("verify")
                    cached.busy = True
                    try:
                        return await self._run_plan(
                            fetches, cached.plans, cached.by_volume, tracker
                        )
                    finally:
                        cached.busy = False
            del self._plan_cache[cache_key]
        from torchstore_amd.controller import layout_fingerprint

        located = await self._controller.locate.call_one(list(fetches.keys()))
        tracker.step("locate")
        plans = {
            key: await self._plan_fetch(key, like, located[key])
            for key, like in fetches.items()
        }
        tracker.step("plan")
        # group sub-fetches per volume, fetch all volumes concurrently
        by_volume: Dict[str, List[_SubFetch]] = {}
        for subs in plans.values():
            for volume_id, sf in subs:
                by_volume.setdefault(volume_id, []).append(sf)
        # plans are reusable only when every destination came from the
        # caller (is not None): plan-owned dests would be returned shared
        # across calls otherwise
        if all(like is not None for like in fetches.values()):
            if len(self._plan_cache) >= 64:  # bound stale plan growth
                self._plan_cache.pop(next(iter(self._plan_cache)))
            self._plan_cache[cache_key] = _CachedFetchPlan(
                likes=dict(fetches),
                fingerprints={
                    key: layout_fingerprint(located[key]) for key in fetches
                },
                plans=plans,
                by_volume=by_volume,
            )
        return await self._run_plan(fetches, plans, by_volume, tracker)

    async def _run_plan(
        self,
        fetches: Dict[str, Any],
        plans: Dict[str, List[Tuple[str, _SubFetch]]],
        by_volume: Dict[str, List[_SubFetch]],
        tracker: LatencyTracker,
    ) -> Dict[str, Any]:
        for subs in by_volume.values():
            for sf in subs:
                sf.result = None
        await asyncio.gather(
            *(self._fetch_volume(vid, sfs) for vid, sfs in by_volume.items())
        )
        tracker.step("fetch")
        out = {
            key: self._finish_fetch(key, fetches[key], [sf for _, sf in subs])
            for key, subs in plans.items()
        }
        tracker.step("finish")
        return out

    async def _plan_fetch(
        self, key: str, like: Any, locations: Dict[str, StorageInfo]
    ) -> List[Tuple[str, _SubFetch]]:
        """Decide which volumes to hit and what to ask each for."""
        # the NEWEST write decides both the key's kind and — for whole
        # tensors/objects — which volume's copy to serve: a re-put routed
        # to a different volume (strategy/client change) leaves the old
        # copy behind, and locality order must never resurrect it
        newest_vid = max(locations, key=lambda v: locations[v].seq)
        newest = locations[newest_vid]
        volume_ids = self._order_by_locality(locations.keys())

        if newest.object_type == ObjectType.OBJECT:
            return [
                (newest_vid, _SubFetch(key, Request(key=key, is_object=True)))
            ]

        dest, wanted = self._dest_and_region(like)

        if newest.object_type == ObjectType.TENSOR:
            # the newest whole-tensor copy — single-volume fetch
            vid = newest_vid
            req = Request(key=key, tensor_slice=wanted, tensor_val=dest)
            if dest is not None:
                req.inplace = True
            sf = _SubFetch(
                key, req,
                region_offsets=wanted.offsets if wanted is not None else None,
            )
            await self._allocate_dests(vid, [sf])
            return [(vid, sf)]

        # sharded key: expand into per-stored-shard intersections
        if wanted is None:
            # full-tensor fetch of a sharded key: region = whole global
            # shape (from the newest entry — a stale whole-tensor location
            # may coexist with no slices at all)
            any_slice = next(iter(newest.tensor_slices))
            wanted = _full_region_slice(any_slice.global_shape)

        subs: List[Tuple[str, _SubFetch]] = []
        covered: set = set()
        for vid in volume_ids:
            for stored in locations[vid].tensor_slices:
                inter = stored.intersect(wanted)
                if inter is None:
                    continue
                region = (inter.offsets, inter.local_shape)
                if region in covered:
                    continue  # replicated shard — fetch once
                covered.add(region)
                req = Request(key=key, tensor_slice=inter)
                if dest is not None:
                    view = region_view(
                        dest, wanted.offsets, inter.offsets, inter.local_shape
                    )
                    req.tensor_val = view
                    req.inplace = True
                subs.append(
                    (vid, _SubFetch(key, req, region_offsets=inter.offsets))
                )
        if not subs:
            raise KeyError(
                f"no stored shard of {key!r} overlaps the requested region"
            )
        for vid in {v for v, _ in subs}:
            await self._allocate_dests(vid, [sf for v, sf in subs if v == vid])
        return subs

    def _order_by_locality(self, volume_ids) -> List[str]:
        def sort_key(vid: str):
            return (0 if self._volume_ref(vid).is_local else 1, vid)

        return sorted(volume_ids, key=sort_key)

    def _dest_and_region(
        self, like: Any
    ) -> Tuple[Optional[torch.Tensor], Optional[TensorSlice]]:
        """Destination local tensor + wanted global region from a `like`."""
        if like is None:
            return None, None
        from torchstore_amd.types import LocalShard

        if isinstance(like, LocalShard):
            return like.tensor, like.slice
        from torch.distributed.tensor import DTensor

        if isinstance(like, DTensor):
            from torchstore_amd.types import (
                _dtensor_is_trivially_local,
                slice_from_dtensor,
            )

            if _dtensor_is_trivially_local(like):
                local = like.to_local()
                return local, _full_region_slice(local.shape)
            return like.to_local(), slice_from_dtensor(like)
        if isinstance(like, torch.Tensor):
            return like, _full_region_slice(like.shape)
        raise TypeError(f"cannot fetch into a {type(like)}")

    async def _allocate_dests(self, volume_id: str, subs: List[_SubFetch]) -> None:
        """Every tensor sub-request needs a destination before transport.get."""
        missing = [
            sf for sf in subs
            if not sf.request.is_object and sf.request.tensor_val is None
        ]
        if not missing:
            return
        ref = self._volume_ref(volume_id)
        metas = await ref.volume.get_meta.call_one(
            [sf.request.meta_only() for sf in missing]
        )
        for sf, meta in zip(missing, metas):
            if meta == OBJ_SENTINEL:
                sf.request.is_object = True
                continue
            device = self._alloc_device(meta)
            sf.request.tensor_val = torch.empty(
                meta.shape, dtype=meta.dtype, device=device
            )
            sf.request.dest_owned = True

    @staticmethod
    def _alloc_device(meta: TensorMeta) -> torch.device:
        if meta.device == "cuda" and torch.cuda.is_available():
            return torch.device("cuda", torch.cuda.current_device())
        return torch.device("cpu")

    async def _fetch_volume(self, volume_id: str, subs: List[_SubFetch]) -> None:
        from torchstore_amd.utils.logging import roctx_range

        ref = self._volume_ref(volume_id)
        with roctx_range(f"ts::get[v{volume_id}:{len(subs)}]"):
            buffer = create_transport(ref)
            results = await buffer.get([sf.request for sf in subs])
        for sf, res in zip(subs, results):
            sf.result = res

    def _finish_fetch(self, key: str, like: Any, subs: List[_SubFetch]) -> Any:
        if len(subs) == 1 and subs[0].region_offsets is None:
            sf = subs[0]
            return self._return_like(like, sf.result)
        dest, wanted = self._dest_and_region(like)
        if dest is not None:
            # verify every piece landed inside dest; copy any that didn't
            for sf in subs:
                if isinstance(sf.result, torch.Tensor) and not same_memory(
                    dest, sf.result
                ):
                    view = region_view(
                        dest, wanted.offsets, sf.region_offsets,
                        tuple(sf.result.shape),
                    )
                    view.copy_(sf.result)
            return like if like is not None else dest
        if len(subs) == 1 and isinstance(subs[0].result, torch.Tensor):
            # one piece covering the whole requested region (the commit gate
            # guarantees full coverage) — no assembly copy needed
            return subs[0].result
        parts = [
            (sf.region_offsets, sf.result)
            for sf in subs
            if isinstance(sf.result, torch.Tensor)
        ]
        out, _origin = assemble(parts)
        return out

    def _return_like(self, like: Any, result: Any) -> Any:
        if like is None:
            return result
        if isinstance(result, torch.Tensor):
            dest, _ = self._dest_and_region(like)
            if dest is not None and not same_memory(dest, result):
                dest.copy_(result)
            return like
        return result

    # -- delete / keys / exists ------------------------------------------
    async def delete(self, key: str, missing_ok: bool = False) -> None:
        await self._ensure_volumes()
        volume_ids = await self._controller.notify_delete.call_one(
            key, missing_ok
        )
        await asyncio.gather(
            *(
                self._volumes[vid].handle.delete.call_one(key, True)
                for vid in volume_ids
            )
        )
        self._ctx.drop_key(key)

    async def delete_batch(self, keys: Sequence[str], missing_ok: bool = True) -> None:
        await self._ensure_volumes()
        removed = await self._controller.notify_delete_batch.call_one(
            list(keys), missing_ok
        )
        by_volume: Dict[str, List[str]] = {}
        for key, vids in removed.items():
            for vid in vids:
                by_volume.setdefault(vid, []).append(key)
        await asyncio.gather(
            *(
                self._volumes[vid].handle.delete_batch.call_one(ks, True)
                for vid, ks in by_volume.items()
            )
        )
        for key in keys:
            self._ctx.drop_key(key)

    async def keys(self, prefix: Optional[str] = None) -> List[str]:
        return await self._controller.list_keys.call_one(prefix)

    async def exists(self, key: str) -> bool:
        return await self._controller.key_exists.call_one(key)

    def close(self) -> None:
        self._plan_cache.clear()
        self._ctx.close()
