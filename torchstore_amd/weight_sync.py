"""Direct weight sync: zero-copy one-hop trainer→generator transfer.

MI355X re-design of the reference's RDMA weight sync
(torchstore ``direct_weight_sync.py``): instead of ibverbs RDMABuffers,
the **source** exports HIP IPC handles pointing at live parameter memory
(optimizer updates are visible without copies) or at dtype-cast staging
buffers refreshed per step (K3 fused cast); only the handle *metadata*
goes through the store (``"{key}/rank_{r}"`` + ``"{key}/num_ranks"``).
The **dest** builds a cached transfer plan on first pull — per
(dest param × overlapping source shard):

* exact region match + contiguous dest ⇒ one-sided read straight into
  parameter memory;
* partial overlap (TP mismatch) ⇒ read the full source shard into a cached
  recv buffer, then one batched K2 scatter into the strided dest views;
* replicated source shards deduplicated by region.

All reads of a pull execute as ONE ``copy_batch`` striped over the
per-device stream pool, so pulls from several peers occupy several xGMI
links concurrently (reference behavior: asyncio.gather of RDMA reads,
``direct_weight_sync.py:338-340``).

The memory codec is swappable (:class:`FakeMemoryCodec`) so the full plan
logic runs in CPU tests — mirroring the reference's MockRDMABuffer tests.
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Any, Dict, List, Optional, Sequence, Tuple

import torch

from torchstore_amd.client import LocalClient
from torchstore_amd.ops.slicing import byte_view, region_view
from torchstore_amd.transport.hip_ipc import IpcDescriptor
from torchstore_amd.types import LocalShard, Request, TensorSlice
from torchstore_amd.utils.logging import LatencyTracker, get_logger

logger = get_logger("torchstore_amd.weight_sync")


@dataclass(frozen=True)
class WeightHandle:
    """Serializable pointer to one source rank's shard of one parameter."""

    name: str
    desc: IpcDescriptor
    slice: Optional[TensorSlice]  # None = full tensor


# ---------------------------------------------------------------------------
# memory codec: real HIP IPC, or an in-process fake for CPU tests
# ---------------------------------------------------------------------------


@dataclass
class _TransferOp:
    """One one-sided read of a pull.

    kinds:
      * ``1d``   — whole descriptor straight into a contiguous dest (the
        zero-copy exact-match case: dest is live parameter memory);
      * ``2d``   — pitched read of ONLY the overlap region straight into a
        (possibly strided) dest view — moves overlap bytes, not whole
        shards (improvement over the reference's full-shard reads);
      * ``recv`` — full shard into a cached recv buffer + a K2 scatter
        (fallback for >2-D overlap geometry).
    """

    kind: str
    desc: IpcDescriptor
    dst: torch.Tensor
    src_offset_bytes: int = 0
    spitch: int = 0          # remote row pitch (bytes), 2d only
    width: int = 0           # row bytes, 2d only
    height: int = 0          # rows, 2d only
    scatter: Optional[Tuple[torch.Tensor, torch.Tensor]] = None

    @property
    def nbytes(self) -> int:
        if self.kind == "2d":
            return self.width * self.height
        return self.desc.nbytes


class IpcMemoryCodec:
    def export(self, t: torch.Tensor) -> IpcDescriptor:
        from torchstore_amd.transport.hip_ipc import export_tensor

        return export_tensor(t)

    def read_batch(self, ops: Sequence[_TransferOp], ctx) -> None:
        """Execute every one-sided read of a pull as ONE kernel launch.

        The kernel reads IPC-mapped source memory directly (peer loads
        over xGMI for cross-GPU sources) and writes the — possibly
        strided — destinations: at N=8 this replaces ~2.3k pitched SDMA
        enqueues per rank (~20 µs host cost each) with a single dispatch,
        and units from different peers are in flight concurrently across
        the grid, so several xGMI links are driven by one launch.  A
        system-scope acquire at workgroup start keeps re-pulls coherent
        (remote lines cached by a previous pull).  Set
        ``TORCHSTORE_AMD_DIRECT_SDMA=1`` for the per-op SDMA fallback.
        """
        from torchstore_amd.transport.hip_ipc import IpcOpenCache
        from torchstore_amd.ops import gpu

        if os.environ.get("TORCHSTORE_AMD_DIRECT_SDMA", "0") == "1":
            return self._read_batch_sdma(ops, ctx)
        cache: IpcOpenCache = ctx.cache(IpcOpenCache)
        descs = []
        device = None
        for op in ops:
            device = op.dst.device
            src_ptr = cache.resolve(op.desc, device.index)
            if op.kind == "2d":
                es = op.dst.element_size()
                dpitch = (
                    op.dst.stride(0) * es if op.dst.dim() == 2 else op.width
                )
                src = src_ptr + op.src_offset_bytes
                if op.height == 1:
                    descs.append((src, op.dst.data_ptr(), op.width, [], [], []))
                else:
                    descs.append(
                        (src, op.dst.data_ptr(), op.width,
                         [op.height], [op.spitch], [dpitch])
                    )
            else:
                # flat read; row_bytes is u32 in the kernel — span in <=1 GiB
                nbytes = op.desc.nbytes
                dstp = op.dst.data_ptr()
                off = 0
                while nbytes - off > 0:
                    n = min(nbytes - off, 1 << 30)
                    descs.append((src_ptr + off, dstp + off, n, [], [], []))
                    off += n
        if descs:
            gpu.ext().copy_slices(
                descs, device.index, gpu._stream(device), True, True
            )

    def _read_batch_sdma(self, ops: Sequence[_TransferOp], ctx) -> None:
        """Per-op SDMA path (pitched engines), striped over the stream pool."""
        from torchstore_amd.transport.hip_ipc import IpcOpenCache
        from torchstore_amd.ops import gpu

        cache: IpcOpenCache = ctx.cache(IpcOpenCache)
        copies_1d = []
        copies_2d = []
        for op in ops:
            dev = op.dst.device.index
            src_ptr = cache.resolve(op.desc, dev)
            if op.kind == "2d":
                es = op.dst.element_size()
                dpitch = (
                    op.dst.stride(0) * es if op.dst.dim() == 2 else op.width
                )
                copies_2d.append(
                    (op.dst.data_ptr(), dev, dpitch,
                     src_ptr + op.src_offset_bytes, op.desc.device_index,
                     op.spitch, op.width, op.height)
                )
            else:
                copies_1d.append(
                    (op.dst.data_ptr(), dev, src_ptr, op.desc.device_index,
                     op.desc.nbytes)
                )
        if copies_1d:
            gpu.copy_batch(copies_1d)
        if copies_2d:
            gpu.ext().copy_batch_2d(copies_2d)


class FakeMemoryCodec:
    """In-process stand-in: descriptors index a registry of live tensors."""

    def __init__(self):
        self.registry: Dict[int, torch.Tensor] = {}
        self._next = 0
        self.read_count = 0

    def export(self, t: torch.Tensor) -> IpcDescriptor:
        self._next += 1
        self.registry[self._next] = t
        return IpcDescriptor(
            handle=self._next.to_bytes(8, "little"),
            offset=0,
            nbytes=t.numel() * t.element_size(),
            dtype=t.dtype,
            shape=tuple(t.shape),
            device_index=-1,
        )

    def read_batch(self, ops: Sequence[_TransferOp], ctx) -> None:
        for op in ops:
            src = self.registry[int.from_bytes(op.desc.handle, "little")]
            src_c = src.contiguous()
            if op.kind == "2d":
                es = src_c.element_size()
                assert op.src_offset_bytes % es == 0 and op.spitch % es == 0
                view = src_c.reshape(-1).as_strided(
                    (op.height, op.width // es),
                    (op.spitch // es, 1),
                    storage_offset=op.src_offset_bytes // es,
                )
                op.dst.copy_(view.reshape(op.dst.shape))
            else:
                byte_view(op.dst)[:].copy_(byte_view(src_c))
            self.read_count += 1


_codec: Any = None


def get_codec():
    global _codec
    if _codec is None:
        _codec = IpcMemoryCodec()
    return _codec


def set_codec(codec) -> None:
    global _codec
    _codec = codec


def _flatten(sd):
    from torch.distributed.checkpoint._nested_dict import flatten_state_dict

    return flatten_state_dict(sd)


def _request_slice(value) -> Tuple[torch.Tensor, Optional[TensorSlice]]:
    req = Request.from_any("_", value)
    return req.tensor_val, req.tensor_slice


def _prod(shape) -> int:
    n = 1
    for s in shape:
        n *= s
    return n


def _full_slice(shape) -> TensorSlice:
    shape = tuple(shape)
    return TensorSlice(
        offsets=(0,) * len(shape), local_shape=shape, global_shape=shape,
        coordinates=(), mesh_shape=(),
    )


class DirectWeightSyncSource:
    """Trainer side: export handles once, refresh staging casts per step."""

    def __init__(
        self,
        client: LocalClient,
        key: str,
        transfer_dtype: Optional[torch.dtype] = None,
        rank: Optional[int] = None,
        world_size: Optional[int] = None,
    ):
        self.client = client
        self.key = key
        self.transfer_dtype = transfer_dtype
        self.rank = rank if rank is not None else int(os.environ.get("RANK", "0"))
        self.world_size = (
            world_size
            if world_size is not None
            else int(os.environ.get("WORLD_SIZE", "1"))
        )
        self.registered = False
        # name -> (live param local tensor, staging buffer or None)
        self._params: Dict[str, Tuple[torch.Tensor, Optional[torch.Tensor]]] = {}

    async def push(self, state_dict: Dict[str, Any]) -> None:
        """First call registers + publishes handles; later calls refresh casts."""
        if self.registered:
            self.refresh(state_dict)
            return
        tracker = LatencyTracker(f"direct_sync.register[{self.key}]")
        flat, _mapping = _flatten(state_dict)
        codec = get_codec()
        handles: List[WeightHandle] = []
        for name, value in flat.items():
            if not isinstance(value, (torch.Tensor, LocalShard)):
                continue
            local, tslice = _request_slice(value)
            local = local.detach()
            staging = None
            if (
                self.transfer_dtype is not None
                and local.is_floating_point()
                and local.dtype != self.transfer_dtype
            ):
                from torchstore_amd.ops.cast import cast_tensor

                staging = cast_tensor(local.contiguous(), self.transfer_dtype)
                export_t = staging
            else:
                if not local.is_contiguous():
                    staging = local.contiguous()
                    export_t = staging
                else:
                    export_t = local  # true zero-copy: live param memory
            self._params[name] = (local, staging)
            handles.append(
                WeightHandle(name=name, desc=codec.export(export_t), slice=tslice)
            )
        tracker.step("export", None)
        await self.client.put(f"{self.key}/rank_{self.rank}", handles)
        if self.rank == 0:
            await self.client.put(f"{self.key}/num_ranks", self.world_size)
        self.registered = True
        tracker.e2e()

    def refresh(self, state_dict: Optional[Dict[str, Any]] = None) -> None:
        """Re-cast staging buffers after optimizer.step() (K3 per param)."""
        from torchstore_amd.ops.cast import cast_into

        for name, (live, staging) in self._params.items():
            if staging is not None:
                cast_into(live.contiguous(), staging)
        if torch.cuda.is_available():
            torch.cuda.synchronize()


class DirectWeightSyncDest:
    """Generator side: cached plan, one batched read per pull."""

    def __init__(self, client: LocalClient, key: str):
        self.client = client
        self.key = key
        self._plan: Optional[List[_TransferOp]] = None
        self._handles: Optional[List[WeightHandle]] = None
        # STRONG reference to the dict the cached plan targets: `is` against
        # a held reference can never alias a new dict (id() could, after GC
        # reuses the address — the plan would then sync into orphaned tensors)
        self._plan_dest: Optional[Dict[str, Any]] = None

    async def _fetch_handles(self) -> List[WeightHandle]:
        if self._handles is not None:
            return self._handles
        try:
            num_ranks = await self.client.get(f"{self.key}/num_ranks")
        except KeyError as exc:
            raise RuntimeError(
                f"no direct weight sync source registered under {self.key!r}"
            ) from exc
        all_handles: List[WeightHandle] = []
        fetched = await self.client.get_batch(
            {f"{self.key}/rank_{r}": None for r in range(num_ranks)}
        )
        for r in range(num_ranks):
            all_handles.extend(fetched[f"{self.key}/rank_{r}"])
        self._handles = all_handles
        return all_handles

    def _build_plan(
        self, handles: Sequence[WeightHandle], dest_flat: Dict[str, Any]
    ) -> List[_TransferOp]:
        by_name: Dict[str, List[WeightHandle]] = {}
        for h in handles:
            by_name.setdefault(h.name, []).append(h)
        plan: List[_TransferOp] = []
        for name, value in dest_flat.items():
            if not isinstance(value, (torch.Tensor, LocalShard)):
                continue
            dest_local, dest_slice = _request_slice(value)
            wanted = dest_slice or _full_slice(dest_local.shape)
            srcs = by_name.get(name)
            if not srcs:
                raise KeyError(f"source has no parameter {name!r}")
            covered: set = set()
            covered_regions: List[Tuple[Tuple[int, ...], Tuple[int, ...]]] = []
            for h in srcs:
                src_region = h.slice or _full_slice(h.desc.shape)
                inter = src_region.intersect(wanted)
                if inter is None:
                    continue
                region_key = (inter.offsets, inter.local_shape)
                if region_key in covered:
                    continue  # replicated shard — read once
                covered.add(region_key)
                covered_regions.append(region_key)
                if dest_local.dtype != h.desc.dtype:
                    raise TypeError(
                        f"{name}: dest dtype {dest_local.dtype} != "
                        f"transfer dtype {h.desc.dtype}"
                    )
                exact = (
                    inter.offsets == wanted.offsets
                    and inter.local_shape == wanted.local_shape
                    and inter.offsets == src_region.offsets
                    and inter.local_shape == src_region.local_shape
                )
                if exact and dest_local.is_contiguous():
                    plan.append(_TransferOp(kind="1d", dst=dest_local, desc=h.desc))
                    continue
                dst_view = region_view(
                    dest_local, wanted.offsets, inter.offsets, inter.local_shape
                )
                es = dest_local.element_size()
                shape = inter.local_shape
                rel = tuple(
                    inter.offsets[d] - src_region.offsets[d]
                    for d in range(len(shape))
                )
                if len(shape) <= 2 and (
                    dst_view.dim() == 0 or dst_view.stride(-1) == 1
                ):
                    # pitched read of exactly the overlap bytes
                    width = shape[-1] * es if shape else es
                    height = shape[0] if len(shape) == 2 else 1
                    if len(h.desc.shape) == 2:
                        spitch = h.desc.shape[1] * es
                        src_off = (rel[0] * h.desc.shape[1] + rel[-1]) * es
                    else:
                        spitch = width
                        src_off = rel[0] * es if rel else 0
                    plan.append(
                        _TransferOp(
                            kind="2d", dst=dst_view, desc=h.desc,
                            src_offset_bytes=src_off, spitch=spitch,
                            width=width, height=height,
                        )
                    )
                    continue
                # >2-D geometry: full-shard read + batched K2 scatter
                recv = torch.empty(
                    h.desc.shape, dtype=h.desc.dtype, device=dest_local.device
                )
                src_view = region_view(
                    recv, src_region.offsets, inter.offsets, inter.local_shape
                )
                plan.append(
                    _TransferOp(
                        kind="recv", dst=recv, desc=h.desc,
                        scatter=(src_view, dst_view),
                    )
                )
            # the source shards' intersections must TILE the wanted region;
            # a gap would silently leave stale weights in the generator's
            # parameters (verify, don't trust the layout)
            from torchstore_amd.ops.slicing import (
                regions_disjoint,
                union_volume,
            )

            want_vol = 1
            for s in wanted.local_shape:
                want_vol *= s
            if not covered_regions:
                got_vol = 0
            elif regions_disjoint(covered_regions):
                got_vol = sum(
                    _prod(shape) for _off, shape in covered_regions
                )
            else:
                got_vol = union_volume(covered_regions)
            if got_vol < want_vol:
                raise RuntimeError(
                    f"direct sync plan for {name!r} covers {got_vol} of "
                    f"{want_vol} destination elements — source shards do "
                    "not tile the wanted region"
                )
        return plan

    def invalidate(self) -> None:
        """Drop the cached plan + handles (call when the source re-registers,
        e.g. after a trainer restart)."""
        self._plan = None
        self._handles = None
        self._plan_dest = None

    async def pull(self, dest_state_dict: Dict[str, Any]) -> None:
        codec = get_codec()
        if self._plan_dest is not None and self._plan_dest is not dest_state_dict:
            # a different destination dict: the cached plan points at the old
            # tensors — rebuild against the new ones
            self._plan = None
        if self._plan is None:
            handles = await self._fetch_handles()
            dest_flat, _ = _flatten(dest_state_dict)
            self._plan = self._build_plan(handles, dest_flat)
            self._plan_dest = dest_state_dict
            logger.info(
                "direct sync plan: %d ops (%d zero-copy, %d pitched, %d recv)",
                len(self._plan),
                sum(1 for op in self._plan if op.kind == "1d"),
                sum(1 for op in self._plan if op.kind == "2d"),
                sum(1 for op in self._plan if op.kind == "recv"),
            )
        tracker = LatencyTracker(f"direct_sync.pull[{self.key}]")
        codec.read_batch(self._plan, self.client._ctx)
        nbytes = sum(op.nbytes for op in self._plan)
        tracker.step("read", nbytes)
        scatters = [op.scatter for op in self._plan if op.scatter is not None]
        if scatters:
            device = scatters[0][1].device
            if device.type == "cuda":
                from torchstore_amd.ops import gpu

                gpu.copy_pairs(scatters, device, blocking=True)
            else:
                for src_view, dst_view in scatters:
                    dst_view.copy_(src_view)
        tracker.step("scatter")
        tracker.e2e(nbytes)
