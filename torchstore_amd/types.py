"""Wire/planning types: :class:`TensorSlice` and :class:`Request`.

These mirror the semantics of the reference store's planning types
(torchstore ``transport/types.py:20-218``): a ``TensorSlice`` pins a shard
into a global tensor + device mesh; a ``Request`` is the unit of work a
client sends through a transport to a storage volume.

Design notes (MI355X build):
* ``TensorSlice`` is a frozen dataclass, hashable, and carries everything the
  controller's commit gate and the reshard planner need — no DTensor object
  ever crosses a process boundary.
* Fully-local DTensors (1-element mesh, or all-``Replicate`` placements) are
  demoted to plain tensors so MoE per-expert replicas and single-rank runs
  store cheaply (reference: ``types.py:58-85``).
"""

from __future__ import annotations

import itertools
from dataclasses import dataclass, replace
from typing import Any, Optional, Tuple

import torch

from torchstore_amd.ops.slicing import overlap


@dataclass(frozen=True)
class TensorSlice:
    """Placement of one shard inside a global logical tensor."""

    offsets: Tuple[int, ...]       # global coords of this shard's [0,..,0]
    local_shape: Tuple[int, ...]   # shape of the shard itself
    global_shape: Tuple[int, ...]  # shape of the full logical tensor
    coordinates: Tuple[int, ...]   # mesh coordinate that produced the shard
    mesh_shape: Tuple[int, ...]    # shape of the device mesh

    def __post_init__(self):
        object.__setattr__(self, "offsets", tuple(int(x) for x in self.offsets))
        object.__setattr__(self, "local_shape", tuple(int(x) for x in self.local_shape))
        object.__setattr__(self, "global_shape", tuple(int(x) for x in self.global_shape))
        object.__setattr__(self, "coordinates", tuple(int(x) for x in self.coordinates))
        object.__setattr__(self, "mesh_shape", tuple(int(x) for x in self.mesh_shape))

    @property
    def ndim(self) -> int:
        return len(self.global_shape)

    def numel(self) -> int:
        n = 1
        for s in self.local_shape:
            n *= s
        return n

    def intersect(self, other: "TensorSlice") -> Optional["TensorSlice"]:
        """The sub-region covered by both shards, or None when disjoint.

        The result keeps *this* slice's global/mesh info with the overlap's
        offsets/shape — i.e. it names a region, not a mesh coordinate.
        """
        hit = overlap(self.offsets, self.local_shape, other.offsets, other.local_shape)
        if hit is None:
            return None
        off, shape = hit
        return replace(self, offsets=off, local_shape=shape)

    @staticmethod
    def expected_coordinates(mesh_shape: Tuple[int, ...]):
        """Every mesh coordinate a fully-committed DTensor must have stored."""
        return itertools.product(*(range(s) for s in mesh_shape))


@dataclass
class LocalShard:
    """A shard + its placement, without requiring a live DTensor/mesh.

    Lets processes outside any torch.distributed world (e.g. a serving
    fleet pulling weights) name sharded layouts explicitly.
    """

    tensor: "torch.Tensor"
    slice: TensorSlice


def _dtensor_is_trivially_local(value) -> bool:
    """A DTensor whose local tensor IS the full tensor on every rank."""
    from torch.distributed.tensor.placement_types import Replicate

    mesh = value.device_mesh
    if mesh.size() == 1:
        return True
    return all(isinstance(p, Replicate) for p in value.placements)


def slice_from_dtensor(value) -> TensorSlice:
    """Compute the TensorSlice for a DTensor's local shard on this rank."""
    from torch.distributed.tensor._utils import (
        compute_local_shape_and_global_offset,
    )

    mesh = value.device_mesh
    local_shape, global_offset = compute_local_shape_and_global_offset(
        value.shape, mesh, value.placements
    )
    coords = mesh.get_coordinate()
    if coords is None:
        raise RuntimeError("this rank is not part of the DTensor's mesh")
    return TensorSlice(
        offsets=tuple(global_offset),
        local_shape=tuple(local_shape),
        global_shape=tuple(value.shape),
        coordinates=tuple(coords),
        mesh_shape=tuple(mesh.shape),
    )


@dataclass
class Request:
    """One key's worth of work travelling client→volume (or back).

    Exactly one of the payload forms is populated:
      * ``tensor_val`` + optional ``tensor_slice``  — tensor / DTensor shard
      * ``objects`` with ``is_object=True``         — arbitrary pickled object
      * none of the above                           — pure metadata (e.g. the
        controller's copy, or a fetch with unknown shape)
    """

    key: str
    tensor_val: Optional[torch.Tensor] = None
    tensor_slice: Optional[TensorSlice] = None
    objects: Any = None
    is_object: bool = False
    # hint: land the fetch directly in tensor_val (in-place get)
    inplace: bool = False
    # tensor_val was allocated by the planner (not user memory) — transports
    # may substitute a zero-copy result for it
    dest_owned: bool = False

    @classmethod
    def from_any(cls, key: str, value: Any) -> "Request":
        from torch.distributed.tensor import DTensor

        if isinstance(value, DTensor):
            if _dtensor_is_trivially_local(value):
                return cls(key=key, tensor_val=value.to_local())
            return cls(
                key=key,
                tensor_val=value.to_local(),
                tensor_slice=slice_from_dtensor(value),
            )
        if isinstance(value, LocalShard):
            return cls(key=key, tensor_val=value.tensor, tensor_slice=value.slice)
        if isinstance(value, torch.Tensor):
            return cls(key=key, tensor_val=value)
        if value is None:
            return cls(key=key)
        return cls(key=key, objects=value, is_object=True)

    @classmethod
    def fetch(cls, key: str, like: Any = None) -> "Request":
        """Build the request side of a get: `like` describes the desired layout."""
        req = cls.from_any(key, like)
        if req.tensor_val is not None:
            req.inplace = True
        return req

    def meta_only(self) -> "Request":
        """Copy with the bulk payload stripped (what the controller sees)."""
        return Request(
            key=self.key,
            tensor_val=None,
            tensor_slice=self.tensor_slice,
            objects=None,
            is_object=self.is_object,
        )

    @property
    def has_payload(self) -> bool:
        return self.tensor_val is not None or self.objects is not None

    def nbytes(self) -> int:
        if self.tensor_val is not None:
            return self.tensor_val.numel() * self.tensor_val.element_size()
        return 0
