"""N-dimensional shard/slice math — the correctness heart of resharding.

A shard of a global tensor is described by a :class:`~torchstore_amd.types.TensorSlice`
(per-dim global offsets + local shape).  Resharding between two arbitrary
layouts reduces to three primitives implemented here:

* :func:`overlap` — per-dim ``[start, end)`` intersection of two shards
  (reference semantics: torchstore ``utils.py:248-307``).
* :func:`region_view` — map a global region into a (possibly strided) view of a
  destination shard so transports can land bytes in place
  (reference: ``utils.py:36-98``).
* :func:`assemble` — scatter K fetched sub-regions into their bounding-box
  union (reference: ``utils.py:158-245``).

Everything here is pure index math plus torch view ops; the GPU fast paths
(slice gather / batched scatter HIP kernels) live in
:mod:`torchstore_amd.ops.gpu` and fall back to these implementations on CPU.
"""

from __future__ import annotations

from typing import Optional, Sequence, Tuple

import torch

Region = Tuple[Tuple[int, ...], Tuple[int, ...]]  # (offsets, shape)


def overlap(
    a_offsets: Sequence[int],
    a_shape: Sequence[int],
    b_offsets: Sequence[int],
    b_shape: Sequence[int],
) -> Optional[Region]:
    """Intersect two N-d regions given by (global offsets, shape).

    Returns ``(offsets, shape)`` of the intersection in *global* coordinates,
    or ``None`` when the regions are disjoint in any dimension.
    """
    if len(a_offsets) != len(b_offsets):
        raise ValueError(
            f"rank mismatch: {len(a_offsets)} vs {len(b_offsets)}"
        )
    starts = []
    sizes = []
    for ao, asz, bo, bsz in zip(a_offsets, a_shape, b_offsets, b_shape):
        s = max(ao, bo)
        e = min(ao + asz, bo + bsz)
        if e <= s:
            return None
        starts.append(s)
        sizes.append(e - s)
    return tuple(starts), tuple(sizes)


def extract_region(
    tensor: torch.Tensor,
    tensor_offsets: Sequence[int],
    region_offsets: Sequence[int],
    region_shape: Sequence[int],
) -> torch.Tensor:
    """Return the view of ``tensor`` covering a global region.

    ``tensor`` is a shard whose element [0,...,0] sits at ``tensor_offsets``
    in global coordinates; the requested region must lie fully inside it.
    The result is a *view* (no copy) — callers decide whether to materialise.
    """
    out = tensor
    for dim, (toff, roff, rsz) in enumerate(
        zip(tensor_offsets, region_offsets, region_shape)
    ):
        rel = roff - toff
        if rel < 0 or rel + rsz > tensor.shape[dim]:
            raise IndexError(
                f"region dim {dim}: [{roff},{roff + rsz}) outside shard "
                f"[{toff},{toff + tensor.shape[dim]})"
            )
        out = out.narrow(dim, rel, rsz)
    return out


def region_view(
    dest: torch.Tensor,
    dest_offsets: Sequence[int],
    region_offsets: Sequence[int],
    region_shape: Sequence[int],
) -> torch.Tensor:
    """View of ``dest`` where a fetched global region should land (may be strided)."""
    return extract_region(dest, dest_offsets, region_offsets, region_shape)


def is_dense_view(view: torch.Tensor) -> bool:
    """True when ``view`` is a single contiguous byte range (safe for raw I/O)."""
    return view.is_contiguous()


def bounding_box(parts: Sequence[Region]) -> Region:
    """Smallest region covering every part."""
    if not parts:
        raise ValueError("no parts")
    ndim = len(parts[0][0])
    lo = [min(p[0][d] for p in parts) for d in range(ndim)]
    hi = [max(p[0][d] + p[1][d] for p in parts) for d in range(ndim)]
    return tuple(lo), tuple(hi[d] - lo[d] for d in range(ndim))


def regions_disjoint(regions: Sequence[Region]) -> bool:
    """Pairwise-disjointness of N-d regions (O(k²), k small in practice)."""
    for i in range(len(regions)):
        for j in range(i + 1, len(regions)):
            if overlap(regions[i][0], regions[i][1],
                       regions[j][0], regions[j][1]) is not None:
                return False
    return True


def union_volume(regions: Sequence[Region]) -> int:
    """Exact element count of the union of N-d regions.

    Coordinate compression: each dim's boundaries split space into at most
    2k-1 intervals; a grid cell is covered iff any region contains it.
    Cost O((2k)^ndim · k) — fine for the shard counts resharding produces.
    """
    import itertools

    ndim = len(regions[0][0])
    bounds = [
        sorted(
            {r[0][d] for r in regions} | {r[0][d] + r[1][d] for r in regions}
        )
        for d in range(ndim)
    ]
    cells = [list(zip(b[:-1], b[1:])) for b in bounds]
    total = 0
    for cell in itertools.product(*cells):
        if any(
            all(
                r[0][d] <= cell[d][0] and cell[d][1] <= r[0][d] + r[1][d]
                for d in range(ndim)
            )
            for r in regions
        ):
            vol = 1
            for lo, hi in cell:
                vol *= hi - lo
            total += vol
    return total


def assemble(
    parts: Sequence[Tuple[Tuple[int, ...], torch.Tensor]],
    out: Optional[torch.Tensor] = None,
    out_offsets: Optional[Sequence[int]] = None,
) -> Tuple[torch.Tensor, Tuple[int, ...]]:
    """Scatter ``(global_offsets, tensor)`` parts into one tensor.

    When ``out`` is None a new tensor covering the parts' bounding box is
    allocated (on the first part's device/dtype) and its global origin is
    returned alongside.  When ``out``/``out_offsets`` are given the parts are
    scattered into that tensor instead.

    Overlapping parts are allowed (replicated shards); later parts win, and
    since replicas are bit-identical the order does not matter.
    """
    if not parts:
        raise ValueError("no parts to assemble")
    regions = [(off, tuple(t.shape)) for off, t in parts]
    if out is None:
        origin, shape = bounding_box(regions)
        first = parts[0][1]
        total = 1
        for s in shape:
            total *= s
        # disjoint parts (the usual case): numel sum is the exact coverage.
        # Partially overlapping parts could hide a gap behind the sum, so
        # they get the exact (compressed-grid) union volume instead —
        # a gap would otherwise return uninitialised torch.empty memory.
        if regions_disjoint(regions):
            covered = sum(t.numel() for _, t in parts)
        else:
            covered = union_volume(regions)
        if covered < total:
            raise ValueError(
                f"parts cover {covered} elements but bounding box has {total}: "
                "fetched shards do not tile the requested region"
            )
        out = torch.empty(shape, dtype=first.dtype, device=first.device)
        out_offsets = origin
    else:
        if out_offsets is None:
            raise ValueError("out_offsets required with out")
        origin = tuple(out_offsets)
    for off, t in parts:
        dst = extract_region(out, origin, off, tuple(t.shape))
        dst.copy_(t)
    return out, origin


def byte_view(t: torch.Tensor) -> torch.Tensor:
    """Flat uint8 view of a contiguous tensor (zero-copy)."""
    if t.numel() == 0:
        return torch.empty(0, dtype=torch.uint8, device=t.device)
    if not t.is_contiguous():
        raise ValueError("byte_view requires a contiguous tensor")
    return t.reshape(-1).view(torch.uint8)


def same_memory(a: torch.Tensor, b: torch.Tensor) -> bool:
    """True when ``b``'s storage bytes lie within ``a``'s storage bytes."""
    if a.device != b.device:
        return False
    if a.untyped_storage().data_ptr() != b.untyped_storage().data_ptr():
        return False
    a_start = a.storage_offset() * a.element_size()
    a_end = a_start + max(0, a.numel()) * a.element_size()
    b_start = b.storage_offset() * b.element_size()
    b_end = b_start + max(0, b.numel()) * b.element_size()
    return a_start <= b_start and b_end <= a_end
