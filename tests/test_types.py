"""Request / TensorSlice wire types (reference test_tensor_slice.py):
DTensor → Request building, trivially-local demotion (the EP/MoE case),
meta stripping, LocalShard passthrough."""

import tempfile

import pytest
import torch

from torchstore_amd.types import LocalShard, Request, TensorSlice


def _pg1():
    import torch.distributed as dist

    if not dist.is_initialized():
        dist.init_process_group(
            "gloo", init_method=f"file://{tempfile.mktemp()}",
            rank=0, world_size=1,
        )
    return dist


def test_mesh1_and_replicate_dtensors_demote_to_plain():
    """Reference types.py:58-85,145-151: a DTensor whose local tensor IS
    the full tensor (mesh size 1, or all-Replicate) stores as a PLAIN
    tensor — no TensorSlice, no commit gate."""
    dist = _pg1()
    try:
        from torch.distributed.device_mesh import init_device_mesh
        from torch.distributed.tensor import (
            Replicate,
            Shard,
            distribute_tensor,
        )

        mesh = init_device_mesh("cpu", (1,))
        t = torch.randn(8, 4)
        for placements in ([Shard(0)], [Shard(1)], [Replicate()]):
            dt = distribute_tensor(t.clone(), mesh, placements)
            req = Request.from_any("k", dt)
            assert req.tensor_slice is None, placements
            assert torch.equal(req.tensor_val, t)
            assert not req.is_object
    finally:
        dist.destroy_process_group()


def test_localshard_and_plain_tensor_requests():
    t = torch.randn(6, 6)
    req = Request.from_any("k", t)
    assert req.tensor_slice is None and req.tensor_val is t

    s = TensorSlice((0, 0), (3, 6), (6, 6), (0,), (2,))
    req2 = Request.from_any("k", LocalShard(tensor=t[:3], slice=s))
    assert req2.tensor_slice == s
    assert torch.equal(req2.tensor_val, t[:3])

    req3 = Request.from_any("k", {"cfg": 1})
    assert req3.is_object and req3.objects == {"cfg": 1}


def test_meta_only_strips_payload():
    t = torch.randn(4)
    req = Request.from_any("k", t)
    m = req.meta_only()
    assert m.tensor_val is None and m.key == "k"
    assert not m.has_payload
    assert req.tensor_val is t  # original untouched

    obj = Request.from_any("k", [1, 2, 3])
    mo = obj.meta_only()
    assert mo.is_object and mo.objects is None


def test_tensor_slice_normalization_and_hash():
    a = TensorSlice([0, 2], [4, 4], [8, 8], [1], [2])
    b = TensorSlice((0, 2), (4, 4), (8, 8), (1,), (2,))
    assert a == b and hash(a) == hash(b)
    assert a.offsets == (0, 2) and isinstance(a.offsets[0], int)
    assert a.numel() == 16


def test_expected_coordinates_cartesian():
    coords = list(TensorSlice.expected_coordinates((2, 3)))
    assert len(coords) == 6
    assert tuple(coords[0]) == (0, 0) and tuple(coords[-1]) == (1, 2)
    assert list(TensorSlice.expected_coordinates(())) == [()]


def test_plan_row_split_math():
    from torchstore_amd.client import _plan_row_split

    G = 1 << 30
    # under the limit: no split
    assert _plan_row_split((1024, 1024), 4) is None
    # 2.5 GB f32 1-D: pieces of 1 GiB rows
    rows = 640_000_000
    rpp, k = _plan_row_split((rows,), 4)
    assert rpp == G // 4 and k == 3
    assert (k - 1) * rpp < rows <= k * rpp
    # 2-D: row granularity respected
    rpp, k = _plan_row_split((300_000_000, 2), 4)
    assert rpp == G // 8 and k == 3
    # single-row giant: unsplittable
    assert _plan_row_split((1, 1 << 30), 4) is None
    # a row itself over the limit: unsplittable
    assert _plan_row_split((4, 1 << 30), 4) is None
    # 0-d / empty
    assert _plan_row_split((), 4) is None
    # exactly at the limit splits
    assert _plan_row_split(((1 << 31) // 4, 1), 4) is not None


def test_strategy_selection_units():
    """Placement-strategy mapping units (reference strategy.py:111-188)."""
    import os

    from torchstore_amd.strategy import (
        HostStrategy,
        LocalRankStrategy,
        SingletonStrategy,
        strategy_from_spec,
    )
    from torchstore_amd.transport import TransportType

    lr = LocalRankStrategy()
    os.environ["RANK"] = "3"
    try:
        assert lr.client_id() == "3"
        assert lr.select_volume_id(["0", "1", "2", "3"]) == "3"
        # more clients than volumes: deterministic modulo placement
        assert lr.select_volume_id(["0", "1"]) == "1"
        assert lr.num_volumes_for(8, 2) == 8
    finally:
        del os.environ["RANK"]

    hs = HostStrategy()
    os.environ["HOSTNAME"] = "hx"
    try:
        assert hs.select_volume_id(["ha", "hx"]) == "hx"
        assert hs.select_volume_id(["ha", "hb"]) == "ha"  # fallback sorted
        assert hs.num_volumes_for(16, 2) == 2
    finally:
        del os.environ["HOSTNAME"]

    ss = SingletonStrategy(transport=TransportType.RPC)
    assert ss.select_volume_id(["b", "a"]) == "a"
    assert ss.num_volumes_for(99, 9) == 1

    # spec round-trips the kind AND the forced transport (attaching
    # clients reconstruct the spawning strategy from it)
    back = strategy_from_spec(ss.spec())
    assert type(back) is SingletonStrategy
    assert back.transport == TransportType.RPC
    assert type(strategy_from_spec(None)) is SingletonStrategy


def test_gpu_descriptor_builders():
    """CPU units for the kernel-descriptor builders: shapes/strides in
    BYTES, contiguous fast paths, rejection of inexpressible layouts."""
    from torchstore_amd.ops.gpu import (
        _desc_view_to_ptr,
        _pitched_params,
        _slice_desc,
    )

    a = torch.zeros(8, 6)
    b = torch.zeros(8, 6)
    # both contiguous → flat byte descriptor
    d = _slice_desc(a, b)
    assert d == (a.data_ptr(), b.data_ptr(), 8 * 6 * 4, [], [], [])
    # strided src → rows of row_bytes with byte strides
    big = torch.zeros(8, 12)
    src = big[:, 3:9]
    d = _slice_desc(src, b)
    assert d[2] == 6 * 4 and d[3] == [8]
    assert d[4] == [12 * 4] and d[5] == [6 * 4]
    # shape/dtype mismatch → None (caller falls back)
    assert _slice_desc(a, torch.zeros(6, 8)) is None
    assert _slice_desc(a, torch.zeros(8, 6, dtype=torch.float16)) is None
    # inner-stride != 1 → None
    assert _slice_desc(a.t(), torch.zeros(6, 8)) is None
    # empty → () sentinel (skip, not fallback)
    assert _slice_desc(torch.zeros(0), torch.zeros(0)) == ()
    # 0-d → single-element descriptor
    d = _slice_desc(torch.zeros(()), torch.zeros(()))
    assert d[2] == 4 and d[3] == []

    # fused view→ptr: contiguous dest strides synthesized for src's shape
    v = big[2:6, 3:9]  # 4x6 strided view
    d = _desc_view_to_ptr(v, 0xDEAD)
    assert d[1] == 0xDEAD and d[2] == 6 * 4
    assert d[3] == [4] and d[4] == [12 * 4] and d[5] == [6 * 4]
    assert _desc_view_to_ptr(a.t(), 0) is None

    # pitched params: (pitch, width, height) in bytes
    assert _pitched_params(v) == (12 * 4, 6 * 4, 4)
    assert _pitched_params(torch.zeros(10)) == (40, 40, 1)
    assert _pitched_params(a.t()) is None
    assert _pitched_params(torch.zeros(2, 3, 4)) is None
