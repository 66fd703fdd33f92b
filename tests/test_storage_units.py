"""InMemoryStore unit coverage: the three entry shapes (tensor / object /
per-coordinate shard dict), slice serving, shard search, meta, in-place
find_existing, epoch clearing (reference storage_volume.py:146-407)."""

import pytest
import torch

from torchstore_amd.storage import OBJ_SENTINEL, InMemoryStore
from torchstore_amd.types import Request, TensorSlice


def _sl(off, shape, gshape, coord=(), mesh=()):
    return TensorSlice(
        offsets=off, local_shape=shape, global_shape=gshape,
        coordinates=coord, mesh_shape=mesh,
    )


def test_three_entry_shapes_and_meta():
    s = InMemoryStore("cpu")
    t = torch.randn(8, 4)
    s.put(Request(key="t"), t)
    s.put(Request(key="o", is_object=True), {"a": 1})
    s.put(
        Request(key="sh", tensor_slice=_sl((0, 0), (4, 4), (8, 4), (0,), (2,))),
        t[:4].clone(),
    )
    assert torch.equal(s.fetch(Request(key="t")), t)
    assert s.fetch(Request(key="o")) == {"a": 1}
    m = s.meta(Request(key="t"))
    assert m.shape == (8, 4) and m.dtype == torch.float32
    assert s.meta(Request(key="o")) == OBJ_SENTINEL
    # sharded meta reports the GLOBAL shape without a slice request
    assert s.meta(Request(key="sh")).shape == (8, 4)
    # ...and the local region shape with one
    assert s.meta(
        Request(key="sh", tensor_slice=_sl((1, 0), (2, 4), (8, 4)))
    ).shape == (2, 4)
    assert sorted(s.keys()) == ["o", "sh", "t"]


def test_subslice_of_full_tensor():
    s = InMemoryStore("cpu")
    t = torch.arange(64, dtype=torch.float32).reshape(8, 8)
    s.put(Request(key="t"), t)
    out = s.fetch(Request(key="t", tensor_slice=_sl((2, 3), (4, 2), (8, 8))))
    assert torch.equal(out, t[2:6, 3:5])


def test_shard_search_and_miss():
    s = InMemoryStore("cpu")
    full = torch.arange(16, dtype=torch.float32)
    for c in range(2):
        s.put(
            Request(key="w", tensor_slice=_sl((c * 8,), (8,), (16,), (c,), (2,))),
            full[c * 8 : (c + 1) * 8].clone(),
        )
    # region fully inside shard 1
    out = s.fetch(Request(key="w", tensor_slice=_sl((9,), (4,), (16,))))
    assert torch.equal(out, full[9:13])
    # region SPANNING both shards: the volume cannot serve it (clients
    # split such requests per shard)
    with pytest.raises(KeyError, match="contains region"):
        s.fetch(Request(key="w", tensor_slice=_sl((6,), (4,), (16,))))
    # full fetch of a sharded key needs a slice
    with pytest.raises(KeyError, match="slice request is required"):
        s.fetch(Request(key="w"))


def test_find_existing_in_place_contract():
    s = InMemoryStore("cpu")
    t = torch.zeros(4, 4)
    s.put(Request(key="k"), t)
    prior = s.find_existing(Request(key="k"))
    assert prior is not None
    prior.fill_(7.0)  # transports overwrite the live stored tensor
    assert s.fetch(Request(key="k")).eq(7.0).all()
    # mismatched kinds return None, never a wrong tensor
    assert s.find_existing(Request(key="k", tensor_slice=_sl((0,), (4,), (8,)))) is None
    assert s.find_existing(Request(key="nope")) is None
    sl = _sl((0,), (4,), (8,), (0,), (2,))
    s.put(Request(key="sh", tensor_slice=sl), torch.ones(4))
    assert s.find_existing(Request(key="sh", tensor_slice=sl)) is not None
    other = _sl((4,), (4,), (8,), (1,), (2,))
    assert s.find_existing(Request(key="sh", tensor_slice=other)) is None


def test_epoch_clear_in_store():
    s = InMemoryStore("cpu")
    for c in range(2):
        s.put(
            Request(key="w", tensor_slice=_sl((c * 4,), (4,), (8,), (c,), (2,))),
            torch.full((4,), float(c)),
        )
    # same key, NEW global shape: stale shards must not survive
    s.put(
        Request(key="w", tensor_slice=_sl((0,), (6,), (12,), (0,), (2,))),
        torch.full((6,), 9.0),
    )
    entry = s.kv["w"]
    assert len(entry.shards) == 1
    (only_slice, tensor), = entry.shards.values()
    assert only_slice.global_shape == (12,) and tensor.eq(9.0).all()


def test_delete_and_reset():
    s = InMemoryStore("cpu")
    s.put(Request(key="a"), torch.ones(2))
    s.delete("a")
    with pytest.raises(KeyError):
        s.delete("a")
    s.delete("a", missing_ok=True)
    s.put(Request(key="b"), torch.ones(2))
    s.reset()
    assert s.keys() == []
