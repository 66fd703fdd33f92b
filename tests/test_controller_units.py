"""Controller index logic, unit-level (no actor processes): commit gate,
sharding epochs, write sequencing, fingerprints, delete bookkeeping."""

import pytest

from torchstore_amd.controller import (
    Controller,
    ObjectType,
    layout_fingerprint,
)
from torchstore_amd.types import Request, TensorSlice


def _slice(coord, mesh=(2,), rows=8):
    return TensorSlice(
        offsets=(coord[0] * rows,), local_shape=(rows,),
        global_shape=(mesh[0] * rows,), coordinates=coord, mesh_shape=mesh,
    )


def _shard_req(key, coord, mesh=(2,)):
    return Request(key=key, tensor_slice=_slice(coord, mesh))


def test_commit_gate_and_keys():
    c = Controller()
    c.notify_put_batch([_shard_req("w", (0,))], "v0")
    assert not c.key_exists("w")
    assert c.list_keys() == []
    with pytest.raises(KeyError, match="partially committed"):
        c.locate(["w"])
    c.notify_put_batch([_shard_req("w", (1,))], "v1")
    assert c.key_exists("w")
    assert c.list_keys() == ["w"]
    located = c.locate(["w"])["w"]
    assert set(located) == {"v0", "v1"}


def test_sharding_epoch_replaces_layout():
    c = Controller()
    c.notify_put_batch([_shard_req("w", (0,)), _shard_req("w", (1,))], "v0")
    assert c.key_exists("w")
    # re-push under mesh (4,): old-layout entries must be dropped, the
    # key re-gates until all four coordinates land
    c.notify_put_batch([_shard_req("w", (0,), mesh=(4,))], "v0")
    assert not c.key_exists("w")
    for j in range(1, 4):
        c.notify_put_batch([_shard_req("w", (j,), mesh=(4,))], "v0")
    assert c.key_exists("w")
    slices = c.locate(["w"])["w"]["v0"].tensor_slices
    assert {s.mesh_shape for s in slices} == {(4,)}
    assert len(slices) == 4


def test_write_seq_newest_kind_wins():
    c = Controller()
    c.notify_put_batch([_shard_req("k", (0,))], "v0")  # half-committed
    c.notify_put_batch([Request(key="k")], "v1")        # plain overwrite
    assert c.key_exists("k")  # newest kind = TENSOR -> readable
    newest = max(c.locate(["k"])["k"].values(), key=lambda i: i.seq)
    assert newest.object_type == ObjectType.TENSOR
    # ...and a newer half-shard makes it unreadable again
    c.notify_put_batch([_shard_req("k", (0,))], "v0")
    assert not c.key_exists("k")


def test_fingerprints_track_layout_not_data():
    c = Controller()
    c.notify_put_batch([Request(key="a")], "v0")
    fp1 = layout_fingerprint(c.locate(["a"])["a"])
    assert c.verify_layouts({"a": fp1})
    # same-layout re-put: fingerprint stable
    c.notify_put_batch([Request(key="a")], "v0")
    assert c.verify_layouts({"a": fp1})
    # new volume holding the key: fingerprint changes
    c.notify_put_batch([Request(key="a")], "v1")
    assert not c.verify_layouts({"a": fp1})
    # deleted key: verification fails (no KeyError)
    c.notify_delete("a")
    assert not c.verify_layouts({"a": fp1})


def test_delete_bookkeeping():
    c = Controller()
    c.notify_put_batch([Request(key="x")], "v0")
    c.notify_put_batch([Request(key="x")], "v1")
    vids = c.notify_delete("x")
    assert sorted(vids) == ["v0", "v1"]
    with pytest.raises(KeyError):
        c.notify_delete("x")
    assert c.notify_delete("x", missing_ok=True) == []
    out = c.notify_delete_batch(["x", "y"], missing_ok=True)
    assert out == {}


def test_meta_only_enforced():
    import torch

    c = Controller()
    bad = Request(key="k", tensor_val=torch.ones(2))
    with pytest.raises(AssertionError, match="meta-only"):
        c.notify_put_batch([bad], "v0")


def test_stats_counts():
    c = Controller()
    c.notify_put_batch([Request(key="o", is_object=True)], "v0")
    c.notify_put_batch([_shard_req("w", (0,)), _shard_req("w", (1,))], "v0")
    s = c.stats()
    assert s["keys"] == 2 and s["sharded_entries"] == 1
    assert s["total_shards"] == 2
