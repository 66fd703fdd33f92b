"""Transport-buffer unit tests (no actors, CPU only)."""

import torch

from torchstore_amd.runtime import serde
from torchstore_amd.transport.hip_ipc import HipIpcTransportBuffer, IpcDescriptor
from torchstore_amd.transport.shm import ShmDescriptor, ShmTransportBuffer
from torchstore_amd.transport.rpc import RpcTransportBuffer
from torchstore_amd.transport.base import TransportContext, TransportType


def test_ipc_buffer_strips_local_tensors():
    """Staged tensors must never ride the RPC frame (the 1 GB/s bug)."""
    buf = HipIpcTransportBuffer()
    buf._hold = [torch.randn(1024), torch.randn(2048)]
    buf._scratch = {0: torch.randn(512)}
    buf.payload = [
        ("ipc", IpcDescriptor(b"h" * 64, 0, 4096, torch.float32, (1024,), 0))
    ]
    header, bufs = serde.dumps(buf)
    assert bufs == [], "local tensors leaked into the serialized buffer"
    back = serde.loads(header, [])
    assert back._hold == [] and back._scratch == {}
    assert back.payload[0][1].nbytes == 4096


def test_shm_buffer_serializes_descriptors_only():
    buf = ShmTransportBuffer()
    buf.payload = [
        ("seg", ShmDescriptor(b"/m", b"/n", 64, torch.float32, (16,))),
        ("obj", {"a": 1}),
    ]
    buf.alloc_sizes = [64, None]
    header, bufs = serde.dumps(buf)
    assert bufs == []
    back = serde.loads(header, [])
    assert back.payload[0][1].name == b"/n"
    assert back.alloc_sizes == [64, None]


def test_rpc_buffer_carries_payload_out_of_band():
    buf = RpcTransportBuffer()
    t = torch.randn(256)
    buf.data = [("tensor", t)]
    header, bufs = serde.dumps(buf)
    assert len(bufs) == 1  # the tensor rides as one out-of-band buffer
    back = serde.loads(header, [bytearray(b) for b in bufs])
    assert torch.equal(back.data[0][1], t)


def test_pin_fail_open(monkeypatch):
    """hipHostRegister failures must warn once per code and keep copies
    working unpinned (reference: _FakeCudart, test_shared_memory.py:342)."""
    import warnings

    from torchstore_amd.transport import shm

    class FakeCudart:
        def cudaHostRegister(self, ptr, nbytes, flags):
            return 712  # hipErrorHostMemoryAlreadyRegistered-style failure

    monkeypatch.setattr(torch.cuda, "is_available", lambda: True)
    monkeypatch.setattr(torch.cuda, "cudart", lambda: FakeCudart())
    monkeypatch.setattr(shm, "_PIN_WARNED", set())

    seg = shm._allocate_segment(4096)
    pinned = {}
    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        shm._try_pin(seg, pinned)
        shm._try_pin(seg, pinned)  # second call: same code, no new warning
    assert pinned == {}  # fail-open: nothing recorded as pinned
    msgs = [str(x.message) for x in w if "hipHostRegister" in str(x.message)]
    assert len(msgs) == 1, msgs

    # a succeeding register IS recorded
    class GoodCudart:
        def cudaHostRegister(self, ptr, nbytes, flags):
            return 0

    monkeypatch.setattr(torch.cuda, "cudart", lambda: GoodCudart())
    shm._try_pin(seg, pinned)
    assert len(pinned) == 1
    pinned.clear()  # avoid unregister on a fake


def test_context_strip_on_serialization():
    buf = RpcTransportBuffer()
    buf.bind_client(object(), TransportContext())
    header, bufs = serde.dumps(buf)
    back = serde.loads(header, bufs)
    assert back._client_ctx is None and back._volume_ref is None


async def test_mutable_shm_zero_copy_get(monkeypatch):
    """TORCHSTORE_AMD_MUTABLE_SHM=1: a get returns the live segment view —
    a later put to the key is visible through the earlier result."""
    monkeypatch.setenv("TORCHSTORE_AMD_MUTABLE_SHM", "1")
    import torchstore_amd as ts
    from torchstore_amd.strategy import SingletonStrategy

    await ts.initialize(
        num_storage_volumes=1,
        strategy=SingletonStrategy(transport=TransportType.SHARED_MEMORY),
        storage_device="cpu",
    )
    try:
        await ts.put("live", torch.ones(16))
        view = await ts.get("live")
        assert view.eq(1).all()
        await ts.put("live", torch.full((16,), 5.0))
        # the earlier result aliases the volume's segment: update visible
        assert view.eq(5).all()
    finally:
        await ts.shutdown()


def test_shm_get_segments_are_client_keyed():
    """Two clients fetching the same key must get DISTINCT volume response
    segments: a shared segment lets a faster reader's refill tear the
    slower reader's copy."""
    import asyncio

    import torch

    from torchstore_amd.transport.base import TransportContext
    from torchstore_amd.transport.shm import (
        ShmTransportBuffer,
        ShmVolumeCache,
    )
    from torchstore_amd.types import Request

    volume_ctx = TransportContext()
    stored = torch.randn(64)

    async def fetch(client_uid):
        buf = ShmTransportBuffer()
        buf.client_uid = client_uid
        buf.attach_volume(volume_ctx)
        req = Request(key="k")
        reply = await buf.volume_send([req], [stored])
        kind, desc = reply[0]
        assert kind == "seg"
        return desc

    d1 = asyncio.run(fetch("client-A"))
    d2 = asyncio.run(fetch("client-B"))
    assert d1.seg_key != d2.seg_key, "response segment shared across clients"
    cache: ShmVolumeCache = volume_ctx.cache(ShmVolumeCache)
    assert len(cache.get_segments) == 2
    # same client re-fetching reuses its own segment
    d1b = asyncio.run(fetch("client-A"))
    assert d1b.seg_key == d1.seg_key
    volume_ctx.close()


def test_shm_put_segment_reuse_and_growth():
    """Re-put of the same key reuses the volume-owned segment; a LARGER
    re-put allocates a bigger one (reference SHM descriptor-reuse
    handshake, shared_memory.py:340-360)."""
    import asyncio

    import torch

    from torchstore_amd.transport.base import TransportContext
    from torchstore_amd.transport.shm import ShmTransportBuffer, ShmVolumeCache
    from torchstore_amd.types import Request

    volume_ctx = TransportContext()

    def handshake(nbytes):
        buf = ShmTransportBuffer()
        buf.client_uid = "c1"
        buf.alloc_sizes = [nbytes]
        buf.attach_volume(volume_ctx)
        (desc,) = buf.recv_handshake([Request(key="k")], "put", None)
        return desc

    d1 = handshake(1024)
    d2 = handshake(512)       # smaller fits -> same segment
    assert d2.seg_key == d1.seg_key
    d3 = handshake(4096)      # larger -> fresh, bigger segment
    assert d3.seg_key != d1.seg_key and d3.nbytes >= 4096
    # a different client never shares the put segment
    buf = ShmTransportBuffer()
    buf.client_uid = "c2"
    buf.alloc_sizes = [1024]
    buf.attach_volume(volume_ctx)
    (d4,) = buf.recv_handshake([Request(key="k")], "put", None)
    assert d4.seg_key != d3.seg_key
    volume_ctx.close()


def test_delete_drops_volume_shm_segments():
    """delete() must release the volume's SHM segments for the key (put-
    and get-side), not just the index entry."""
    import asyncio

    import torch

    from torchstore_amd.transport.base import TransportContext
    from torchstore_amd.transport.shm import ShmTransportBuffer, ShmVolumeCache
    from torchstore_amd.types import Request

    ctx = TransportContext()
    cache: ShmVolumeCache = ctx.cache(ShmVolumeCache)
    # put-side segment via the handshake
    buf = ShmTransportBuffer()
    buf.client_uid = "c"
    buf.alloc_sizes = [256]
    buf.attach_volume(ctx)
    buf.recv_handshake([Request(key="k1")], "put", None)
    # get-side segment via volume_send
    buf2 = ShmTransportBuffer()
    buf2.client_uid = "c"
    buf2.attach_volume(ctx)
    asyncio.run(buf2.volume_send([Request(key="k1")], [torch.randn(8)]))
    asyncio.run(buf2.volume_send([Request(key="k2")], [torch.randn(8)]))
    assert len(cache.put_segments) == 1 and len(cache.get_segments) == 2
    ctx.drop_key("k1")
    assert not cache.put_segments
    assert len(cache.get_segments) == 1
    assert next(iter(cache.get_segments))[1] == "k2"
    ctx.close()


def test_shm_adopted_entry_zero_copy_warm_get():
    """A CPU volume ADOPTS the typed view of its own put segment as
    storage; warm gets of that key return the SAME segment descriptor
    with no volume-side copy (reference warm-get reuse,
    shared_memory.py:340-360 + our ownership inversion)."""
    import asyncio

    import torch

    from torchstore_amd.transport.base import TransportContext
    from torchstore_amd.transport.shm import (
        ShmTransportBuffer,
        ShmVolumeCache,
        _typed_view,
    )
    from torchstore_amd.types import Request

    ctx = TransportContext()
    cache: ShmVolumeCache = ctx.cache(ShmVolumeCache)
    # put handshake allocates the segment; the volume adopts the view
    buf = ShmTransportBuffer()
    buf.client_uid = "c"
    buf.alloc_sizes = [1024]
    buf.attach_volume(ctx)
    (desc,) = buf.recv_handshake([Request(key="k")], "put", None)
    seg = cache.put_segments[("c", "k")][1]
    stored = _typed_view(seg, desc.with_layout(torch.float32, (256,)))
    stored.copy_(torch.arange(256, dtype=torch.float32))

    # warm get: the reply descriptor POINTS AT the adopted segment —
    # same seg_key, zero copies (desc_by_storage fast path)
    getbuf = ShmTransportBuffer()
    getbuf.client_uid = "reader"
    getbuf.attach_volume(ctx)
    reply = asyncio.run(getbuf.volume_send([Request(key="k")], [stored]))
    kind, rdesc = reply[0]
    assert kind == "seg"
    assert rdesc.seg_key == desc.seg_key, "warm get must reuse the segment"
    assert rdesc.dtype == torch.float32 and rdesc.shape == (256,)
    assert not cache.get_segments, "no response segment should be allocated"
    ctx.close()


def test_shm_descriptor_layout_roundtrip():
    """with_layout carries dtype/shape; _typed_view reconstructs exactly
    (incl. empty and odd shapes) from the raw byte segment."""
    import torch

    from torchstore_amd.transport.shm import (
        _allocate_segment,
        _segment_descriptor,
        _typed_view,
    )

    seg = _allocate_segment(4096)
    base = _segment_descriptor(seg)
    for dtype, shape in [
        (torch.float32, (16, 16)),
        (torch.bfloat16, (3, 5, 7)),
        (torch.int64, (512,)),
        (torch.float16, (0,)),
        (torch.uint8, ()),
    ]:
        d = base.with_layout(dtype, shape)
        v = _typed_view(seg, d)
        assert v.dtype == dtype and tuple(v.shape) == tuple(shape)
        if v.numel():
            v.fill_(1 if dtype == torch.int64 else 1.0)
            v2 = _typed_view(seg, d)
            assert v2.reshape(-1)[0].item() == 1


def test_shm_unpin_on_close(monkeypatch):
    """ctx.close() unregisters every pinned segment exactly once."""
    import torch

    from torchstore_amd.transport import shm

    calls = {"reg": 0, "unreg": 0}

    class FakeCudart:
        def cudaHostRegister(self, ptr, nbytes, flags):
            calls["reg"] += 1
            return 0

        def cudaHostUnregister(self, ptr):
            calls["unreg"] += 1
            return 0

    monkeypatch.setattr(torch.cuda, "is_available", lambda: True)
    monkeypatch.setattr(torch.cuda, "cudart", lambda: FakeCudart())
    pinned = {}
    s1 = shm._allocate_segment(1024)
    s2 = shm._allocate_segment(1024)
    shm._try_pin(s1, pinned)
    shm._try_pin(s1, pinned)  # idempotent per segment
    shm._try_pin(s2, pinned)
    assert calls["reg"] == 2 and len(pinned) == 2
    shm._unpin_all(pinned)
    assert calls["unreg"] == 2 and not pinned
