"""Per-phase transport fault injection (CPU, via fakes).

Mirrors the reference's per-phase failure suites
(/root/reference/tests/test_torchcomms_transport.py:190-203,558-658):
every phase of the chunked (windowed ≥2 GiB) HIP-IPC protocol can fail and
must release volume staging back to the pool; a failed PG rendezvous must
not poison the pair cache (publish-on-success).

The fakes replace only the native byte movers (export/resolve/copy) with
in-process equivalents (ctypes.memmove between CPU tensors), so the FULL
client↔volume protocol — double-buffered windows, commit ordering, abort —
runs exactly as on hardware.
"""

import asyncio
import ctypes

import pytest
import torch

from torchstore_amd.transport import hip_ipc
from torchstore_amd.transport.base import TransportContext
from torchstore_amd.transport.hip_ipc import (
    ChunkStagingCache,
    HipIpcTransportBuffer,
    IpcDescriptor,
    IpcOpenCache,
)
from torchstore_amd.types import Request


def _fake_export(t: torch.Tensor, generation=None) -> IpcDescriptor:
    return IpcDescriptor(
        handle=t.data_ptr().to_bytes(8, "little"),
        offset=0,
        nbytes=t.numel() * t.element_size(),
        dtype=t.dtype,
        shape=tuple(t.shape),
        device_index=-1,
    )


def _fake_resolve(self, desc: IpcDescriptor, local_device) -> int:
    return int.from_bytes(desc.handle, "little") + desc.offset


def _fake_run_copies(copies):
    for dst, _dd, src, _sd, n in copies:
        ctypes.memmove(dst, src, n)


class _FakeStore:
    """Volume-side store stub for chunk_get_init."""

    def __init__(self, value: torch.Tensor):
        self.value = value

    def fetch(self, request):
        return self.value


class _FakeVolume:
    """In-process 'volume': dispatches handshake RPCs to the same buffer's
    recv_handshake, optionally raising at a scheduled (phase, index)."""

    def __init__(self, buffer, device=torch.device("cpu"), store=None):
        self._buffer = buffer
        self.device = device
        self.store = store
        self.fail_at = None  # (phase, nth-call-of-that-phase)
        self._counts = {}
        self.handshake = self

    async def call_one(self, buffer, args, phase):
        n = self._counts.get(phase, 0)
        self._counts[phase] = n + 1
        if self.fail_at == (phase, n):
            raise RuntimeError(f"injected failure at {phase}[{n}]")
        return self._buffer.recv_handshake(args, phase, self)


@pytest.fixture
def chunked_env(monkeypatch):
    """Small windows + in-process fakes for the native layer."""
    monkeypatch.setattr(hip_ipc, "CHUNK_BYTES", 1024)
    monkeypatch.setattr(hip_ipc, "export_tensor", _fake_export)
    monkeypatch.setattr(hip_ipc, "_run_copies", _fake_run_copies)
    monkeypatch.setattr(IpcOpenCache, "resolve", _fake_resolve)
    # K1 pack / stream sync are GPU-only; the fake store serves contiguous
    # CPU tensors so pack_region is identity and sync is a no-op
    from torchstore_amd.ops import gpu as gpu_ops

    monkeypatch.setattr(gpu_ops, "pack_region", lambda t: t.contiguous())
    monkeypatch.setattr(
        torch.cuda, "current_stream", lambda dev=None: _NoopStream()
    )
    buffer = HipIpcTransportBuffer()
    client_ctx, volume_ctx = TransportContext(), TransportContext()
    buffer.bind_client(_Ref(), client_ctx)
    buffer.attach_volume(volume_ctx)
    volume = _FakeVolume(buffer)
    buffer._volume_ref.volume = volume
    return buffer, volume, volume_ctx


class _NoopStream:
    def synchronize(self):
        return None


class _Ref:
    volume = None
    device = "cpu"


async def _drive_put(buffer, t):
    return await buffer._chunked_put_windows(t)


def test_chunked_put_roundtrip_pipelined(chunked_env):
    """Happy path: windows pipelined over 3 staging chunks land bit-exact
    in the volume payload; the op's chunks return to the pool."""
    buffer, volume, volume_ctx = chunked_env
    t = torch.arange(1200, dtype=torch.uint8)  # 2 full windows + partial
    token = asyncio.run(_drive_put(buffer, t))
    cache: ChunkStagingCache = volume_ctx.cache(ChunkStagingCache)
    payload = cache.payload(token)
    assert torch.equal(payload, t)
    released = cache.release(token)
    assert released is payload
    assert len(cache.free) == 3 and not cache.by_token


@pytest.mark.parametrize("fail_phase,fail_idx", [
    ("chunk_put_init", 0),
    ("chunk_put_commit", 0),
    ("chunk_put_commit", 1),
])
def test_chunked_put_abort_releases_staging(chunked_env, fail_phase, fail_idx):
    buffer, volume, volume_ctx = chunked_env
    volume.fail_at = (fail_phase, fail_idx)
    t = torch.arange(3000, dtype=torch.uint8)
    with pytest.raises(RuntimeError, match="injected"):
        asyncio.run(_drive_put(buffer, t))
    cache: ChunkStagingCache = volume_ctx.cache(ChunkStagingCache)
    # the abort path's chunk_release must return every chunk to the pool
    assert not cache.by_token, "aborted op leaked staging chunks"
    if fail_phase != "chunk_put_init":
        assert len(cache.free) == 3


@pytest.mark.parametrize("fail_phase,fail_idx", [
    ("chunk_get_init", 0),
    ("chunk_get_fill", 1),
])
def test_chunked_get_abort_releases_staging(chunked_env, fail_phase, fail_idx):
    buffer, volume, volume_ctx = chunked_env
    stored = torch.arange(2500, dtype=torch.uint8)
    volume.store = _FakeStore(stored)
    volume.fail_at = (fail_phase, fail_idx)
    dest = torch.zeros_like(stored)
    req = Request(key="k")
    with pytest.raises(RuntimeError, match="injected"):
        asyncio.run(buffer._chunked_get_windows(req, dest))
    cache: ChunkStagingCache = volume_ctx.cache(ChunkStagingCache)
    assert not cache.by_token, "aborted get leaked staging chunks"


def test_chunked_get_roundtrip_pipelined(chunked_env):
    buffer, volume, volume_ctx = chunked_env
    stored = torch.arange(5000, dtype=torch.uint8)
    volume.store = _FakeStore(stored)
    dest = torch.zeros_like(stored)
    token = asyncio.run(buffer._chunked_get_windows(Request(key="k"), dest))
    assert torch.equal(dest, stored)
    cache: ChunkStagingCache = volume_ctx.cache(ChunkStagingCache)
    cache.release(token)
    assert not cache.by_token


def test_volume_reset_reclaims_leaked_chunks(chunked_env):
    """A client that dies mid-transfer leaks its chunks until the volume's
    transport context is reset (reference: reset reclaims resources)."""
    buffer, volume, volume_ctx = chunked_env
    cache: ChunkStagingCache = volume_ctx.cache(ChunkStagingCache)
    cache.acquire("dead-token", torch.zeros(8, dtype=torch.uint8),
                  torch.device("cpu"))
    assert cache.by_token
    volume_ctx.close()  # what StorageVolume.reset does
    assert not cache.by_token and not cache.free


def test_pg_pair_cache_publish_on_success(monkeypatch):
    """A pair whose first data op fails must be discarded (uniflow
    publish-on-success), so a retry builds a fresh rendezvous."""
    from torchstore_amd.transport.pg import GlooTransportBuffer, PgClientCache

    ctx = TransportContext()
    buffer = GlooTransportBuffer()

    class Ref:
        volume_id = "v0"
        volume = None

    buffer.bind_client(Ref(), ctx)

    class FailingVolume:
        class put:  # noqa: N801 — endpoint-shaped stub
            @staticmethod
            async def call_one(*a, **kw):
                raise ConnectionError("volume died before rendezvous")

    buffer._volume_ref.volume = FailingVolume()
    cache: PgClientCache = ctx.cache(PgClientCache)
    req = Request(key="k", tensor_val=torch.randn(4))
    with pytest.raises(ConnectionError):
        asyncio.run(buffer.put([req]))
    assert "v0" not in cache.pairs, "failed op published a poisoned pair"

    # a successful op (no tensors -> no PG needed) confirms the pair
    buffer2 = GlooTransportBuffer()
    buffer2.bind_client(Ref(), ctx)

    class OkVolume:
        class put:  # noqa: N801
            @staticmethod
            async def call_one(*a, **kw):
                return None

    buffer2._volume_ref.volume = OkVolume()
    obj_req = Request(key="o", objects={"x": 1}, is_object=True)
    asyncio.run(buffer2.put([obj_req]))
    assert cache.pairs["v0"].confirmed


def test_pg_pair_retry_builds_fresh_rendezvous(monkeypatch):
    """After a failed (discarded) pair, the next op creates a NEW pair id
    and TCPStore — the failed rendezvous is never retried in place."""
    from torchstore_amd.transport.pg import GlooTransportBuffer, PgClientCache

    ctx = TransportContext()

    class Ref:
        volume_id = "v1"
        volume = None

    cache: PgClientCache = ctx.cache(PgClientCache)

    b1 = GlooTransportBuffer()
    b1.bind_client(Ref(), ctx)
    first = cache.get_or_create("v1", "gloo")
    first_id = first.info.pair_id

    class Dead:
        class put:  # noqa: N801
            @staticmethod
            async def call_one(*a, **kw):
                raise ConnectionError("down")

    b1._volume_ref.volume = Dead()
    with pytest.raises(ConnectionError):
        asyncio.run(b1.put([Request(key="o", objects=1, is_object=True)]))
    assert "v1" not in cache.pairs
    fresh = cache.get_or_create("v1", "gloo")
    assert fresh.info.pair_id != first_id
    assert fresh.info.port != 0


@pytest.fixture
def pushpull_env(chunked_env, monkeypatch):
    """chunked_env plus a faked try_export (push/pull use it volume-side)."""
    monkeypatch.setattr(
        hip_ipc, "try_export", lambda t, generation=None: _fake_export(t)
    )
    return chunked_env


def test_push_put_roundtrip(pushpull_env):
    """Direct push: volume exports payloads, client writes them in place,
    volume_receive adopts; pending state fully consumed."""
    buffer, volume, volume_ctx = pushpull_env
    cache: ChunkStagingCache = volume_ctx.cache(ChunkStagingCache)
    t1 = torch.arange(100, dtype=torch.uint8)
    t2 = torch.arange(50, dtype=torch.uint8) + 7
    reqs = [Request(key="a"), Request(key="b")]
    payload = [("pending", None), ("pending", None)]
    asyncio.run(buffer._push_put([(0, t1), (1, t2)], reqs, payload))
    assert payload[0][0] == "pushed" and payload[1][0] == "pushed"
    assert len(cache.push_pending) == 1
    buffer.payload = payload
    out = asyncio.run(
        buffer.volume_receive(reqs, [None, None], torch.device("cpu"))
    )
    assert torch.equal(out[0], t1) and torch.equal(out[1], t2)
    assert not cache.push_pending, "consumed tokens must clear"


def test_push_abort_releases_pending(pushpull_env, monkeypatch):
    buffer, volume, volume_ctx = pushpull_env
    cache: ChunkStagingCache = volume_ctx.cache(ChunkStagingCache)

    def boom(copies):
        raise RuntimeError("injected copy failure")

    monkeypatch.setattr(hip_ipc, "_run_copies", boom)
    t = torch.ones(64, dtype=torch.uint8)
    with pytest.raises(RuntimeError, match="injected"):
        asyncio.run(
            buffer._push_put([(0, t)], [Request(key="a")], [("pending", None)])
        )
    assert not cache.push_pending, "aborted push leaked pending payloads"


def test_pull_get_roundtrip_and_stash_release(pushpull_env):
    buffer, volume, volume_ctx = pushpull_env
    cache: ChunkStagingCache = volume_ctx.cache(ChunkStagingCache)
    stored = torch.arange(200, dtype=torch.uint8)
    volume.store = _FakeStore(stored)
    dest = torch.zeros_like(stored)
    payload = {0: None}
    asyncio.run(
        buffer._batched_pull([(0, Request(key="k"), dest)], payload)
    )
    assert payload[0] == ("pulled", None)
    assert torch.equal(dest, stored)
    assert not cache.pull_stash, "pull_release must clear the stash"


def test_pull_abort_releases_stash(pushpull_env, monkeypatch):
    buffer, volume, volume_ctx = pushpull_env
    cache: ChunkStagingCache = volume_ctx.cache(ChunkStagingCache)
    volume.store = _FakeStore(torch.ones(64, dtype=torch.uint8))

    def boom(copies):
        raise RuntimeError("injected pull copy failure")

    monkeypatch.setattr(hip_ipc, "_run_copies", boom)
    dest = torch.zeros(64, dtype=torch.uint8)
    with pytest.raises(RuntimeError, match="injected"):
        asyncio.run(
            buffer._batched_pull([(0, Request(key="k"), dest)], {0: None})
        )
    assert not cache.pull_stash
