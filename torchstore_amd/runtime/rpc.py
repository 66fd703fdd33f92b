"""Asyncio TCP RPC with 64-bit frames and out-of-band tensor payloads.

This is the message layer of the actor runtime (the Monarch-RPC
replacement — reference surface at SURVEY §2.3).  One frame:

    [kind:u8][req_id:u64][nbufs:u32][header_len:u64][header bytes]
    ([buf_len:u64][buf bytes]) * nbufs

kinds: 0 = request, 1 = ok-response, 2 = error-response.
Request header: ``(method_name, args, kwargs)``; response header: the result
object or the remote exception.  Tensors inside args/results ride as the
out-of-band buffers (see :mod:`torchstore_amd.runtime.serde`).

Servers handle each request in its own asyncio task, so endpoints can await
each other without head-of-line blocking; responses are written under a
per-connection lock.
"""

from __future__ import annotations

import asyncio
import itertools
import struct
import traceback
from typing import Any, Dict, Optional, Tuple

from torchstore_amd.runtime import serde
from torchstore_amd.utils.logging import get_logger

logger = get_logger("torchstore_amd.rpc")

_HDR = struct.Struct("<BQIQ")  # kind, req_id, nbufs, header_len

KIND_REQUEST = 0
KIND_OK = 1
KIND_ERR = 2

SHUTDOWN_METHOD = "__shutdown__"
HEALTH_METHOD = "__health__"

# Raised client-side when the remote raised something unpicklable.
class RemoteError(RuntimeError):
    pass


async def _write_frame(
    writer: asyncio.StreamWriter,
    lock: asyncio.Lock,
    kind: int,
    req_id: int,
    obj: Any,
) -> None:
    # bufs alias live tensor memory (zero-copy): the memoryviews keep the
    # storage alive until flushed; a concurrent in-place overwrite of a
    # stored tensor during a reply can tear VALUES (never memory safety) —
    # the same window the reference documents for mutable SHM reads
    header, bufs = serde.dumps(obj)
    async with lock:
        writer.write(_HDR.pack(kind, req_id, len(bufs), len(header)))
        writer.write(header)
        for b in bufs:
            writer.write(struct.pack("<Q", b.nbytes))
            writer.write(b)
        await writer.drain()


async def _read_frame(
    reader: asyncio.StreamReader,
) -> Optional[Tuple[int, int, Any]]:
    try:
        head = await reader.readexactly(_HDR.size)
    except (asyncio.IncompleteReadError, ConnectionResetError):
        return None
    kind, req_id, nbufs, header_len = _HDR.unpack(head)
    header = await reader.readexactly(header_len)
    bufs = []
    for _ in range(nbufs):
        (blen,) = struct.unpack("<Q", await reader.readexactly(8))
        buf = bytearray(blen)
        view = memoryview(buf)
        got = 0
        while got < blen:
            chunk = await reader.read(min(blen - got, 16 << 20))
            if not chunk:
                raise ConnectionResetError("peer closed mid-frame")
            view[got : got + len(chunk)] = chunk
            got += len(chunk)
        bufs.append(buf)
    return kind, req_id, serde.loads(header, bufs)


class RpcServer:
    """Serves endpoint calls on a target object."""

    def __init__(self, target: Any, host: str = "127.0.0.1"):
        self._target = target
        self._host = host
        self._server: Optional[asyncio.AbstractServer] = None
        self.port: Optional[int] = None
        self.stopped = asyncio.Event()

    async def start(self) -> Tuple[str, int]:
        self._server = await asyncio.start_server(
            self._on_connection, self._host, 0
        )
        self.port = self._server.sockets[0].getsockname()[1]
        return self._host, self.port

    async def _on_connection(
        self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter
    ) -> None:
        wlock = asyncio.Lock()
        tasks = set()
        try:
            while True:
                frame = await _read_frame(reader)
                if frame is None:
                    break
                kind, req_id, payload = frame
                if kind != KIND_REQUEST:
                    logger.warning("server got non-request frame kind=%s", kind)
                    continue
                task = asyncio.create_task(
                    self._dispatch(writer, wlock, req_id, payload)
                )
                tasks.add(task)
                task.add_done_callback(tasks.discard)
        finally:
            for t in tasks:
                t.cancel()
            writer.close()

    async def _dispatch(self, writer, wlock, req_id: int, payload) -> None:
        method, args, kwargs = payload
        try:
            if method == SHUTDOWN_METHOD:
                await _write_frame(writer, wlock, KIND_OK, req_id, None)
                self.stopped.set()
                return
            if method == HEALTH_METHOD:
                await _write_frame(writer, wlock, KIND_OK, req_id, "ok")
                return
            fn = getattr(self._target, method, None)
            if fn is None or not getattr(fn, "_is_endpoint", False):
                raise AttributeError(
                    f"{type(self._target).__name__} has no endpoint {method!r}"
                )
            result = fn(*args, **kwargs)
            if asyncio.iscoroutine(result):
                result = await result
            await _write_frame(writer, wlock, KIND_OK, req_id, result)
        except Exception as exc:  # noqa: BLE001 — forwarded to caller
            try:
                await _write_frame(writer, wlock, KIND_ERR, req_id, exc)
            except Exception:
                tb = traceback.format_exc()
                await _write_frame(
                    writer, wlock, KIND_ERR, req_id,
                    RemoteError(f"{type(exc).__name__}: {exc}\n{tb}"),
                )

    async def serve_until_stopped(self) -> None:
        await self.stopped.wait()
        self._server.close()
        await self._server.wait_closed()

    async def close(self) -> None:
        self.stopped.set()


class RpcConnection:
    """Client side of one TCP connection; multiplexes concurrent calls."""

    def __init__(self, host: str, port: int):
        self.host = host
        self.port = port
        self._reader: Optional[asyncio.StreamReader] = None
        self._writer: Optional[asyncio.StreamWriter] = None
        self._wlock = asyncio.Lock()
        self._pending: Dict[int, asyncio.Future] = {}
        self._ids = itertools.count()
        self._reader_task: Optional[asyncio.Task] = None
        self._closed = False

    async def connect(self) -> None:
        self._reader, self._writer = await asyncio.open_connection(
            self.host, self.port
        )
        self._reader_task = asyncio.create_task(self._read_loop())

    async def _read_loop(self) -> None:
        try:
            while True:
                frame = await _read_frame(self._reader)
                if frame is None:
                    break
                kind, req_id, payload = frame
                fut = self._pending.pop(req_id, None)
                if fut is None or fut.done():
                    continue
                if kind == KIND_OK:
                    fut.set_result(payload)
                else:
                    fut.set_exception(
                        payload
                        if isinstance(payload, BaseException)
                        else RemoteError(str(payload))
                    )
        except Exception as exc:  # noqa: BLE001
            if not self._closed:
                logger.debug("rpc read loop ended: %s", exc)
        finally:
            err = ConnectionResetError(
                f"connection to {self.host}:{self.port} lost"
            )
            for fut in self._pending.values():
                if not fut.done():
                    fut.set_exception(err)
            self._pending.clear()

    @property
    def is_open(self) -> bool:
        return (
            self._writer is not None
            and not self._writer.is_closing()
            and not self._closed
        )

    async def call(self, method: str, *args, **kwargs) -> Any:
        req_id = next(self._ids)
        fut: asyncio.Future = asyncio.get_running_loop().create_future()
        self._pending[req_id] = fut
        try:
            await _write_frame(
                self._writer, self._wlock, KIND_REQUEST, req_id,
                (method, args, kwargs),
            )
        except BaseException:
            # the request never left (serde error on args, closed socket):
            # drop the pending entry or it leaks for the connection's life
            self._pending.pop(req_id, None)
            raise
        return await fut

    async def close(self) -> None:
        self._closed = True
        if self._reader_task:
            self._reader_task.cancel()
        if self._writer is not None:
            self._writer.close()
