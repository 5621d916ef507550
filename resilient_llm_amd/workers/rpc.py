"""Router <-> worker RPC over unix sockets (length-prefixed msgpack).

The reference's router->backend transport is HTTPS to Bedrock
(SURVEY.md §5.8); here workers are processes on the same node (one per
GPU, pinned via HIP_VISIBLE_DEVICES), so the transport is a unix socket
with multiplexed request ids and push-style streaming frames.
"""

from __future__ import annotations

import asyncio
import dataclasses
import struct
from typing import AsyncIterator, Optional

import msgpack

from .base import (
    GenerationChunk, GenerationRequest, GenerationResult, Worker,
    WorkerDead, WorkerError, WorkerMigrated, WorkerThrottled,
)

_LEN = struct.Struct("<I")
MAX_FRAME = 64 * 1024 * 1024


async def send_frame(writer: asyncio.StreamWriter, obj) -> None:
    data = msgpack.packb(obj, use_bin_type=True)
    writer.write(_LEN.pack(len(data)) + data)
    await writer.drain()


async def recv_frame(reader: asyncio.StreamReader):
    head = await reader.readexactly(_LEN.size)
    (n,) = _LEN.unpack(head)
    if n > MAX_FRAME:
        raise ValueError("frame too large")
    data = await reader.readexactly(n)
    return msgpack.unpackb(data, raw=False)


# --------------------------------------------------------------- server
class WorkerRpcServer:
    """Wraps a Worker (EngineWorker) behind a unix socket."""

    def __init__(self, worker: Worker, socket_path: str,
                 on_kill=None) -> None:
        self.worker = worker
        self.socket_path = socket_path
        self.on_kill = on_kill          # subprocess: hard-exit hook
        self._server: Optional[asyncio.base_events.Server] = None

    async def start(self) -> None:
        self._server = await asyncio.start_unix_server(self._conn,
                                                       path=self.socket_path)

    async def serve_forever(self) -> None:
        await self.start()
        async with self._server:
            await self._server.serve_forever()

    async def stop(self) -> None:
        if self._server:
            self._server.close()
            await self._server.wait_closed()

    async def _conn(self, reader: asyncio.StreamReader,
                    writer: asyncio.StreamWriter) -> None:
        lock = asyncio.Lock()   # serialize frame writes across tasks

        async def reply(obj):
            async with lock:
                await send_frame(writer, obj)

        try:
            while True:
                try:
                    msg = await recv_frame(reader)
                except (asyncio.IncompleteReadError, ConnectionResetError):
                    return
                except Exception:
                    # malformed frame (bad length, not msgpack): drop the
                    # CONNECTION, never the worker
                    return
                if not isinstance(msg, dict):
                    continue
                asyncio.ensure_future(self._dispatch(msg, reply))
        finally:
            writer.close()

    async def _dispatch(self, msg, reply) -> None:
        mid = msg.get("id")
        op = msg.get("op")
        data = msg.get("data") or {}
        try:
            if op == "generate":
                req = GenerationRequest(**data)
                res = await self.worker.generate(req)
                await reply({"id": mid, "type": "result",
                             "data": dataclasses.asdict(res)})
            elif op == "stream":
                req = GenerationRequest(**data)
                async for chunk in self.worker.generate_stream(req):
                    await reply({"id": mid, "type": "chunk",
                                 "data": dataclasses.asdict(chunk)})
                await reply({"id": mid, "type": "end", "data": None})
            elif op == "health":
                await reply({"id": mid, "type": "result",
                             "data": await self.worker.health()})
            elif op == "list_requests":
                await reply({"id": mid, "type": "result",
                             "data": {"ids": self.worker.list_requests()}})
            elif op == "migrate_out":
                blob = await self.worker.migrate_out(data["request_id"])
                await reply({"id": mid, "type": "result",
                             "data": {"state": blob}})
            elif op == "migrate_in":
                await self.worker.migrate_in(data["state"])
                await reply({"id": mid, "type": "result", "data": {}})
            elif op == "release_migrated":
                await self.worker.release_migrated(data["request_id"])
                await reply({"id": mid, "type": "result", "data": {}})
            elif op == "fault":
                mode = data.get("mode", "none")
                if mode == "kill" and self.on_kill is not None:
                    await reply({"id": mid, "type": "result", "data": {"mode": mode}})
                    self.on_kill()
                    return
                await self.worker.inject_fault(mode)
                await reply({"id": mid, "type": "result", "data": {"mode": mode}})
            else:
                raise ValueError(f"unknown op {op!r}")
        except Exception as e:
            kind = ("migrated" if isinstance(e, WorkerMigrated) else
                    "throttled" if isinstance(e, WorkerThrottled) else
                    "dead" if isinstance(e, WorkerDead) else "error")
            try:
                await reply({"id": mid, "type": "error",
                             "data": {"kind": kind, "message": str(e)}})
            except Exception:
                pass


# --------------------------------------------------------------- client
def _raise_remote(data: dict) -> None:
    kind = data.get("kind", "error")
    msg = data.get("message", "remote error")
    if kind == "migrated":
        raise WorkerMigrated(msg)
    if kind == "throttled":
        raise WorkerThrottled(msg)
    if kind == "dead":
        raise WorkerDead(msg)
    raise WorkerError(msg)


class RpcWorkerClient(Worker):
    """Gateway-side proxy for a worker process (device string mirrors the
    remote's, e.g. gpu:3)."""

    def __init__(self, device: str, models: set[str], socket_path: str) -> None:
        super().__init__(device=device, models=models)
        self.socket_path = socket_path
        self._reader: Optional[asyncio.StreamReader] = None
        self._writer: Optional[asyncio.StreamWriter] = None
        self._wlock = asyncio.Lock()
        self._next_id = 0
        self._queues: dict[int, asyncio.Queue] = {}
        self._reader_task: Optional[asyncio.Task] = None
        self._in_flight = 0
        self.proc = None              # set by the spawner (subprocess handle)
        self.proc_group = None        # TP pools: every rank's Popen
        self.respawn = None           # callable -> new Popen (elastic recovery)
        self.last_respawn = 0.0

    @property
    def in_flight(self) -> int:
        return self._in_flight

    async def connect(self, timeout: float = 300.0) -> None:
        deadline = asyncio.get_running_loop().time() + timeout
        last_err = None
        while asyncio.get_running_loop().time() < deadline:
            try:
                self._reader, self._writer = await asyncio.open_unix_connection(
                    self.socket_path)
                self._reader_task = asyncio.create_task(self._read_loop())
                return
            except (ConnectionRefusedError, FileNotFoundError, OSError) as e:
                last_err = e
                if self.proc is not None and self.proc.poll() is not None:
                    raise WorkerDead(
                        f"worker process for {self.device} exited "
                        f"rc={self.proc.returncode} before serving")
                await asyncio.sleep(0.25)
        raise WorkerDead(f"cannot connect to {self.socket_path}: {last_err}")

    def _connection_lost(self) -> None:
        for q in self._queues.values():
            q.put_nowait({"type": "error",
                          "data": {"kind": "dead",
                                   "message": f"{self.device} connection lost"}})
        self._queues.clear()
        self._reader = None
        self._writer = None

    async def _read_loop(self) -> None:
        try:
            while True:
                msg = await recv_frame(self._reader)
                q = self._queues.get(msg.get("id"))
                if q is not None:
                    q.put_nowait(msg)
        except (asyncio.IncompleteReadError, ConnectionResetError, OSError):
            self._connection_lost()
        except asyncio.CancelledError:
            pass

    async def _call(self, op: str, data) -> tuple[int, asyncio.Queue]:
        if self._writer is None:
            # mid-operation reconnect: keep this SHORT.  A dead worker
            # raises WorkerDead immediately (poll() check in connect);
            # a BOOTING respawn can take longer than any caller should
            # block — respawn_now() owns the long reconnect (900 s), and
            # callers see a fast WorkerDead instead of a hung RPC (an
            # r02 soak hung /admin/drain for minutes exactly here).
            await self.connect(timeout=15.0)
        self._next_id += 1
        mid = self._next_id
        q: asyncio.Queue = asyncio.Queue()
        self._queues[mid] = q
        try:
            async with self._wlock:
                await send_frame(self._writer, {"id": mid, "op": op, "data": data})
        except (ConnectionResetError, BrokenPipeError, OSError) as e:
            self._queues.pop(mid, None)
            self._connection_lost()
            raise WorkerDead(f"{self.device} send failed: {e}") from e
        return mid, q

    async def generate(self, req: GenerationRequest) -> GenerationResult:
        self._in_flight += 1
        mid = None
        try:
            mid, q = await self._call("generate", dataclasses.asdict(req))
            msg = await q.get()
            if msg["type"] == "error":
                _raise_remote(msg["data"])
            return GenerationResult(**msg["data"])
        finally:
            self._in_flight -= 1
            if mid is not None:
                self._queues.pop(mid, None)

    async def _stream_impl(self, req: GenerationRequest) -> AsyncIterator[GenerationChunk]:
        self._in_flight += 1
        mid, q = await self._call("stream", dataclasses.asdict(req))
        try:
            while True:
                msg = await q.get()
                if msg["type"] == "error":
                    _raise_remote(msg["data"])
                if msg["type"] == "end":
                    return
                yield GenerationChunk(**msg["data"])
        finally:
            self._in_flight -= 1
            self._queues.pop(mid, None)

    def generate_stream(self, req: GenerationRequest) -> AsyncIterator[GenerationChunk]:
        return self._stream_impl(req)

    async def health(self) -> dict:
        if self.proc is not None and self.proc.poll() is not None:
            raise WorkerDead(f"worker process {self.device} is dead "
                             f"(rc={self.proc.returncode})")
        _, q = await self._call("health", {})
        msg = await asyncio.wait_for(q.get(), timeout=10.0)
        if msg["type"] == "error":
            _raise_remote(msg["data"])
        return msg["data"]

    async def list_requests(self) -> list:
        _, q = await self._call("list_requests", {})
        msg = await asyncio.wait_for(q.get(), timeout=30.0)
        if msg["type"] == "error":
            _raise_remote(msg["data"])
        return msg["data"]["ids"]

    async def migrate_out(self, request_id: str) -> bytes:
        _, q = await self._call("migrate_out", {"request_id": request_id})
        msg = await asyncio.wait_for(q.get(), timeout=60.0)
        if msg["type"] == "error":
            _raise_remote(msg["data"])
        return msg["data"]["state"]

    async def migrate_in(self, blob: bytes) -> None:
        _, q = await self._call("migrate_in", {"state": blob})
        msg = await asyncio.wait_for(q.get(), timeout=60.0)
        if msg["type"] == "error":
            _raise_remote(msg["data"])

    async def release_migrated(self, request_id: str) -> None:
        _, q = await self._call("release_migrated",
                                {"request_id": request_id})
        msg = await asyncio.wait_for(q.get(), timeout=30.0)
        if msg["type"] == "error":
            _raise_remote(msg["data"])

    async def inject_fault(self, mode: str) -> None:
        _, q = await self._call("fault", {"mode": mode})
        try:
            msg = await asyncio.wait_for(q.get(), timeout=10.0)
            if msg["type"] == "error":
                _raise_remote(msg["data"])
        except (asyncio.TimeoutError, WorkerDead):
            if mode != "kill":
                raise

    async def respawn_now(self) -> bool:
        """Elastic recovery: restart a dead worker
        process and reconnect.  Returns True once serving again.

        TP pools restart as a WHOLE group (every rank must rejoin a
        fresh RCCL rendezvous): surviving followers are reaped first,
        and a ``respawn`` hook that returns a list re-populates
        ``proc_group`` with the new ranks (leader first)."""
        import time as _time
        if self.respawn is None:
            return False
        if self.proc is not None and self.proc.poll() is None:
            return True
        self.last_respawn = _time.monotonic()
        followers = (self.proc_group or [])[1:]
        for p in followers:
            if p.poll() is None:
                p.terminate()
        for p in followers:
            try:
                p.wait(timeout=10)
            except Exception:
                p.kill()
        if self._reader_task:
            self._reader_task.cancel()
            self._reader_task = None
        self._connection_lost()
        try:
            import os as _os
            _os.unlink(self.socket_path)
        except OSError:
            pass
        spawned = self.respawn()
        if isinstance(spawned, list):
            self.proc_group = spawned
            self.proc = spawned[0]
        else:
            self.proc = spawned
        await self.connect(timeout=900)
        return True

    async def close(self) -> None:
        if self._reader_task:
            self._reader_task.cancel()
        if self._writer is not None:
            try:
                self._writer.close()
            except Exception:
                pass
        procs = self.proc_group or ([self.proc] if self.proc else [])
        for p in procs:
            if p.poll() is None:
                p.terminate()
        for p in procs:
            try:
                p.wait(timeout=10)
            except Exception:
                p.kill()
