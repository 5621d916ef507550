"""Minimal asyncio HTTP/1.1 server for the gateway.

Hand-rolled on purpose: the serving hot path is one POST with a JSON body
and either a JSON response or an SSE stream — a full framework stack adds
latency and buys nothing here.  Supports keep-alive, Content-Length
bodies, chunked responses for streaming (SSE), and nothing else.

Replaces the HTTP surface LiteLLM's proxy provides in the reference
(OpenAI-compatible REST on a configured port — reference
config/config.yaml:31-33).
"""

from __future__ import annotations

import asyncio
import json
import urllib.parse
from typing import AsyncIterator, Awaitable, Callable, Optional

MAX_HEADER_BYTES = 64 * 1024
MAX_BODY_BYTES = 32 * 1024 * 1024

_STATUS_TEXT = {
    200: "OK", 204: "No Content", 400: "Bad Request", 401: "Unauthorized",
    403: "Forbidden", 404: "Not Found", 405: "Method Not Allowed",
    408: "Request Timeout", 413: "Payload Too Large",
    429: "Too Many Requests", 500: "Internal Server Error",
    502: "Bad Gateway", 503: "Service Unavailable",
}


class Request:
    def __init__(self, method: str, target: str, headers: dict[str, str],
                 body: bytes, http_version: str) -> None:
        self.method = method
        self.http_version = http_version
        parsed = urllib.parse.urlsplit(target)
        self.path = parsed.path
        self.query = {k: v[-1] for k, v in
                      urllib.parse.parse_qs(parsed.query, keep_blank_values=True).items()}
        self.headers = headers            # keys lower-cased
        self.body = body

    def json(self):
        if not self.body:
            return None
        return json.loads(self.body.decode("utf-8"))


class Response:
    def __init__(self, status: int = 200, body: bytes | str | None = b"",
                 headers: Optional[dict[str, str]] = None,
                 content_type: str = "application/json",
                 body_iter: Optional[AsyncIterator[bytes]] = None) -> None:
        self.status = status
        self.headers = dict(headers or {})
        self.body_iter = body_iter
        if isinstance(body, str):
            body = body.encode("utf-8")
        self.body = body or b""
        self.headers.setdefault("content-type", content_type)

    @classmethod
    def json_response(cls, obj, status: int = 200,
                      headers: Optional[dict[str, str]] = None) -> "Response":
        return cls(status=status, body=json.dumps(obj), headers=headers)

    @classmethod
    def error(cls, status: int, message: str, err_type: str = "api_error",
              code: Optional[str] = None,
              headers: Optional[dict[str, str]] = None) -> "Response":
        """OpenAI-style error body — the SDK maps 429 to RateLimitError
        (reference demo_quota_isolation.py:95-107 relies on this, X9)."""
        return cls.json_response(
            {"error": {"message": message, "type": err_type,
                       "param": None, "code": code or str(status)}},
            status=status, headers=headers)


Handler = Callable[[Request], Awaitable[Response]]


class HttpServer:
    # a client that sends headers but never the body must not pin the
    # connection forever (idle KEEP-ALIVE waits between requests are
    # fine and unbounded; only the body read is deadline-bounded)
    body_timeout_s: float = 120.0

    def __init__(self, handler: Handler, host: str = "127.0.0.1", port: int = 4000) -> None:
        self.handler = handler
        self.host = host
        self.port = port
        self._server: Optional[asyncio.base_events.Server] = None

    async def start(self, reuse_port: bool = False) -> None:
        self._server = await asyncio.start_server(
            self._handle_conn, self.host, self.port,
            limit=MAX_HEADER_BYTES, reuse_address=True,
            reuse_port=reuse_port)

    def close_listener(self) -> None:
        """Stop ACCEPTING; existing connections keep being served
        (graceful-drain phase 1)."""
        if self._server is not None:
            self._server.close()

    async def stop(self) -> None:
        if self._server is not None:
            self._server.close()
            await self._server.wait_closed()
            self._server = None

    # ------------------------------------------------------------ parsing
    async def _read_request(self, reader: asyncio.StreamReader) -> Optional[Request]:
        try:
            head = await reader.readuntil(b"\r\n\r\n")
        except (asyncio.IncompleteReadError, ConnectionResetError):
            return None
        except asyncio.LimitOverrunError:
            raise ValueError("headers too large")
        lines = head.decode("latin-1").split("\r\n")
        try:
            method, target, version = lines[0].split(" ", 2)
        except ValueError:
            raise ValueError(f"bad request line: {lines[0]!r}")
        headers: dict[str, str] = {}
        for line in lines[1:]:
            if not line:
                continue
            k, _, v = line.partition(":")
            headers[k.strip().lower()] = v.strip()
        length = int(headers.get("content-length", "0") or "0")
        if length > MAX_BODY_BYTES:
            raise ValueError("body too large")
        if length:
            try:
                body = await asyncio.wait_for(reader.readexactly(length),
                                              timeout=self.body_timeout_s)
            except asyncio.TimeoutError:
                raise ValueError("body read timeout")
            except asyncio.IncompleteReadError:
                return None
        else:
            body = b""
        return Request(method, target, headers, body, version)

    # ----------------------------------------------------------- writing
    @staticmethod
    def _head_bytes(resp: Response, chunked: bool) -> bytes:
        lines = [f"HTTP/1.1 {resp.status} {_STATUS_TEXT.get(resp.status, 'Unknown')}"]
        headers = dict(resp.headers)
        if chunked:
            headers["transfer-encoding"] = "chunked"
            headers.pop("content-length", None)
        else:
            headers["content-length"] = str(len(resp.body))
        headers.setdefault("connection", "keep-alive")
        for k, v in headers.items():
            lines.append(f"{k}: {v}")
        return ("\r\n".join(lines) + "\r\n\r\n").encode("latin-1")

    async def _write_response(self, writer: asyncio.StreamWriter, resp: Response) -> None:
        if resp.body_iter is None:
            writer.write(self._head_bytes(resp, chunked=False) + resp.body)
            await writer.drain()
            return
        writer.write(self._head_bytes(resp, chunked=True))
        await writer.drain()
        try:
            async for chunk in resp.body_iter:
                if not chunk:
                    continue
                writer.write(f"{len(chunk):x}\r\n".encode() + chunk + b"\r\n")
                await writer.drain()
        finally:
            # on client disconnect the generator is abandoned mid-yield;
            # close it NOW so its cleanup (router ticket settlement) runs
            # promptly instead of at interpreter shutdown
            aclose = getattr(resp.body_iter, "aclose", None)
            if aclose is not None:
                try:
                    await aclose()
                except Exception:
                    pass
            try:
                writer.write(b"0\r\n\r\n")
                await writer.drain()
            except (ConnectionResetError, BrokenPipeError, OSError):
                pass

    # --------------------------------------------------------- connection
    async def _handle_conn(self, reader: asyncio.StreamReader,
                           writer: asyncio.StreamWriter) -> None:
        try:
            while True:
                try:
                    req = await self._read_request(reader)
                except ValueError as e:
                    await self._write_response(writer, Response.error(400, str(e)))
                    break
                if req is None:
                    break
                try:
                    resp = await self.handler(req)
                except Exception as e:  # handler bug -> 500, keep serving
                    resp = Response.error(500, f"internal error: {type(e).__name__}: {e}")
                await self._write_response(writer, resp)
                conn = req.headers.get("connection", "").lower()
                if conn == "close" or req.http_version == "HTTP/1.0":
                    break
        except (ConnectionResetError, BrokenPipeError, asyncio.CancelledError):
            pass
        finally:
            try:
                writer.close()
                await writer.wait_closed()
            except Exception:
                pass
