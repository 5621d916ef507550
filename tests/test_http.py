"""HTTP layer unit tests (hand-rolled server: parsing, keep-alive,
errors, chunked SSE framing)."""

import asyncio
import json

from resilient_llm_amd.gateway.http import HttpServer, Request, Response
from tests.gateway_harness import free_port


def run(coro):
    return asyncio.run(coro)


async def _start(handler):
    port = free_port()
    server = HttpServer(handler, port=port)
    await server.start()
    return server, port


def test_keepalive_and_parsing():
    async def main():
        seen = []

        async def handler(req: Request) -> Response:
            seen.append((req.method, req.path, dict(req.query),
                         req.headers.get("x-test"), req.body))
            return Response.json_response({"n": len(seen)})

        server, port = await _start(handler)
        reader, writer = await asyncio.open_connection("127.0.0.1", port)
        # two pipelined-ish requests on ONE connection (keep-alive)
        body = b'{"a": 1}'
        for i in range(2):
            writer.write(
                b"POST /x/y?q=1&r=two HTTP/1.1\r\nHost: t\r\nX-Test: v\r\n"
                b"Content-Length: " + str(len(body)).encode() + b"\r\n\r\n" + body)
            await writer.drain()
            line = await reader.readline()
            assert b"200" in line
            headers = b""
            while True:
                h = await reader.readline()
                headers += h
                if h == b"\r\n":
                    break
            n = int([l for l in headers.split(b"\r\n")
                     if l.lower().startswith(b"content-length")][0].split(b":")[1])
            data = await reader.readexactly(n)
            assert json.loads(data)["n"] == i + 1
        writer.close()
        assert seen[0][0] == "POST" and seen[0][1] == "/x/y"
        assert seen[0][2] == {"q": "1", "r": "two"}
        assert seen[0][3] == "v" and seen[0][4] == body
        await server.stop()

    run(main())


def test_bad_request_line_400():
    async def main():
        async def handler(req):
            return Response.json_response({})
        server, port = await _start(handler)
        reader, writer = await asyncio.open_connection("127.0.0.1", port)
        writer.write(b"GARBAGE\r\n\r\n")
        await writer.drain()
        line = await reader.readline()
        assert b"400" in line
        writer.close()
        await server.stop()
    run(main())


def test_handler_exception_becomes_500():
    async def main():
        async def handler(req):
            raise RuntimeError("boom")
        server, port = await _start(handler)
        reader, writer = await asyncio.open_connection("127.0.0.1", port)
        writer.write(b"GET / HTTP/1.1\r\n\r\n")
        await writer.drain()
        line = await reader.readline()
        assert b"500" in line
        writer.close()
        await server.stop()
    run(main())


def test_chunked_streaming_response():
    async def main():
        async def gen():
            yield b"data: one\n\n"
            yield b"data: two\n\n"

        async def handler(req):
            return Response(status=200, content_type="text/event-stream",
                            body_iter=gen())

        server, port = await _start(handler)
        reader, writer = await asyncio.open_connection("127.0.0.1", port)
        writer.write(b"GET /s HTTP/1.1\r\n\r\n")
        await writer.drain()
        head = await reader.readuntil(b"\r\n\r\n")
        assert b"transfer-encoding: chunked" in head.lower()
        chunks = []
        while True:
            size_line = await reader.readline()
            size = int(size_line.strip(), 16)
            if size == 0:
                await reader.readline()
                break
            data = await reader.readexactly(size)
            await reader.readline()
            chunks.append(data)
        assert b"".join(chunks) == b"data: one\n\ndata: two\n\n"
        writer.close()
        await server.stop()
    run(main())


def test_body_read_timeout():
    """Headers promising a body that never arrives: the server answers
    400 within the body deadline instead of pinning the connection."""
    import socket as _socket

    from resilient_llm_amd.gateway.http import HttpServer, Response

    async def handler(req):
        return Response.json_response({"ok": True})

    async def run():
        srv = HttpServer(handler, port=free_port())
        srv.body_timeout_s = 0.5
        await srv.start()
        try:
            reader, writer = await asyncio.open_connection("127.0.0.1", srv.port)
            writer.write(b"POST /chat/completions HTTP/1.1\r\n"
                         b"content-length: 100\r\n\r\n")   # body never sent
            await writer.drain()
            data = await asyncio.wait_for(reader.read(4096), timeout=5.0)
            assert b"400" in data and b"body read timeout" in data
            writer.close()
        finally:
            await srv.stop()

    asyncio.run(run())
