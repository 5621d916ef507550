"""Minimal OpenAI-compatible sync client for the demos and tests.

The reference demos use the ``openai`` SDK against the local gateway
(reference src/demo_fallback.py:30).  This client reproduces the slice of
that surface the demos exercise — ``client.chat.completions.create(...)``
returning an object with ``.model``, ``.choices[0].message.content``,
``.usage`` — plus the response headers (the reference reads
``x-litellm-model-id``; ours is ``x-gateway-model-id``,
demo_load_balancing.py:121-132 analogue) and raises :class:`RateLimitError`
on HTTP 429 exactly like the SDK does (demo_quota_isolation.py:95-107).

Uses stdlib ``http.client`` so the demos are thread-parallel with zero
extra dependencies.
"""

from __future__ import annotations

import http.client
import json
import urllib.parse
from typing import Iterator, Optional


class APIError(Exception):
    def __init__(self, status: int, message: str, body: Optional[dict] = None) -> None:
        super().__init__(f"HTTP {status}: {message}")
        self.status = status
        self.message = message
        self.body = body or {}


class RateLimitError(APIError):
    pass


class APIConnectionError(Exception):
    pass


class _Obj:
    """Attribute view over parsed JSON."""

    def __init__(self, data: dict) -> None:
        self._data = data

    def __getattr__(self, name: str):
        try:
            v = self._data[name]
        except KeyError:
            raise AttributeError(name) from None
        return _wrap(v)

    def get(self, name, default=None):
        return _wrap(self._data.get(name, default))

    def __repr__(self) -> str:
        return f"_Obj({self._data!r})"


def _wrap(v):
    if isinstance(v, dict):
        return _Obj(v)
    if isinstance(v, list):
        return [_wrap(x) for x in v]
    return v


class ChatCompletionResponse(_Obj):
    def __init__(self, data: dict, headers: dict[str, str]) -> None:
        super().__init__(data)
        self.headers = headers

    @property
    def model_id_header(self) -> Optional[str]:
        return self.headers.get("x-gateway-model-id")

    @property
    def device_header(self) -> Optional[str]:
        return self.headers.get("x-gateway-device")

    @property
    def was_fallback(self) -> bool:
        return self.headers.get("x-gateway-fallback") == "true"


class _Completions:
    def __init__(self, client: "OpenAIClient") -> None:
        self._client = client

    def create(self, *, model: str, messages: list, max_tokens: int = 128,
               temperature: float = 0.0, top_p: float = 1.0,
               stream: bool = False, timeout: Optional[float] = None,
               seed: Optional[int] = None,
               extra_headers: Optional[dict] = None, **kw):
        payload = {"model": model, "messages": messages, "max_tokens": max_tokens,
                   "temperature": temperature, "top_p": top_p, "stream": stream}
        if seed is not None:
            payload["seed"] = seed
        payload.update(kw)
        if stream:
            return self._client._post_stream("/chat/completions", payload,
                                             timeout, extra_headers)
        return self._client._post_json("/chat/completions", payload, timeout,
                                       extra_headers)


class _Chat:
    def __init__(self, client: "OpenAIClient") -> None:
        self.completions = _Completions(client)


class OpenAIClient:
    def __init__(self, base_url: str = "http://127.0.0.1:4000",
                 api_key: str = "anonymous", timeout: float = 60.0) -> None:
        parsed = urllib.parse.urlsplit(base_url)
        self.host = parsed.hostname or "127.0.0.1"
        self.port = parsed.port or 4000
        self.api_key = api_key
        self.timeout = timeout
        self.chat = _Chat(self)

    def _connect(self, timeout: Optional[float]) -> http.client.HTTPConnection:
        return http.client.HTTPConnection(self.host, self.port,
                                          timeout=timeout or self.timeout)

    def _headers(self) -> dict:
        return {"content-type": "application/json",
                "authorization": f"Bearer {self.api_key}"}

    @staticmethod
    def _raise_for(status: int, data: dict) -> None:
        msg = (data.get("error") or {}).get("message", "") if isinstance(data, dict) else ""
        if status == 429:
            raise RateLimitError(status, msg or "rate limited", data)
        raise APIError(status, msg or "request failed", data)

    def _post_json(self, path: str, payload: dict,
                   timeout: Optional[float],
                   extra_headers: Optional[dict] = None) -> ChatCompletionResponse:
        conn = self._connect(timeout)
        try:
            try:
                conn.request("POST", path, body=json.dumps(payload),
                             headers={**self._headers(), **(extra_headers or {})})
                resp = conn.getresponse()
                body = resp.read()
            except (ConnectionError, OSError) as e:
                raise APIConnectionError(str(e)) from e
            data = json.loads(body) if body else {}
            if resp.status >= 400:
                self._raise_for(resp.status, data)
            return ChatCompletionResponse(data, {k.lower(): v for k, v in resp.getheaders()})
        finally:
            conn.close()

    def _post_stream(self, path: str, payload: dict,
                     timeout: Optional[float],
                     extra_headers: Optional[dict] = None) -> "StreamResponse":
        conn = self._connect(timeout)
        try:
            conn.request("POST", path, body=json.dumps(payload),
                         headers={**self._headers(), **(extra_headers or {})})
            resp = conn.getresponse()
        except (ConnectionError, OSError) as e:
            conn.close()
            raise APIConnectionError(str(e)) from e
        if resp.status >= 400:
            body = resp.read()
            conn.close()
            self._raise_for(resp.status, json.loads(body) if body else {})
        return StreamResponse(conn, resp)

    # ------------------------------------------------------ admin helpers
    def _get(self, path: str, params: Optional[dict] = None) -> dict:
        if params:
            path = path + "?" + urllib.parse.urlencode(params)
        conn = self._connect(None)
        try:
            conn.request("GET", path, headers=self._headers())
            resp = conn.getresponse()
            data = json.loads(resp.read() or b"{}")
            if resp.status >= 400:
                self._raise_for(resp.status, data)
            return data
        finally:
            conn.close()

    def _post(self, path: str, payload: dict) -> dict:
        conn = self._connect(None)
        try:
            conn.request("POST", path, body=json.dumps(payload), headers=self._headers())
            resp = conn.getresponse()
            data = json.loads(resp.read() or b"{}")
            if resp.status >= 400:
                self._raise_for(resp.status, data)
            return data
        finally:
            conn.close()

    def distribution(self, by: str = "device", since_s: Optional[float] = None,
                     alias: Optional[str] = None, status: str = "ok") -> dict:
        params: dict = {"by": by, "status": status}
        if since_s is not None:
            params["since_s"] = since_s
        if alias is not None:
            params["alias"] = alias
        return self._get("/admin/distribution", params)

    def router_state(self) -> dict:
        return self._get("/admin/router")

    def health(self) -> dict:
        return self._get("/health")

    def inject_fault(self, device: str, mode: str) -> dict:
        return self._post("/admin/fault", {"device": device, "mode": mode})


class StreamResponse:
    """Iterator over SSE chunks; each item is the parsed event dict."""

    def __init__(self, conn: http.client.HTTPConnection,
                 resp: http.client.HTTPResponse) -> None:
        self._conn = conn
        self._resp = resp
        self.headers = {k.lower(): v for k, v in resp.getheaders()}
        self.comments: list[str] = []

    def __iter__(self) -> Iterator[dict]:
        buf = b""
        try:
            while True:
                chunk = self._resp.read1(65536)
                if not chunk:
                    break
                buf += chunk
                while b"\n\n" in buf:
                    event, buf = buf.split(b"\n\n", 1)
                    for line in event.split(b"\n"):
                        line = line.strip()
                        if line.startswith(b": "):
                            self.comments.append(line[2:].decode())
                        elif line.startswith(b"data: "):
                            data = line[6:]
                            if data == b"[DONE]":
                                return
                            yield json.loads(data)
        finally:
            self._conn.close()

    def collect_text(self) -> tuple[str, Optional[str]]:
        """Drain the stream; return (full_text, final_model_id)."""
        parts: list[str] = []
        model = None
        for evt in self:
            if "error" in evt:
                raise APIError(500, evt["error"].get("message", "stream error"), evt)
            model = evt.get("model", model)
            for c in evt.get("choices", []):
                content = (c.get("delta") or {}).get("content")
                if content:
                    parts.append(content)
        return "".join(parts), model
