"""The gateway application: OpenAI-compatible routes + dispatch loop.

This is the native equivalent of the LiteLLM proxy the reference launches
(reference bin/start-gateway.sh:56) — the full capability surface of
SURVEY.md §2.2 in one place:

- X1  POST /chat/completions (and /v1/...) with model alias, messages,
      max_tokens, temperature, stream; `response.model` carries the
      serving deployment id (demo_fallback.py:166 reads it).
- X8  response header ``x-gateway-model-id`` names the deployment
      (the reference reads ``x-litellm-model-id``, demo_load_balancing.py:132).
- X9  429 + OpenAI error body when every bucket/ladder is exhausted.
- X5/X6/X14  retry ladder onto fallback deployments, cooldowns, and SSE
      streaming with hot failover mid-stream (stateless replay: the
      replacement worker regenerates and the gateway discards the tokens
      the client already has — SURVEY.md §5.3).
- X10/X12  /admin/distribution — the Logs-Insights-style query over the
      in-process ledger (no propagation wait, same table content).
- Fault injection: POST /admin/fault {device, mode} (drives the failover
      demo where the reference could only starve quotas).
"""

from __future__ import annotations

import asyncio
import json
import time
import uuid
from typing import AsyncIterator, Optional

from ..config import Config
from ..obs.ledger import InvocationLedger, InvocationRecord
from ..router.core import (
    NoDeploymentAvailable, Router, RouterRateLimit, Ticket, UnknownAlias,
)
from ..router.token_bucket import MinuteWindowLimiter
from ..workers.base import (
    GenerationRequest, GenerationResult, Worker, WorkerDead, WorkerError,
    WorkerRegistry, WorkerMigrated, WorkerThrottled,
)

from .http import Request, Response

DEFAULT_MAX_TOKENS = 128
STREAM_CHUNK_TIMEOUT_S = 30.0
REQUEST_TIMEOUT_S = 120.0


class ConsumerLimiter:
    """Optional per-consumer (api-key) buckets on top of per-deployment
    ones — SURVEY.md §3.4 'MI355X equivalent: per-consumer token buckets
    keyed on api_key'.  Configured via a ``consumer_limits:`` section."""

    def __init__(self, raw: dict | None, clock=time.monotonic) -> None:
        raw = raw or {}
        self._default = raw.get("default")          # {"rpm":..,"tpm":..} | None
        self._per_key_cfg = raw.get("keys", {})
        self._clock = clock
        self._limiters: dict[str, MinuteWindowLimiter] = {}

    def _limiter_for(self, key: str) -> Optional[MinuteWindowLimiter]:
        cfg = self._per_key_cfg.get(key, self._default)
        if not cfg:
            return None
        if key not in self._limiters:
            self._limiters[key] = MinuteWindowLimiter(
                rpm=cfg.get("rpm"), tpm=cfg.get("tpm"), clock=self._clock)
        return self._limiters[key]

    def try_acquire(self, key: str, tokens: int) -> bool:
        lim = self._limiter_for(key)
        return True if lim is None else lim.try_acquire(tokens)

    def reconcile(self, key: str, estimated: int, actual: int) -> None:
        lim = self._limiter_for(key)
        if lim is not None:
            lim.reconcile(estimated, actual)


def _normalize_stop(stop) -> Optional[list]:
    """OpenAI `stop`: a string or up to 4 strings."""
    if stop is None:
        return None
    if isinstance(stop, str):
        stop = [stop]
    stop = [s for s in stop if isinstance(s, str) and s]
    return stop[:4] or None


class AccessPolicy:
    """API-key -> alias ACL — the least-privilege IAM policy analogue
    (reference iam/policy.json, C8 in SURVEY.md §2.1): which consumers
    may invoke which model aliases, and who may touch admin routes.
    Configured via an ``auth:`` section; absent -> allow everything
    (matching the reference demos' cosmetic api keys)."""

    def __init__(self, raw: dict | None) -> None:
        raw = raw or {}
        self.enforce = bool(raw.get("enforce", False))
        self.keys: dict = raw.get("keys", {})
        self.default = raw.get("default", {"allow": ["*"]})

    def _rule(self, key: str) -> dict:
        return self.keys.get(key, self.default)

    def allows(self, key: str, alias: str) -> bool:
        if not self.enforce:
            return True
        allow = self._rule(key).get("allow", [])
        return "*" in allow or alias in allow

    def is_admin(self, key: str) -> bool:
        if not self.enforce:
            return True
        return bool(self._rule(key).get("admin", False))


class GatewayApp:
    def __init__(self, config: Config, registry: WorkerRegistry,
                 router: Optional[Router] = None,
                 ledger: Optional[InvocationLedger] = None,
                 health_interval_s: float = 2.0,
                 shared_limits_path: Optional[str] = None) -> None:
        self.config = config
        self.registry = registry
        self.router = router or Router(config.deployments, config.router,
                                       shared_limits_path=shared_limits_path)
        self.ledger = ledger or InvocationLedger(
            jsonl_path=config.cluster.ledger_path)
        self.consumers = ConsumerLimiter(config.raw.get("consumer_limits"))
        self.policy = AccessPolicy(config.raw.get("auth"))
        self.health_interval_s = health_interval_s
        self.respawn_cooldown_s = 10.0
        self._restart_tasks: dict = {}
        self.migration_stats = {"migrated": 0, "failed": 0, "sweeps": 0,
                                "restarts": 0}
        self.last_health: dict = {}     # worker key -> last health dict
        self._health_task: Optional[asyncio.Task] = None
        self.started_at = time.time()

    # ---------------------------------------------------------- lifecycle
    async def start_background(self) -> None:
        if self._health_task is None:
            self._health_task = asyncio.create_task(self._health_loop())

    async def stop_background(self) -> None:
        if self._health_task is not None:
            self._health_task.cancel()
            try:
                await self._health_task
            except asyncio.CancelledError:
                pass
            self._health_task = None
        for t in list(self._restart_tasks.values()):
            if not t.done():
                t.cancel()
                try:
                    await t
                except asyncio.CancelledError:
                    pass
        self._restart_tasks.clear()

    def _worker_for(self, ticket: Ticket) -> Worker:
        d = ticket.deployment
        return self.registry.get(d.backend_kind, d.backend_target, d.backend_model)

    def _deployments_on(self, worker: Worker) -> list[str]:
        """model_ids of deployments that resolve to this worker (for
        health marking).  ``*`` targets are skipped — they re-resolve.
        The MODEL must match too: two
        different-model deployments co-located on one target must not
        share health flags."""
        kind, _, target = worker.device.partition(":")
        out = []
        for d in self.config.deployments:
            if d.backend_kind != kind or d.backend_target != target:
                continue
            if (d.backend_model or d.model_id) not in worker.models:
                continue
            out.append(d.model_id)
        return out

    async def _health_loop(self) -> None:
        """Per-worker heartbeat -> router health flags (X6) and elastic
        recovery: a dead worker PROCESS with a respawn hook is restarted
        (with a cooldown) and rejoins rotation once healthy
        (SURVEY.md §5.3)."""
        respawning: set = set()
        fail_streak: dict = {}
        while True:
            await asyncio.sleep(self.health_interval_s)
            for key, worker in self.registry.all().items():
                try:
                    # probe timeout is NOT the poll interval: a busy
                    # worker (GIL-heavy engine step, co-located load)
                    # can miss a sub-second deadline while being
                    # perfectly alive — r02 chaos soaks flapped healthy
                    # workers out of rotation exactly this way
                    h = await asyncio.wait_for(
                        worker.health(),
                        timeout=max(5.0, self.health_interval_s))
                    self.last_health[key] = h
                    ok = True
                    fail_streak[key] = 0
                except Exception as e:
                    streak = fail_streak.get(key, 0) + 1
                    fail_streak[key] = streak
                    proc = getattr(worker, "proc", None)
                    proc_dead = proc is not None and proc.poll() is not None
                    # an explicit dead signal (WorkerDead) or a dead
                    # process is immediate; a slow/errored probe with a
                    # LIVE process is a blip until it repeats
                    ok = (not (proc_dead or isinstance(e, WorkerDead))
                          and streak < 2)
                    if (streak >= 8 and proc is not None
                            and proc.poll() is None
                            and getattr(worker, "respawn", None) is not None):
                        # wedged-but-alive process (hung loop, stuck
                        # engine): unresponsive for ~8 probes straight —
                        # terminate it so the dead-proc respawn path
                        # below recovers it (r02: a hung worker sat
                        # unhealthy forever because respawn only fired
                        # for DEAD processes)
                        from ..utils.logging import log_with_timestamp
                        log_with_timestamp(
                            f"worker {key} unresponsive "
                            f"({streak} probes) — terminating for "
                            f"respawn", "red")
                        if streak >= 12:
                            # SIGTERM can't reach a process wedged in a
                            # C extension (Python only runs handlers at
                            # bytecode boundaries) — escalate
                            proc.kill()
                        else:
                            proc.terminate()
                if ok and getattr(worker, "proc_group", None):
                    # a dead FOLLOWER strands the collective while the
                    # leader still answers health RPCs: detect it, take
                    # the leader down too, and let whole-group respawn
                    # rebuild the pool on a fresh rendezvous
                    if any(p.poll() is not None for p in worker.proc_group):
                        ok = False
                        if worker.proc is not None and worker.proc.poll() is None:
                            worker.proc.terminate()
                for model_id in self._deployments_on(worker):
                    self.router.set_healthy(model_id, ok)
                if (not ok
                        and getattr(worker, "proc_group", None)
                        and worker.proc is not None
                        and worker.proc.poll() is not None):
                    # a dead TP-pool LEADER strands its followers in the
                    # collective: reap the whole group so GPUs free up
                    for pr in worker.proc_group[1:]:
                        if pr.poll() is None:
                            pr.terminate()
                if (not ok and key not in respawning
                        and getattr(worker, "respawn", None) is not None
                        and getattr(worker, "proc", None) is not None
                        and worker.proc.poll() is not None
                        and time.monotonic() - worker.last_respawn
                        > self.respawn_cooldown_s):
                    respawning.add(key)

                    async def _do(worker=worker, key=key):
                        from ..utils.logging import log_with_timestamp
                        try:
                            log_with_timestamp(
                                f"respawning dead worker {key}", "yellow")
                            await worker.respawn_now()
                            log_with_timestamp(
                                f"worker {key} back online", "green")
                        except Exception as e:
                            log_with_timestamp(
                                f"respawn of {key} failed: {e}", "red")
                        finally:
                            respawning.discard(key)

                    asyncio.ensure_future(_do())

    # ------------------------------------------------------------ routing
    async def handle(self, req: Request) -> Response:
        route = (req.method, req.path)
        if req.path in ("/chat/completions", "/v1/chat/completions"):
            if req.method != "POST":
                return Response.error(405, "use POST")
            return await self.chat_completions(req)
        if req.path in ("/completions", "/v1/completions"):
            if req.method != "POST":
                return Response.error(405, "use POST")
            return await self.text_completions(req)
        if route == ("GET", "/health"):
            return await self.health(req)
        if req.path in ("/models", "/v1/models") and req.method == "GET":
            return self.list_models()
        if req.path.startswith("/admin/") and not self.policy.is_admin(
                self._consumer_of(req)):
            return Response.error(403, "admin access denied",
                                  err_type="permission_error")
        if route == ("GET", "/admin/distribution"):
            return self.admin_distribution(req)
        if route == ("GET", "/admin/router"):
            return Response.json_response({"deployments": self.router.describe(),
                                           "settings": {
                                               "routing_strategy": self.config.router.routing_strategy,
                                               "enable_pre_call_checks": self.config.router.enable_pre_call_checks,
                                               "allowed_fails": self.config.router.allowed_fails,
                                               "cooldown_time": self.config.router.cooldown_time,
                                               "fallbacks": self.config.router.fallbacks,
                                           }})
        if route == ("GET", "/admin/requests"):
            n = min(int(req.query.get("n", "50")), 1000)
            return Response.json_response({"requests": self.ledger.recent(n)})
        if route == ("POST", "/admin/fault"):
            return await self.admin_fault(req)
        if route == ("POST", "/admin/drain"):
            return await self.admin_drain(req)
        if route == ("POST", "/admin/migrate"):
            return await self.admin_migrate(req)
        if route == ("POST", "/admin/restart"):
            return await self.admin_restart(req)
        if route == ("GET", "/metrics"):
            return self.prometheus_metrics()
        return Response.error(404, f"no route {req.method} {req.path}",
                              err_type="invalid_request_error")

    # ----------------------------------------------------- chat endpoint
    @staticmethod
    def _consumer_of(req: Request) -> str:
        auth = req.headers.get("authorization", "")
        if auth.lower().startswith("bearer "):
            return auth[7:].strip() or "anonymous"
        return req.headers.get("x-api-key", "anonymous")

    @staticmethod
    def _parse_chat_body(req: Request) -> dict:
        body = req.json()
        if not isinstance(body, dict):
            raise ValueError("body must be a JSON object")
        if not isinstance(body.get("model"), str):
            raise ValueError("'model' is required")
        msgs = body.get("messages")
        if not isinstance(msgs, list) or not msgs:
            raise ValueError("'messages' must be a non-empty list")
        # numeric fields must coerce NOW so a bad type is a 400, not a
        # 500 from deep inside the dispatch path (found by the API fuzz
        # test)
        for field, kind in (("max_tokens", int),
                            ("max_completion_tokens", int),
                            ("seed", int),
                            ("temperature", float), ("top_p", float),
                            ("presence_penalty", float),
                            ("frequency_penalty", float)):
            v = body.get(field)
            if v is None:
                continue
            if isinstance(v, bool) or not isinstance(v, (int, float)):
                raise ValueError(f"'{field}' must be a number")
            body[field] = kind(v)
        if body.get("max_tokens") is not None and body["max_tokens"] < 1:
            raise ValueError("'max_tokens' must be >= 1")
        stop = body.get("stop")
        if stop is not None and not isinstance(stop, str) and not (
                isinstance(stop, list)
                and all(isinstance(s, str) for s in stop)):
            raise ValueError("'stop' must be a string or list of strings")
        return body

    def _estimate(self, body: dict) -> tuple[int, int]:
        cpt = self.config.router.estimate_chars_per_token
        prompt_est = sum(max(1, len(str(m.get("content", ""))) // cpt)
                         for m in body["messages"])
        max_tokens = int(body.get("max_tokens")
                       or body.get("max_completion_tokens")
                       or DEFAULT_MAX_TOKENS)
        return prompt_est, prompt_est + max_tokens

    def _gen_request(self, body: dict, consumer: str, ticket: Ticket,
                     stream: bool, rid: Optional[str] = None) -> GenerationRequest:
        return GenerationRequest(
            request_id=rid or f"req-{uuid.uuid4().hex[:16]}",
            model=ticket.deployment.backend_model,
            messages=body["messages"],
            max_tokens=int(body.get("max_tokens")
                       or body.get("max_completion_tokens")
                       or DEFAULT_MAX_TOKENS),
            temperature=float(body.get("temperature", 0.0)),
            top_p=float(body.get("top_p", 1.0)),
            presence_penalty=float(body.get("presence_penalty", 0.0)),
            frequency_penalty=float(body.get("frequency_penalty", 0.0)),
            seed=body.get("seed"),
            stream=stream,
            consumer=consumer,
            stop=_normalize_stop(body.get("stop")),
        )

    def _record(self, ticket: Ticket, greq: Optional[GenerationRequest],
                consumer: str, status: str, t0: float,
                result: Optional[GenerationResult] = None,
                device: str = "") -> None:
        self.ledger.record(InvocationRecord(
            ts=time.time(),
            request_id=greq.request_id if greq else "-",
            alias=ticket.alias_requested,
            model_id=ticket.deployment.model_id,
            device=device,
            consumer=consumer,
            status=status,
            is_fallback=ticket.is_fallback,
            prompt_tokens=result.prompt_tokens if result else 0,
            completion_tokens=result.completion_tokens if result else 0,
            latency_ms=(time.monotonic() - t0) * 1000.0,
            ttft_ms=result.ttft_ms if result else None,
        ))

    async def chat_completions(self, req: Request) -> Response:
        try:
            body = self._parse_chat_body(req)
        except (ValueError, json.JSONDecodeError) as e:
            return Response.error(400, str(e), err_type="invalid_request_error")
        alias = body["model"]
        consumer = self._consumer_of(req)
        if not self.policy.allows(consumer, alias):
            return Response.error(403, f"api key not permitted for model "
                                       f"{alias!r}", err_type="permission_error",
                                  code="model_access_denied")
        prompt_est, total_est = self._estimate(body)
        if not self.consumers.try_acquire(consumer, total_est):
            return Response.error(429, f"consumer {consumer!r} rate limit exceeded",
                                  err_type="rate_limit_error", code="rate_limit_exceeded")
        # request-id propagation: honor the client's x-request-id (or
        # mint one); it tags the ledger rows and every response
        rid = req.headers.get("x-request-id") or f"req-{uuid.uuid4().hex[:16]}"
        # sampled-request seed is minted HERE, once per request (VERDICT
        # r01 #5): a mid-stream failover replays with the SAME seed, so
        # the replacement worker regenerates the identical prefix and the
        # discard-by-count skip concatenates coherently.  (Engine-side
        # default seeds are arrival-order-derived and differ per worker.)
        if body.get("seed") is None and float(body.get("temperature") or 0.0) > 0:
            import hashlib
            body["seed"] = int.from_bytes(
                hashlib.blake2b(rid.encode(), digest_size=8).digest(),
                "big") >> 1
        if bool(body.get("stream")):
            return await self._chat_stream(body, alias, consumer, total_est,
                                           rid)
        return await self._chat_once(body, alias, consumer, total_est, rid)

    async def text_completions(self, req: Request) -> Response:
        """Legacy /completions: the prompt is wrapped as a single user
        message and the chat result is re-shaped to text_completion.
        Streaming is chat-endpoint-only (the reference exercises only
        /chat/completions — SURVEY.md X1)."""
        try:
            body = req.json()
            if not isinstance(body, dict) or not isinstance(body.get("model"), str):
                raise ValueError("'model' is required")
            prompt = body.get("prompt")
            if isinstance(prompt, list):
                prompt = prompt[0] if prompt else ""
            if not isinstance(prompt, str):
                raise ValueError("'prompt' must be a string")
        except (ValueError, json.JSONDecodeError) as e:
            return Response.error(400, str(e), err_type="invalid_request_error")
        if body.get("stream"):
            return Response.error(
                400, "streaming is supported on /chat/completions",
                err_type="invalid_request_error")
        chat_body = dict(body)
        chat_body.pop("prompt", None)
        chat_body["messages"] = [{"role": "user", "content": prompt}]
        chat_req = Request(req.method, req.path, req.headers,
                           json.dumps(chat_body).encode(), req.http_version)
        resp = await self.chat_completions(chat_req)
        if resp.status != 200:
            return resp
        payload = json.loads(resp.body)
        text = payload["choices"][0]["message"]["content"]
        out = {
            "id": payload["id"].replace("chatcmpl-", "cmpl-", 1),
            "object": "text_completion",
            "created": payload["created"],
            "model": payload["model"],
            "choices": [{"index": 0, "text": text, "logprobs": None,
                         "finish_reason": payload["choices"][0]["finish_reason"]}],
            "usage": payload["usage"],
        }
        return Response.json_response(out, headers=resp.headers)

    # max attempts across replicas + fallback chain before giving up
    def _max_attempts(self, alias: str) -> int:
        n = len(self.router.alias_states(alias))
        for fb in self.config.router.fallbacks.get(alias, []):
            try:
                n += len(self.router.alias_states(fb))
            except UnknownAlias:
                pass
        return max(2, min(n + 1, 8))

    def _affinity_key(self, body: dict) -> Optional[str]:
        """Prompt-prefix key for ``prefix-affinity`` routing: the first
        512 chars of the serialized messages.  Requests sharing a system
        prompt / conversation head map to the same replica, whose engine
        prefix cache then serves the shared blocks (None for any other
        strategy — zero cost on the default paths)."""
        if self.router.settings.routing_strategy != "prefix-affinity":
            return None
        parts: list[str] = []
        n = 0
        for m in body.get("messages") or []:
            frag = f"{m.get('role', '')}\x1f{m.get('content', '')}\x1e"
            parts.append(frag)
            n += len(frag)
            if n >= 512:
                break
        return "".join(parts)[:512] or None

    async def _chat_once(self, body: dict, alias: str, consumer: str,
                         total_est: int, rid: Optional[str] = None) -> Response:
        t0 = time.monotonic()
        exclude: set = set()
        last_err: Optional[Exception] = None
        try:
            attempts = self._max_attempts(alias)
        except UnknownAlias:
            self.consumers.reconcile(consumer, total_est, 0)
            return Response.error(404, f"unknown model {alias!r}",
                                  err_type="invalid_request_error", code="model_not_found")
        for _ in range(attempts):
            try:
                ticket = self.router.acquire(alias, total_est, exclude=exclude,
                                             affinity_key=self._affinity_key(body))
            except RouterRateLimit as e:
                # once a worker has FAILED (non-throttle), an
                # exhausted router is a consequence of the failure, not a
                # rate limit — fall through to the 502 path below
                if last_err is not None and not isinstance(last_err, WorkerThrottled):
                    break
                self.consumers.reconcile(consumer, total_est, 0)
                return Response.error(429, str(e), err_type="rate_limit_error",
                                      code="rate_limit_exceeded")
            greq = self._gen_request(body, consumer, ticket, stream=False,
                                     rid=rid)
            try:
                worker = self._worker_for(ticket)
                result = await asyncio.wait_for(worker.generate(greq),
                                                timeout=REQUEST_TIMEOUT_S)
            except WorkerMigrated:
                # live migration: not a failure — release the ticket
                # without penalty and re-route; the adopting worker
                # continues token-exact (any other replica regenerates
                # identically from the sticky seed)
                self.router.complete(ticket, actual_tokens=0)
                exclude.add(id(ticket.state))
                continue
            except (WorkerError, asyncio.TimeoutError) as e:
                last_err = e
                throttled = isinstance(e, WorkerThrottled)
                self.router.fail(ticket, charge=throttled)
                self._record(ticket, greq, consumer,
                             "throttled" if throttled else "error", t0)
                exclude.add(id(ticket.state))
                continue
            actual = result.prompt_tokens + result.completion_tokens
            self.router.complete(ticket, actual_tokens=actual)
            self.consumers.reconcile(consumer, total_est, actual)
            self._record(ticket, greq, consumer, "ok", t0, result, worker.device)
            return self._completion_response(greq, ticket, result, worker)
        from ..utils.logging import sanitize_error
        # no backend work happened: release the consumer's token charge
        # (failed attempts over-throttled clients for a full
        # minute window during backend outages)
        self.consumers.reconcile(consumer, total_est, 0)
        msg = sanitize_error(last_err) if last_err else "no deployment available"
        status = 429 if isinstance(last_err, WorkerThrottled) else 502
        return Response.error(status, f"all deployments failed: {msg}",
                              err_type="api_error")

    def _completion_response(self, greq: GenerationRequest, ticket: Ticket,
                             result: GenerationResult, worker: Worker) -> Response:
        payload = {
            "id": f"chatcmpl-{greq.request_id}",
            "object": "chat.completion",
            "created": int(time.time()),
            "model": ticket.deployment.model_id,
            "choices": [{
                "index": 0,
                "message": {"role": "assistant", "content": result.text},
                "finish_reason": result.finish_reason,
            }],
            "usage": {
                "prompt_tokens": result.prompt_tokens,
                "completion_tokens": result.completion_tokens,
                "total_tokens": result.prompt_tokens + result.completion_tokens,
            },
        }
        headers = {
            "x-gateway-model-id": ticket.deployment.model_id,
            "x-gateway-device": worker.device,
            "x-gateway-fallback": "true" if ticket.is_fallback else "false",
            "x-request-id": greq.request_id,
        }
        return Response.json_response(payload, headers=headers)

    # -------------------------------------------------------- streaming
    async def _chat_stream(self, body: dict, alias: str, consumer: str,
                           total_est: int, rid: Optional[str] = None) -> Response:
        """SSE streaming (X14) with hot failover: a mid-stream worker
        death/hang re-routes to the next deployment, which regenerates;
        tokens the client already received are discarded (stateless
        replay)."""
        try:
            first_ticket = self.router.acquire(
                alias, total_est, affinity_key=self._affinity_key(body))
        except UnknownAlias:
            self.consumers.reconcile(consumer, total_est, 0)
            return Response.error(404, f"unknown model {alias!r}",
                                  err_type="invalid_request_error", code="model_not_found")
        except RouterRateLimit as e:
            self.consumers.reconcile(consumer, total_est, 0)
            return Response.error(429, str(e), err_type="rate_limit_error",
                                  code="rate_limit_exceeded")

        app = self

        include_usage = bool((body.get("stream_options") or {})
                             .get("include_usage"))

        prompt_est, _ = self._estimate(body)

        async def body_iter() -> AsyncIterator[bytes]:
            t0 = time.monotonic()
            sent_tokens = 0
            prompt_toks = prompt_est       # refined by the first chunk
            exclude: set = set()
            ticket: Optional[Ticket] = first_ticket
            last_err: Optional[Exception] = None
            attempts = app._max_attempts(alias)
            comp_id = f"chatcmpl-{rid or uuid.uuid4().hex[:16]}"
            try:
                for attempt in range(attempts):
                    if ticket is None:
                        try:
                            ticket = app.router.acquire(
                                alias, total_est, exclude=exclude,
                                affinity_key=app._affinity_key(body))
                        except NoDeploymentAvailable as e:
                            last_err = e
                            break
                    greq = app._gen_request(body, consumer, ticket,
                                            stream=True, rid=rid)
                    try:
                        worker = app._worker_for(ticket)
                    except WorkerError as e:
                        last_err = e
                        app.router.fail(ticket)
                        exclude.add(id(ticket.state))
                        ticket = None
                        continue
                    model_id = ticket.deployment.model_id
                    n_emitted = 0
                    completion_tokens = 0
                    finish = None
                    try:
                        it = worker.generate_stream(greq).__aiter__()
                        if attempt > 0:
                            yield (f": failover -> {model_id}\n\n").encode()
                        while True:
                            try:
                                chunk = await asyncio.wait_for(
                                    it.__anext__(), timeout=STREAM_CHUNK_TIMEOUT_S)
                            except StopAsyncIteration:
                                break
                            completion_tokens += 1
                            if chunk.prompt_tokens is not None:
                                prompt_toks = chunk.prompt_tokens
                            if completion_tokens <= sent_tokens:
                                finish = chunk.finish_reason or finish
                                continue   # replay skip after failover
                            n_emitted += 1
                            finish = chunk.finish_reason or finish
                            delta = ({"role": "assistant", "content": chunk.text}
                                     if sent_tokens + n_emitted == 1
                                     else {"content": chunk.text})
                            evt = {
                                "id": comp_id, "object": "chat.completion.chunk",
                                "created": int(time.time()), "model": model_id,
                                "choices": [{"index": 0, "delta": delta,
                                             "finish_reason": chunk.finish_reason}],
                            }
                            yield f"data: {json.dumps(evt)}\n\n".encode()
                        sent_tokens += n_emitted
                        result = GenerationResult(
                            text="", prompt_tokens=prompt_toks,
                            completion_tokens=sent_tokens,
                            finish_reason=finish or "stop")
                        actual = prompt_toks + sent_tokens
                        app.router.complete(ticket, actual_tokens=actual)
                        app.consumers.reconcile(consumer, total_est, actual)
                        app._record(ticket, greq, consumer, "ok", t0, result, worker.device)
                        if include_usage:
                            # OpenAI stream_options.include_usage: one
                            # final chunk with empty choices + usage
                            usage_evt = {
                                "id": comp_id,
                                "object": "chat.completion.chunk",
                                "created": int(time.time()),
                                "model": model_id, "choices": [],
                                "usage": {
                                    "prompt_tokens": result.prompt_tokens,
                                    "completion_tokens": sent_tokens,
                                    "total_tokens":
                                        result.prompt_tokens + sent_tokens,
                                },
                            }
                            yield f"data: {json.dumps(usage_evt)}\n\n".encode()
                        yield b"data: [DONE]\n\n"
                        return
                    except WorkerMigrated:
                        # live migration mid-stream: no failure penalty;
                        # re-route — the adopting worker replays the
                        # already-generated prefix (attach) and the
                        # emitted-token skip below dedupes it for the
                        # client, then continues token-exact
                        sent_tokens += n_emitted
                        app.router.complete(ticket, actual_tokens=0)
                        exclude.add(id(ticket.state))
                        ticket = None
                        try:
                            await asyncio.wait_for(it.aclose(), timeout=2.0)
                        except Exception:
                            pass
                        continue
                    except (WorkerError, asyncio.TimeoutError) as e:
                        last_err = e
                        sent_tokens += n_emitted
                        throttled = isinstance(e, WorkerThrottled)
                        app.router.fail(ticket, charge=throttled)
                        app._record(ticket, greq, consumer,
                                    "throttled" if throttled else "error", t0)
                        exclude.add(id(ticket.state))
                        ticket = None
                        # release the failed worker's stream NOW (its
                        # cleanup decrements in-flight / aborts the
                        # engine request) rather than at GC
                        try:
                            await asyncio.wait_for(it.aclose(), timeout=2.0)
                        except Exception:
                            pass
                        continue
                from ..utils.logging import sanitize_error
                # settle the consumer charge to what was actually served
                app.consumers.reconcile(
                    consumer, total_est,
                    prompt_toks + sent_tokens if sent_tokens else 0)
                err_evt = {"error": {"message": f"stream failed: "
                                                f"{sanitize_error(last_err) if last_err else 'exhausted'}",
                                     "type": "api_error"}}
                yield f"data: {json.dumps(err_evt)}\n\n".encode()
                yield b"data: [DONE]\n\n"
            finally:
                # a client disconnect closes this generator at a yield:
                # settle the open ticket so in-flight counts and the
                # ledger stay truthful (X12)
                if ticket is not None and not ticket.done:
                    app.router.complete(ticket, actual_tokens=None)
                    app.consumers.reconcile(
                        consumer, total_est,
                        prompt_toks + sent_tokens if sent_tokens else 0)
                    app.ledger.record(InvocationRecord(
                        ts=time.time(), request_id=comp_id, alias=alias,
                        model_id=ticket.deployment.model_id, device="",
                        consumer=consumer, status="cancelled",
                        is_fallback=ticket.is_fallback,
                        completion_tokens=sent_tokens,
                        latency_ms=(time.monotonic() - t0) * 1000.0))

        headers = {
            "x-gateway-model-id": first_ticket.deployment.model_id,
            "x-gateway-fallback": "true" if first_ticket.is_fallback else "false",
            "cache-control": "no-cache",
            "x-request-id": rid or "",
        }
        return Response(status=200, headers=headers,
                        content_type="text/event-stream", body_iter=body_iter())

    # ------------------------------------------------------------- admin
    async def health(self, req: Request) -> Response:
        workers = {}
        for key, w in self.registry.all().items():
            try:
                workers[key] = await w.health()
            except Exception as e:
                workers[key] = {"device": key, "status": "unhealthy", "error": str(e)}
        return Response.json_response({
            "status": "ok",
            "uptime_s": time.time() - self.started_at,
            "workers": workers,
        })

    def list_models(self) -> Response:
        return Response.json_response({
            "object": "list",
            "data": [{"id": a, "object": "model", "owned_by": "resilient_llm_amd"}
                     for a in self.config.aliases],
        })

    def admin_distribution(self, req: Request) -> Response:
        """The Logs-Insights-style aggregation (X12): counts by device /
        model_id / consumer / alias plus serving stats — synchronously,
        no propagation retry ladder needed."""
        by = req.query.get("by", "device")
        if by not in ("device", "model_id", "alias", "consumer", "status"):
            return Response.error(400, f"bad 'by' dimension {by!r}")
        since = None
        if "since_s" in req.query:
            since = time.time() - float(req.query["since_s"])
        alias = req.query.get("alias")
        status = req.query.get("status", "ok") or None
        dist = self.ledger.distribution(by=by, since=since, alias=alias, status=status)
        total = sum(dist.values())
        return Response.json_response({
            "by": by, "alias": alias, "total": total,
            "distribution": dist,
            "percentages": {k: round(100.0 * v / total, 1) for k, v in dist.items()}
            if total else {},
            "stats": self.ledger.stats(since=since, alias=alias),
        })

    async def admin_fault(self, req: Request) -> Response:
        try:
            body = req.json() or {}
            device = body["device"]
            mode = body.get("mode", "none")
        except (KeyError, json.JSONDecodeError):
            return Response.error(400, "need {device, mode}")
        workers = self.registry.all()
        if device not in workers:
            return Response.error(404, f"no worker {device!r} "
                                       f"(have {sorted(workers)})")
        try:
            await workers[device].inject_fault(mode)
        except WorkerError as e:
            return Response.error(409, f"worker {device} cannot take fault "
                                       f"command: {e}")
        return Response.json_response({"device": device, "mode": mode})

    # -------------------------------------------- drain / rolling restart
    def _set_drain(self, worker: Worker, draining: bool) -> None:
        worker.draining = draining          # spread (`*`) targets consult this
        for model_id in self._deployments_on(worker):
            self.router.set_draining(model_id, draining)

    def _worker_in_flight(self, worker: Worker) -> int:
        n = getattr(worker, "in_flight", 0)
        for d in self.config.deployments:
            st = self.router.state_for_id(d.model_id)
            kind, _, target = worker.device.partition(":")
            if (st is not None and d.backend_kind == kind
                    and d.backend_target == target):
                n = max(n, st.in_flight)
        return n

    async def admin_drain(self, req: Request) -> Response:
        """POST {worker, drain=true|false}: stop routing NEW requests to
        a worker; in-flight requests finish normally.  The building
        block of zero-downtime maintenance (see /admin/restart)."""
        try:
            body = req.json() or {}
            device = body["worker"]
        except (KeyError, json.JSONDecodeError):
            return Response.error(400, "need {worker, drain?}")
        worker = self.registry.all().get(device)
        if worker is None:
            return Response.error(404, f"no worker {device!r}")
        draining = bool(body.get("drain", True))
        t_drain0 = time.monotonic()
        self._set_drain(worker, draining)
        migrated: list = []
        errors: list = []
        target_dev = body.get("migrate_to")
        if draining and target_dev:
            # evacuate live requests to the target: zero-recompute
            # drain, BOUNDED by timeout_s — a slow/contended worker must
            # never hang the admin call (requests left behind simply
            # finish in place, which a drain tolerates by definition)
            target = self.registry.all().get(target_dev)
            if target is None:
                return Response.error(404, f"no worker {target_dev!r}")
            deadline = time.monotonic() + float(body.get("timeout_s", 30.0))
            # evacuating to a dead/booting target would burn the whole
            # deadline on doomed RPCs while the source sits draining —
            # leave the requests in place instead (a drain tolerates
            # that by definition)
            target_ok = True
            try:
                await asyncio.wait_for(target.health(), timeout=5.0)
            except Exception as e:                        # noqa: BLE001
                target_ok = False
                errors.append(f"target {target_dev} unhealthy, evacuation "
                              f"skipped: {e!r:.120}")
            ids = []
            if target_ok:
                try:
                    ids = await worker.list_requests()
                except (WorkerError, AttributeError,
                        asyncio.TimeoutError) as e:
                    errors.append(str(e) or "list_requests timed out")

            def _left() -> float:
                # every awaited migrate op is clamped to the deadline:
                # ONE hung RPC must not blow past timeout_s (r02 GPU
                # soak: migrate_in spinning in reconnect held the drain
                # >60 s while clients saw "no deployment")
                return max(0.5, deadline - time.monotonic())
            for req_id in ids:
                if time.monotonic() > deadline:
                    errors.append(f"evacuation deadline: "
                                  f"{len(ids) - len(migrated)} left in place")
                    break
                try:
                    blob = await asyncio.wait_for(
                        worker.migrate_out(req_id), timeout=_left())
                    try:
                        await asyncio.wait_for(
                            target.migrate_in(blob), timeout=_left())
                        migrated.append(req_id)
                        self.migration_stats["migrated"] += 1
                    finally:
                        # release the blocked client only once the
                        # state landed (or demonstrably failed): the
                        # re-routed attach then always finds it
                        await asyncio.wait_for(
                            worker.release_migrated(req_id),
                            timeout=max(2.0, _left()))
                except (WorkerError, asyncio.TimeoutError) as e:
                    errors.append(f"{req_id}: {e!r}")
                    self.migration_stats["failed"] += 1
                    # best-effort release: a migrate_out cancelled
                    # mid-extraction leaves the client blocked on a
                    # request whose state already left the engine —
                    # releasing re-routes it NOW (it regenerates from
                    # its sticky seed) instead of hanging to timeout
                    try:
                        await asyncio.wait_for(
                            worker.release_migrated(req_id), timeout=5.0)
                    except Exception:                 # noqa: BLE001
                        pass
            self.migration_stats["sweeps"] += 1
        return Response.json_response({
            "worker": device, "draining": draining,
            "migrated": migrated, "migrate_errors": errors,
            "elapsed_s": round(time.monotonic() - t_drain0, 3),
            "in_flight": self._worker_in_flight(worker)})

    async def admin_migrate(self, req: Request) -> Response:
        """POST {request_id, from, to}: live-migrate one in-flight
        request between workers — tokens + sampling identity + KV blocks
        move, generation continues token-exact on the target, and the
        client's blocked/streaming call transparently re-routes (the
        source raises WorkerMigrated into the gateway retry path)."""
        try:
            body = req.json() or {}
            request_id = body["request_id"]
            src_dev, dst_dev = body["from"], body["to"]
        except (KeyError, json.JSONDecodeError):
            return Response.error(400, "need {request_id, from, to}")
        workers = self.registry.all()
        src = workers.get(src_dev)
        dst = workers.get(dst_dev)
        if src is None or dst is None:
            return Response.error(404, f"unknown worker in {src_dev!r} -> "
                                       f"{dst_dev!r} (have {sorted(workers)})")
        try:
            blob = await src.migrate_out(request_id)
            try:
                await dst.migrate_in(blob)
            finally:
                await src.release_migrated(request_id)
        except (WorkerError, asyncio.TimeoutError) as e:
            return Response.error(409, f"migration failed: {e!r}")
        return Response.json_response({"request_id": request_id,
                                       "from": src_dev, "to": dst_dev,
                                       "state_bytes": len(blob)})

    async def admin_restart(self, req: Request) -> Response:
        """POST {worker, timeout_s?}: zero-downtime rolling restart —
        drain, wait for in-flight to land, terminate + respawn the
        worker process, wait healthy, undrain.  Other replicas keep
        serving throughout; returns immediately with 202."""
        try:
            body = req.json() or {}
            device = body["worker"]
        except (KeyError, json.JSONDecodeError):
            return Response.error(400, "need {worker, timeout_s?}")
        worker = self.registry.all().get(device)
        if worker is None:
            return Response.error(404, f"no worker {device!r}")
        if getattr(worker, "respawn", None) is None:
            return Response.error(409, f"worker {device} is not "
                                       "process-backed (no respawn hook)")
        timeout_s = float(body.get("timeout_s", 120.0))
        migrate_to = body.get("migrate_to")
        if migrate_to and migrate_to not in self.registry.all():
            return Response.error(404, f"no worker {migrate_to!r}")
        self.migration_stats["restarts"] += 1
        task = asyncio.get_running_loop().create_task(
            self._rolling_restart(device, worker, timeout_s, migrate_to))
        self._restart_tasks[device] = task
        return Response.json_response({"worker": device,
                                       "status": "restarting",
                                       "migrate_to": migrate_to}, status=202)

    async def _rolling_restart(self, device: str, worker: Worker,
                               timeout_s: float,
                               migrate_to: Optional[str] = None) -> None:
        from ..utils.logging import log_with_timestamp, sanitize_error
        log_with_timestamp(f"rolling restart of {device}: draining", "yellow")
        self._set_drain(worker, True)
        t0 = time.monotonic()
        try:
            if migrate_to:
                # evacuate instead of waiting for long generations:
                # zero-recompute restart (failures fall back to the
                # in-flight wait below, then stateless replay)
                target = self.registry.all().get(migrate_to)
                try:
                    ids = await worker.list_requests()
                except Exception:                      # noqa: BLE001
                    ids = []
                for req_id in ids:
                    if time.monotonic() - t0 > timeout_s / 2:
                        break
                    try:
                        blob = await worker.migrate_out(req_id)
                        try:
                            await target.migrate_in(blob)
                        finally:
                            await worker.release_migrated(req_id)
                    except (WorkerError, asyncio.TimeoutError):
                        pass
            while (self._worker_in_flight(worker) > 0
                   and time.monotonic() - t0 < timeout_s):
                await asyncio.sleep(0.1)
            proc = getattr(worker, "proc", None)
            if proc is not None and proc.poll() is None:
                proc.terminate()
                for _ in range(100):
                    if proc.poll() is not None:
                        break
                    await asyncio.sleep(0.1)
                if proc.poll() is None:
                    proc.kill()
            await worker.respawn_now()
            while time.monotonic() - t0 < timeout_s:
                try:
                    h = await asyncio.wait_for(worker.health(), timeout=5.0)
                    self.last_health[worker.device] = h
                    break
                except Exception:
                    await asyncio.sleep(0.2)
            for model_id in self._deployments_on(worker):
                self.router.set_healthy(model_id, True)
            log_with_timestamp(f"rolling restart of {device}: back online",
                               "green")
        except Exception as e:                            # noqa: BLE001
            log_with_timestamp(f"rolling restart of {device} FAILED: "
                               f"{sanitize_error(e)}", "red")
        finally:
            self._set_drain(worker, False)

    def prometheus_metrics(self) -> Response:
        lines = []
        for row in self.router.describe():
            labels = (f'alias="{row["model_name"]}",model_id="{row["model_id"]}"')
            lines.append(f'gateway_requests_total{{{labels}}} {row["total_requests"]}')
            lines.append(f'gateway_failures_total{{{labels}}} {row["total_failures"]}')
            lines.append(f'gateway_in_flight{{{labels}}} {row["in_flight"]}')
            lines.append(f'gateway_rpm_used{{{labels}}} {row["rpm_used"]}')
            lines.append(f'gateway_healthy{{{labels}}} {int(row["healthy"])}')
            lines.append(f'gateway_draining{{{labels}}} {int(row["draining"])}')
        st = self.ledger.stats()
        for k in ("total", "ok", "errors", "throttled", "fallbacks"):
            lines.append(f'gateway_ledger_{k} {st[k]}')
        for k, v in self.migration_stats.items():
            lines.append(f'gateway_migration_{k} {v}')
        lines.append(f'gateway_last_resort_routes_total '
                     f'{self.router.last_resort_total}')
        for key, h in self.last_health.items():
            wl = f'worker="{key}"'
            for field in ("in_flight", "queued", "running",
                          "kv_free_blocks", "total_served"):
                if field in h:
                    lines.append(f'worker_{field}{{{wl}}} {h[field]}')
            pc = h.get("prefix_cache")
            if pc:
                lines.append(f'worker_prefix_hits{{{wl}}} {pc["hits"]}')
                lines.append(f'worker_prefix_lookups{{{wl}}} {pc["lookups"]}')
        return Response(body="\n".join(lines) + "\n",
                        content_type="text/plain; version=0.0.4")
