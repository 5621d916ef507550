"""Typed config loader for the reference ``config.yaml`` schema.

The reference loads its YAML raw via ``yaml.safe_load`` in five separate
scripts (reference src/demo_cris.py:22-24 et al.) and re-validates ad hoc
(demo_cris.py:41-80, bin/start-gateway.sh:20-29).  Here there is ONE typed
loader.  Schema kept from the reference:

- ``model_list[].{model_name, litellm_params.model, model_info.id, rpm, tpm}``
- ``router_settings.{routing_strategy, enable_pre_call_checks, allowed_fails,
  cooldown_time, fallbacks}``

Reinterpretation for MI355X (BASELINE.json north star):

- ``litellm_params.model: bedrock/<id>``  becomes  ``gpu/<device>/<model>``
  (a replica pinned to one GPU), ``pool/<pool>/<model>`` (a TP pool of
  GPUs), or ``stub/<name>`` (in-process CPU stub backend for tests and the
  plumbing demo).
- The reference's ``aws:`` namespace becomes ``cluster:`` — GPU pools
  replace AWS accounts, the gateway port replaces ``litellm.port`` (both
  are accepted).
- ``cris.model_id`` names the alias whose requests the scheduler spreads
  across every available GPU (cross-GPU inference = CRIS analogue).
"""

from __future__ import annotations

import dataclasses
import os
import re
from typing import Any, Optional

import yaml

_VALID_PORT = range(1024, 65536)
_ALIAS_RE = re.compile(r"^[A-Za-z0-9._:\-]+$")


class ConfigError(ValueError):
    """Raised when config.yaml fails validation."""


@dataclasses.dataclass
class Deployment:
    """One entry of ``model_list`` — an alias -> backend binding.

    Mirrors the reference deployment shape (reference config/config.yaml:36-100):
    several deployments may share ``model_name`` (that is what load
    balancing fans out over, config.yaml:45-61).
    """

    model_name: str                  # routing alias
    model: str                       # backend spec: gpu/<dev>/<model> | pool/<p>/<model> | stub/<name>
    model_id: str                    # model_info.id — deployment identity (X8)
    rpm: Optional[int] = None        # requests/minute bucket (X4)
    tpm: Optional[int] = None        # tokens/minute bucket (X4)
    weight: int = 1                  # shuffle weight (LiteLLM uses rpm when present)
    params: dict = dataclasses.field(default_factory=dict)  # backend kwargs

    @property
    def backend_kind(self) -> str:
        return self.model.split("/", 1)[0]

    @property
    def backend_target(self) -> str:
        """Device / pool / stub discriminator, e.g. ``0`` of ``gpu/0/llama-3-8b``."""
        parts = self.model.split("/")
        return parts[1] if len(parts) > 1 else ""

    @property
    def backend_model(self) -> str:
        parts = self.model.split("/", 2)
        return parts[2] if len(parts) > 2 else ""


@dataclasses.dataclass
class RouterSettings:
    """``router_settings`` namespace (reference config/config.yaml:106-114)."""

    routing_strategy: str = "simple-shuffle"
    enable_pre_call_checks: bool = True
    allowed_fails: int = 2
    cooldown_time: float = 15.0
    # alias -> ordered list of fallback aliases (X5)
    fallbacks: dict[str, list[str]] = dataclasses.field(default_factory=dict)
    # retries against OTHER deployments of the same alias before falling back
    num_retries: int = 1
    # chars-per-token for the pre-call TPM estimate: 4 matches BPE-ish
    # tokenizers (the reference's Bedrock models); set 1 for the byte
    # tokenizer so admission does not under-charge 4x (VERDICT weak #8)
    estimate_chars_per_token: int = 4


@dataclasses.dataclass
class PoolDef:
    """A named GPU pool — the account-sharding analogue (X11)."""

    name: str
    gpus: list[int]
    tensor_parallel: int = 1


@dataclasses.dataclass
class ClusterConfig:
    """``cluster:`` namespace — replaces the reference ``aws:`` block."""

    port: int = 4000
    host: str = "127.0.0.1"
    gpus: Optional[int] = None            # None -> autodetect
    pools: dict[str, PoolDef] = dataclasses.field(default_factory=dict)
    ledger_path: Optional[str] = None     # JSONL invocation log (X12)
    # decode-cadence SLO for single-GPU workers: AIMD-bounds the tokens
    # a mixed prefill step may take (engine target_step_ms); ignored by
    # TP pools (wall-clock tuning diverges under lockstep)
    target_step_ms: Optional[float] = None


@dataclasses.dataclass
class Config:
    cluster: ClusterConfig
    deployments: list[Deployment]
    router: RouterSettings
    cris_model: Optional[str] = None      # alias spread across all GPUs (X10)
    raw: dict = dataclasses.field(default_factory=dict)

    def deployments_for(self, alias: str) -> list[Deployment]:
        return [d for d in self.deployments if d.model_name == alias]

    @property
    def aliases(self) -> list[str]:
        seen: dict[str, None] = {}
        for d in self.deployments:
            seen.setdefault(d.model_name, None)
        return list(seen)


def _require(cond: bool, msg: str) -> None:
    if not cond:
        raise ConfigError(msg)


def _parse_deployment(entry: Any, idx: int) -> Deployment:
    _require(isinstance(entry, dict), f"model_list[{idx}] must be a mapping")
    name = entry.get("model_name")
    _require(isinstance(name, str) and bool(_ALIAS_RE.match(name)),
             f"model_list[{idx}].model_name invalid: {name!r}")
    lp = entry.get("litellm_params") or entry.get("params") or {}
    _require(isinstance(lp, dict), f"model_list[{idx}].litellm_params must be a mapping")
    model = lp.get("model")
    _require(isinstance(model, str) and model.count("/") >= 1,
             f"model_list[{idx}].litellm_params.model must look like "
             f"'gpu/<device>/<model>', 'pool/<pool>/<model>' or 'stub/<name>': {model!r}")
    kind = model.split("/", 1)[0]
    _require(kind in ("gpu", "pool", "stub", "bedrock"),
             f"model_list[{idx}]: unknown backend kind {kind!r}")
    info = entry.get("model_info") or {}
    model_id = str(info.get("id") or model)
    rpm = entry.get("rpm")
    tpm = entry.get("tpm")
    for label, v in (("rpm", rpm), ("tpm", tpm)):
        _require(v is None or (isinstance(v, int) and v > 0),
                 f"model_list[{idx}].{label} must be a positive int, got {v!r}")
    weight = entry.get("weight", rpm if isinstance(rpm, int) else 1)
    extra = {k: v for k, v in lp.items() if k != "model"}
    # validate the per-deployment engine options up front (config-fuzz
    # tier: a typo'd value must fail at load, not as a dead worker)
    if "kv_dtype" in extra:
        _require(extra["kv_dtype"] in ("bf16", "fp8"),
                 f"model_list[{idx}].litellm_params.kv_dtype must be "
                 f"bf16|fp8, got {extra['kv_dtype']!r}")
    if "quantization" in extra:
        _require(extra["quantization"] in ("fp8",),
                 f"model_list[{idx}].litellm_params.quantization must be "
                 f"fp8, got {extra['quantization']!r}")
    if "spec_lookup" in extra:
        _require(isinstance(extra["spec_lookup"], int)
                 and 0 <= extra["spec_lookup"] <= 16,
                 f"model_list[{idx}].litellm_params.spec_lookup must be "
                 f"an int in [0, 16], got {extra['spec_lookup']!r}")
    return Deployment(model_name=name, model=model, model_id=model_id,
                      rpm=rpm, tpm=tpm, weight=int(weight), params=extra)


def _parse_fallbacks(raw: Any) -> dict[str, list[str]]:
    """Reference shape: a list of single-key dicts (config.yaml:111-114)."""
    out: dict[str, list[str]] = {}
    if raw is None:
        return out
    _require(isinstance(raw, list), "router_settings.fallbacks must be a list")
    for i, item in enumerate(raw):
        _require(isinstance(item, dict) and len(item) == 1,
                 f"fallbacks[{i}] must be a single-key mapping alias -> [aliases]")
        ((alias, targets),) = item.items()
        _require(isinstance(targets, list) and all(isinstance(t, str) for t in targets),
                 f"fallbacks[{i}] targets must be a list of aliases")
        out[str(alias)] = [str(t) for t in targets]
    return out


def _parse_pools(raw: Any) -> dict[str, PoolDef]:
    out: dict[str, PoolDef] = {}
    if raw is None:
        return out
    _require(isinstance(raw, dict), "cluster.pools must be a mapping")
    for name, spec in raw.items():
        _require(isinstance(spec, dict) and isinstance(spec.get("gpus"), list),
                 f"cluster.pools.{name} must have a gpus list")
        gpus = [int(g) for g in spec["gpus"]]
        tp = int(spec.get("tensor_parallel", 1))
        _require(tp >= 1 and len(gpus) % tp == 0,
                 f"cluster.pools.{name}: len(gpus) must be a multiple of tensor_parallel")
        out[str(name)] = PoolDef(name=str(name), gpus=gpus, tensor_parallel=tp)
    return out


def load_config(path: str | os.PathLike[str] | None = None,
                data: Optional[dict] = None) -> Config:
    """Load and validate a config.  Pass ``data`` to load from a dict (tests)."""
    if data is None:
        _require(path is not None, "load_config needs a path or a data dict")
        with open(path, "r") as f:
            data = yaml.safe_load(f)
    _require(isinstance(data, dict), "config root must be a mapping")

    cluster_raw = data.get("cluster") or {}
    # accept the reference's `litellm: {port:}` spelling too (config.yaml:31-33)
    port = cluster_raw.get("port", (data.get("litellm") or {}).get("port", 4000))
    _require(isinstance(port, int) and port in _VALID_PORT,
             f"gateway port must be in [1024, 65535], got {port!r}")
    cluster = ClusterConfig(
        port=port,
        host=str(cluster_raw.get("host", "127.0.0.1")),
        gpus=cluster_raw.get("gpus"),
        pools=_parse_pools(cluster_raw.get("pools")),
        ledger_path=cluster_raw.get("ledger_path"),
        target_step_ms=cluster_raw.get("target_step_ms"),
    )

    model_list = data.get("model_list") or []
    _require(isinstance(model_list, list) and model_list,
             "model_list must be a non-empty list")
    deployments = [_parse_deployment(e, i) for i, e in enumerate(model_list)]

    rs_raw = data.get("router_settings") or {}
    router = RouterSettings(
        routing_strategy=str(rs_raw.get("routing_strategy", "simple-shuffle")),
        enable_pre_call_checks=bool(rs_raw.get("enable_pre_call_checks", True)),
        allowed_fails=int(rs_raw.get("allowed_fails", 2)),
        cooldown_time=float(rs_raw.get("cooldown_time", 15.0)),
        fallbacks=_parse_fallbacks(rs_raw.get("fallbacks")),
        num_retries=int(rs_raw.get("num_retries", 1)),
        estimate_chars_per_token=int(rs_raw.get("estimate_chars_per_token", 4)),
    )
    _require(router.estimate_chars_per_token >= 1,
             "router_settings.estimate_chars_per_token must be >= 1")
    _require(router.routing_strategy in ("simple-shuffle", "least-busy", "round-robin", "prefix-affinity"),
             f"unknown routing_strategy {router.routing_strategy!r}")

    cris_model = (data.get("cris") or {}).get("model_id")

    # every pool referenced by a deployment must exist
    for d in deployments:
        if d.backend_kind == "pool":
            _require(d.backend_target in cluster.pools,
                     f"deployment {d.model_name}: pool {d.backend_target!r} "
                     f"not defined under cluster.pools")
    # every fallback target must be a known alias
    known = {d.model_name for d in deployments}
    for alias, targets in router.fallbacks.items():
        for t in targets:
            _require(t in known, f"fallback target {t!r} (for {alias!r}) is not in model_list")

    return Config(cluster=cluster, deployments=deployments, router=router,
                  cris_model=cris_model, raw=data)
