"""Checkpoint load (safetensors) and the full CPU
end-to-end slice: gateway -> engine worker -> tiny model."""

import asyncio
import json
import os

import pytest
import torch

from resilient_llm_amd.engine.engine import CapacityExceeded
from resilient_llm_amd.models import LlamaForCausalLM, get_config


def test_safetensors_roundtrip(tmp_path):
    """Export a tiny model in HF-Llama naming, load it into a fresh
    model, verify identical forward outputs."""
    import safetensors.torch as st
    cfg = get_config("tiny")
    src = LlamaForCausalLM(cfg, device="cpu", dtype=torch.float32, seed=9)
    hf = {"model.embed_tokens.weight": src.params["embed"],
          "model.norm.weight": src.params["final_ln"],
          "lm_head.weight": src.params["lm_head"]}
    for i in range(cfg.n_layers):
        q, k, v = torch.split(src.params[f"l{i}.qkv"],
                              [cfg.q_size, cfg.kv_size, cfg.kv_size], dim=0)
        g, u = torch.split(src.params[f"l{i}.gate_up"],
                           [cfg.intermediate_size, cfg.intermediate_size], dim=0)
        hf.update({
            f"model.layers.{i}.self_attn.q_proj.weight": q.contiguous(),
            f"model.layers.{i}.self_attn.k_proj.weight": k.contiguous(),
            f"model.layers.{i}.self_attn.v_proj.weight": v.contiguous(),
            f"model.layers.{i}.self_attn.o_proj.weight": src.params[f"l{i}.o"],
            f"model.layers.{i}.mlp.gate_proj.weight": g.contiguous(),
            f"model.layers.{i}.mlp.up_proj.weight": u.contiguous(),
            f"model.layers.{i}.mlp.down_proj.weight": src.params[f"l{i}.down"],
            f"model.layers.{i}.input_layernorm.weight": src.params[f"l{i}.ln1"],
            f"model.layers.{i}.post_attention_layernorm.weight": src.params[f"l{i}.ln2"],
        })
    st.save_file(hf, os.path.join(tmp_path, "model.safetensors"))

    dst = LlamaForCausalLM(cfg, device="cpu", dtype=torch.float32, seed=123)
    n = dst.load_safetensors(str(tmp_path))
    assert n == 3 + 9 * cfg.n_layers

    from resilient_llm_amd.engine import LLMEngine, PagedKVCache, SamplingParams
    def toks(model):
        kv = PagedKVCache.for_model(cfg, 16, device="cpu")
        kv.k = kv.k.float(); kv.v = kv.v.float()
        e = LLMEngine(model, kv)
        e.add_request("a", list(range(8, 30)), SamplingParams(max_tokens=5))
        out = []
        while e.has_work():
            out += [o.token_id for o in e.step()]
        return out
    assert toks(src) == toks(dst)


def test_max_model_len_guard():
    from resilient_llm_amd.engine import LLMEngine, PagedKVCache, SamplingParams
    cfg = get_config("tiny")
    m = LlamaForCausalLM(cfg, device="cpu", dtype=torch.float32)
    kv = PagedKVCache.for_model(cfg, 4096, device="cpu")
    e = LLMEngine(m, kv, max_model_len=64)
    with pytest.raises(CapacityExceeded):
        e.add_request("a", list(range(60)), SamplingParams(max_tokens=10))


def test_gateway_with_engine_worker_cpu():
    """The minimum end-to-end slice with no GPU: OpenAI request through
    the gateway into a real engine worker (tiny model), plus streaming."""
    from tests.gateway_harness import free_port
    from resilient_llm_amd.client import OpenAIClient
    from resilient_llm_amd.config import load_config
    from resilient_llm_amd.gateway.app import GatewayApp
    from resilient_llm_amd.gateway.http import HttpServer
    from resilient_llm_amd.workers.base import WorkerRegistry
    from resilient_llm_amd.workers.engine_worker import EngineWorker
    import threading

    port = free_port()
    cfg = load_config(data={
        "cluster": {"port": port},
        "model_list": [
            {"model_name": "tiny-serve",
             "litellm_params": {"model": "gpu/0/tiny"},
             "model_info": {"id": "gpu0/tiny"}}],
        "router_settings": {"routing_strategy": "simple-shuffle"},
    })
    loop = asyncio.new_event_loop()
    ready = threading.Event()
    holder = {}

    async def main():
        holder["stop"] = asyncio.Event()
        registry = WorkerRegistry()
        worker = EngineWorker(device="cpu", model_name="tiny",
                              device_label="gpu:0", num_blocks=64)
        registry.register("gpu", "0", worker)
        app = GatewayApp(cfg, registry, health_interval_s=1.0)
        server = HttpServer(app.handle, port=port)
        await server.start()
        await app.start_background()
        ready.set()
        await holder["stop"].wait()
        await app.stop_background()
        await server.stop()
        await registry.close()

    th = threading.Thread(target=lambda: loop.run_until_complete(main()),
                          daemon=True)
    th.start()
    assert ready.wait(30)
    try:
        client = OpenAIClient(f"http://127.0.0.1:{port}")
        r = client.chat.completions.create(
            model="tiny-serve",
            messages=[{"role": "user", "content": "end to end"}],
            max_tokens=6)
        assert r.usage.completion_tokens == 6
        assert r.device_header == "gpu:0"
        stream = client.chat.completions.create(
            model="tiny-serve",
            messages=[{"role": "user", "content": "stream me"}],
            max_tokens=5, stream=True)
        text, model = stream.collect_text()
        assert model == "gpu0/tiny"
        d = client.distribution(by="device")
        assert d["distribution"] == {"gpu:0": 2}
    finally:
        loop.call_soon_threadsafe(holder["stop"].set)
        th.join(timeout=10)


def test_ledger_jsonl(tmp_path):
    from resilient_llm_amd.obs.ledger import InvocationLedger, InvocationRecord
    path = os.path.join(tmp_path, "ledger.jsonl")
    led = InvocationLedger(jsonl_path=path)
    led.record(InvocationRecord(ts=1.0, request_id="r1", alias="a",
                                model_id="m", device="gpu:0", consumer="c",
                                status="ok", completion_tokens=5))
    led.record(InvocationRecord(ts=2.0, request_id="r2", alias="a",
                                model_id="m", device="gpu:1", consumer="c",
                                status="throttled"))
    led.close()
    rows = [json.loads(l) for l in open(path)]
    assert len(rows) == 2 and rows[0]["device"] == "gpu:0"
    assert led.distribution(by="device") == {"gpu:0": 1}
    st = led.stats()
    assert st["ok"] == 1 and st["throttled"] == 1
