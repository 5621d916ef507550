import pytest

from resilient_llm_amd.config import ConfigError, load_config


def base_cfg():
    return {
        "cluster": {"port": 4000, "pools": {
            "pool-a": {"gpus": [0, 1, 2, 3], "tensor_parallel": 4},
        }},
        "model_list": [
            {"model_name": "llama-fallback-demo",
             "litellm_params": {"model": "gpu/0/llama-3-8b"},
             "rpm": 3, "tpm": 100000},
            {"model_name": "llama-loadbalance-demo",
             "litellm_params": {"model": "gpu/0/llama-3-8b"},
             "model_info": {"id": "gpu0/llama-3-8b"},
             "rpm": 3, "tpm": 100000},
            {"model_name": "llama-loadbalance-demo",
             "litellm_params": {"model": "gpu/1/llama-3-8b"},
             "model_info": {"id": "gpu1/llama-3-8b"},
             "rpm": 3, "tpm": 100000},
            {"model_name": "llama-fallback-pool",
             "litellm_params": {"model": "pool/pool-a/llama-3-70b"},
             "rpm": 25, "tpm": 250000},
        ],
        "router_settings": {
            "routing_strategy": "simple-shuffle",
            "enable_pre_call_checks": True,
            "allowed_fails": 2,
            "cooldown_time": 15,
            "fallbacks": [{"llama-fallback-demo": ["llama-fallback-pool"]}],
        },
        "cris": {"model_id": "llama-cris-demo"},
    }


def test_load_valid():
    cfg = load_config(data=base_cfg())
    assert cfg.cluster.port == 4000
    assert len(cfg.deployments) == 4
    assert cfg.aliases == ["llama-fallback-demo", "llama-loadbalance-demo",
                           "llama-fallback-pool"]
    lb = cfg.deployments_for("llama-loadbalance-demo")
    assert [d.model_id for d in lb] == ["gpu0/llama-3-8b", "gpu1/llama-3-8b"]
    assert cfg.router.fallbacks == {"llama-fallback-demo": ["llama-fallback-pool"]}
    assert cfg.cris_model == "llama-cris-demo"


def test_backend_parsing():
    cfg = load_config(data=base_cfg())
    d = cfg.deployments[0]
    assert d.backend_kind == "gpu"
    assert d.backend_target == "0"
    assert d.backend_model == "llama-3-8b"
    p = cfg.deployments[3]
    assert p.backend_kind == "pool"
    assert p.backend_target == "pool-a"
    assert cfg.cluster.pools["pool-a"].tensor_parallel == 4


def test_weight_defaults_to_rpm():
    cfg = load_config(data=base_cfg())
    assert cfg.deployments[3].weight == 25


def test_reference_litellm_port_spelling():
    data = base_cfg()
    data["cluster"].pop("port")
    data["litellm"] = {"port": 4321}
    cfg = load_config(data=data)
    assert cfg.cluster.port == 4321


def test_rejects_bad_port():
    data = base_cfg()
    data["cluster"]["port"] = 80
    with pytest.raises(ConfigError):
        load_config(data=data)


def test_rejects_unknown_pool():
    data = base_cfg()
    data["model_list"][3]["litellm_params"]["model"] = "pool/nope/llama"
    with pytest.raises(ConfigError):
        load_config(data=data)


def test_rejects_unknown_fallback_target():
    data = base_cfg()
    data["router_settings"]["fallbacks"] = [{"llama-fallback-demo": ["missing"]}]
    with pytest.raises(ConfigError):
        load_config(data=data)


def test_rejects_empty_model_list():
    data = base_cfg()
    data["model_list"] = []
    with pytest.raises(ConfigError):
        load_config(data=data)


def test_rejects_bad_rpm():
    data = base_cfg()
    data["model_list"][0]["rpm"] = -1
    with pytest.raises(ConfigError):
        load_config(data=data)


def test_shipped_config_loads():
    import os
    path = os.path.join(os.path.dirname(__file__), "..", "config", "config.yaml")
    cfg = load_config(path)
    assert cfg.deployments
    assert cfg.router.routing_strategy == "simple-shuffle"


def test_cluster_target_step_ms_parsed():
    cfg = load_config(data={
        "cluster": {"port": 4101, "target_step_ms": 25.0},
        "model_list": [{"model_name": "m",
                        "litellm_params": {"model": "stub/0/llama-3-8b"}}],
    })
    assert cfg.cluster.target_step_ms == 25.0
    cfg2 = load_config(data={
        "cluster": {"port": 4101},
        "model_list": [{"model_name": "m",
                        "litellm_params": {"model": "stub/0/llama-3-8b"}}],
    })
    assert cfg2.cluster.target_step_ms is None


def test_70b_pool_config_loads():
    cfg = load_config("config/config.70b.yaml")
    assert set(cfg.cluster.pools) == {"pool-a", "pool-b"}
    assert cfg.cluster.pools["pool-a"].tensor_parallel == 4
    assert cfg.cluster.pools["pool-a"].gpus == [0, 1, 2, 3]
    pool_deps = [d for d in cfg.deployments if d.backend_kind == "pool"]
    assert len(pool_deps) == 4
    assert all(d.backend_model == "llama-3-70b" for d in pool_deps)
    assert len(cfg.deployments_for("llama-70b")) == 2
    assert cfg.router.fallbacks["llama-70b-pool-a"] == ["llama-70b-pool-b"]


def test_config_fuzz_always_config_error():
    """Malformed configs raise ConfigError (a ValueError) with a clear
    message — never KeyError/TypeError/AttributeError from deep inside."""
    from resilient_llm_amd.config import ConfigError

    bad = [
        None,
        [],
        {"model_list": None},
        {"model_list": []},
        {"model_list": "nope"},
        {"model_list": [None]},
        {"model_list": [{"litellm_params": {"model": "gpu/0/m"}}]},   # no name
        {"model_list": [{"model_name": 7,
                         "litellm_params": {"model": "gpu/0/m"}}]},
        {"model_list": [{"model_name": "m", "litellm_params": None}]},
        {"model_list": [{"model_name": "m",
                         "litellm_params": {"model": "nodelimiter"}}]},
        {"model_list": [{"model_name": "m",
                         "litellm_params": {"model": "weird/0/m"}}]},
        {"model_list": [{"model_name": "m",
                         "litellm_params": {"model": "gpu/0/x"},
                         "rpm": -3}]},
        {"cluster": {"port": 80},          # privileged port rejected
         "model_list": [{"model_name": "m",
                         "litellm_params": {"model": "stub/0/x"}}]},
        {"cluster": {"pools": {"p": {"gpus": "zero"}}},
         "model_list": [{"model_name": "m",
                         "litellm_params": {"model": "stub/0/x"}}]},
        {"cluster": {"pools": {"p": {"gpus": [0, 1], "tensor_parallel": 3}}},
         "model_list": [{"model_name": "m",
                         "litellm_params": {"model": "stub/0/x"}}]},
        {"model_list": [{"model_name": "m",
                         "litellm_params": {"model": "stub/0/x"}}],
         "router_settings": {"fallbacks": "broken"}},
        {"model_list": [{"model_name": "m",
                         "litellm_params": {"model": "stub/0/x"}}],
         "router_settings": {"fallbacks": [{"a": "not-a-list"}]}},
    ]
    for data in bad:
        with pytest.raises(ConfigError):
            load_config(data=data)


def test_engine_option_validation():
    import pytest as _pytest
    from resilient_llm_amd.config import ConfigError, load_config

    def cfg(**lp):
        return {"cluster": {"port": 4100},
                "model_list": [{"model_name": "m",
                                "litellm_params": {"model": "stub/0/tiny",
                                                   **lp}}],
                "router_settings": {}}

    load_config(data=cfg(kv_dtype="fp8", quantization="fp8", spec_lookup=4))
    for bad in (cfg(kv_dtype="int4"), cfg(quantization="awq"),
                cfg(spec_lookup="four"), cfg(spec_lookup=99)):
        with _pytest.raises(ConfigError):
            load_config(data=bad)
