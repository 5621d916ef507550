"""resilient_llm_amd — MI355X-native resilient LLM serving framework.

A from-scratch rebuild of the capability surface of
``aws-samples/sample-resilient-llm-inference``: an
OpenAI-compatible gateway + router (simple-shuffle load balancing, RPM/TPM
token buckets, pre-call checks, fallback chains, cooldown health management)
in front of per-GPU inference workers running Llama-family models with
hand-written CDNA4 (gfx950) HIP kernels, RCCL-over-xGMI tensor parallelism
for 70B-class pools, and an in-process invocation ledger replacing
CloudWatch Logs Insights.

The reference delegates its routing engine to LiteLLM (reference
config/config.yaml:35-114) and model execution to Amazon Bedrock; here both
are native: "region" -> GPU, "account" -> GPU pool, CloudWatch -> per-GPU
ledger + rocprof counters.
"""

__version__ = "0.1.0"
