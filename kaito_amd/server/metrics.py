"""Prometheus metrics with the engine metric names the KAITO ecosystem
scrapes (SURVEY.md §8 contract appendix):

  vllm:cache_config_info          — block pool size (benchmark probe reads
                                    num_gpu_blocks to size saturation
                                    concurrency, benchmark_entrypoint.py)
  vllm:generation_tokens_total    — TPM computation (probe + KEDA)
  vllm:prompt_tokens_total
  vllm:num_requests_running / waiting — EPP load-aware routing + 429 guard
  kaito_ratelimit_rejected_total  — rate_limit parity
  kaito_model_download_* — download progress gauges
"""
from __future__ import annotations

from prometheus_client import (CollectorRegistry, Counter, Gauge, Histogram,
                               generate_latest, CONTENT_TYPE_LATEST)

REGISTRY = CollectorRegistry()

CACHE_CONFIG_INFO = Gauge(
    "vllm:cache_config_info",
    "KV cache config (labels carry the config; value is 1)",
    ["block_size", "num_gpu_blocks", "num_cpu_blocks"],
    registry=REGISTRY)

GENERATION_TOKENS = Counter(
    "vllm:generation_tokens", "Total generated tokens", registry=REGISTRY)
PROMPT_TOKENS = Counter(
    "vllm:prompt_tokens", "Total prompt tokens processed", registry=REGISTRY)
REQUESTS_RUNNING = Gauge(
    "vllm:num_requests_running", "Sequences currently decoding",
    registry=REGISTRY)
REQUESTS_WAITING = Gauge(
    "vllm:num_requests_waiting", "Sequences queued for prefill",
    registry=REGISTRY)
GPU_CACHE_USAGE = Gauge(
    "vllm:gpu_cache_usage_perc", "Fraction of KV blocks in use (EPP "
    "kv-cache-utilization scorer input)", registry=REGISTRY)
PREFIX_CACHE_HIT_TOKENS = Gauge(
    "vllm:prefix_cache_hit_tokens", "Prompt tokens served from the "
    "block-hash prefix cache (prefill skipped)", registry=REGISTRY)
TTFT = Histogram(
    "vllm:time_to_first_token_seconds", "Request TTFT",
    buckets=(0.05, 0.1, 0.25, 0.5, 1, 2, 5, 10, 30, 60),
    registry=REGISTRY)
TPOT = Histogram(
    "vllm:time_per_output_token_seconds", "Mean inter-token latency "
    "per request",
    buckets=(0.005, 0.01, 0.02, 0.04, 0.08, 0.15, 0.3, 0.6),
    registry=REGISTRY)
E2E_LATENCY = Counter(
    "vllm:e2e_request_latency_seconds", "Sum of request latencies",
    registry=REGISTRY)

RATELIMIT_REJECTED = Counter(
    "kaito_ratelimit_rejected", "Requests rejected with 429",
    registry=REGISTRY)

MODEL_DOWNLOAD_PROGRESS = Gauge(
    "kaito_model_download_progress", "Weight download progress 0-1",
    registry=REGISTRY)
MODEL_DOWNLOAD_DONE = Gauge(
    "kaito_model_download_completed", "1 when weights are loaded",
    registry=REGISTRY)


def set_cache_config(block_size: int, num_gpu_blocks: int,
                     num_cpu_blocks: int = 0) -> None:
    CACHE_CONFIG_INFO.labels(
        block_size=str(block_size),
        num_gpu_blocks=str(num_gpu_blocks),
        num_cpu_blocks=str(num_cpu_blocks)).set(1)


def render() -> tuple[bytes, str]:
    return generate_latest(REGISTRY), CONTENT_TYPE_LATEST
