"""Prometheus metrics for the emulator — the vLLM metric names the
collector consumes (wva_amd/controller/constants.py), complete.

Uses an isolated CollectorRegistry per instance so tests can run several
emulated servers in one process.
"""

from __future__ import annotations

from prometheus_client import CollectorRegistry, Counter, Gauge, Histogram, generate_latest

REQUEST_LATENCY_BUCKETS = [
    0.3, 0.5, 0.8, 1.0, 1.5, 2.0, 2.5, 5.0, 10.0, 15.0, 20.0, 30.0,
    40.0, 50.0, 60.0, 120.0, 240.0, 480.0, 960.0, 1920.0, 7680.0,
]
ITL_BUCKETS = [0.01, 0.025, 0.05, 0.075, 0.1, 0.15, 0.2, 0.3, 0.4, 0.5, 0.75, 1.0, 2.5]
TTFT_BUCKETS = [0.001, 0.005, 0.01, 0.02, 0.04, 0.06, 0.08, 0.1, 0.25, 0.5, 0.75, 1.0, 2.5, 5.0, 7.5, 10.0]
TOKEN_BUCKETS = [1, 2, 5, 10, 20, 50, 100, 200, 500, 1000, 2000, 5000]


class EmulatorMetrics:
    def __init__(self, model_name: str) -> None:
        self.registry = CollectorRegistry()
        self.model_name = model_name
        labels = ["model_name"]
        reg = self.registry

        self.running = Gauge(
            "vllm:num_requests_running", "Requests currently running on GPU.",
            labels, registry=reg)
        self.waiting = Gauge(
            "vllm:num_requests_waiting", "Requests waiting to be processed.",
            labels, registry=reg)
        self.kv_cache_usage = Gauge(
            "vllm:gpu_cache_usage_perc", "GPU KV-cache usage fraction.",
            labels, registry=reg)
        self.request_arrival = Counter(
            "vllm:request_arrival", "Total request arrivals.", labels, registry=reg)
        self.request_success = Counter(
            "vllm:request_success", "Total requests completed.", labels, registry=reg)
        self.tokens_total = Counter(
            "vllm:tokens", "Total tokens generated.", labels, registry=reg)
        self.time_per_output_token = Histogram(
            "vllm:time_per_output_token_seconds", "Inter-token latency (s).",
            labels, registry=reg, buckets=ITL_BUCKETS)
        self.time_to_first_token = Histogram(
            "vllm:time_to_first_token_seconds", "Time to first token (s).",
            labels, registry=reg, buckets=TTFT_BUCKETS)
        self.queue_time = Histogram(
            "vllm:request_queue_time_seconds", "Time in WAITING phase (s).",
            labels, registry=reg, buckets=REQUEST_LATENCY_BUCKETS)
        self.prompt_tokens = Histogram(
            "vllm:request_prompt_tokens", "Prompt token count per request.",
            labels, registry=reg, buckets=TOKEN_BUCKETS)
        self.generation_tokens = Histogram(
            "vllm:request_generation_tokens", "Generated token count per request.",
            labels, registry=reg, buckets=TOKEN_BUCKETS)

        # pre-register the labeled children so every series is exported
        # from process start (as real vLLM does) — otherwise the collector's
        # availability probe sees no series until the first request
        for metric in (
            self.running, self.waiting, self.kv_cache_usage, self.request_arrival,
            self.request_success, self.tokens_total, self.time_per_output_token,
            self.time_to_first_token, self.queue_time, self.prompt_tokens,
            self.generation_tokens,
        ):
            metric.labels(model_name=model_name)

    def expose(self) -> bytes:
        return generate_latest(self.registry)

    def l(self, metric):  # bound to this emulator's model
        return metric.labels(model_name=self.model_name)
