"""Prometheus metrics (the reference's util/prometheus is an explicit
stub, prometheus.go:7-9 — here it is real: request counters/latency,
LLM token/TTFT metrics, runner gauges, served at /metrics).

Each app instance gets its own CollectorRegistry so tests can build
many apps in one process without duplicate-registration errors.
"""
from __future__ import annotations

from prometheus_client import (CollectorRegistry, Counter, Gauge,
                               Histogram, generate_latest,
                               CONTENT_TYPE_LATEST)


class Metrics:
    def __init__(self):
        self.registry = CollectorRegistry()
        self.http_requests = Counter(
            "helix_http_requests_total", "HTTP requests",
            ["method", "route", "status"], registry=self.registry)
        self.http_latency = Histogram(
            "helix_http_request_seconds", "HTTP request latency",
            ["route"], registry=self.registry,
            buckets=(.005, .02, .05, .1, .25, .5, 1, 2.5, 5, 10, 30, 60))
        self.llm_calls = Counter(
            "helix_llm_calls_total", "LLM calls",
            ["provider", "model"], registry=self.registry)
        self.llm_tokens = Counter(
            "helix_llm_tokens_total", "LLM tokens",
            ["model", "kind"], registry=self.registry)
        self.ttft_ms = Histogram(
            "helix_llm_ttft_ms", "Time to first token (ms)",
            registry=self.registry,
            buckets=(5, 10, 25, 50, 100, 250, 500, 1000, 2500, 5000,
                     10000))
        self.runners_online = Gauge(
            "helix_runners_online", "Connected runners",
            registry=self.registry)
        self.models_loaded = Gauge(
            "helix_models_loaded", "Models loaded on the local runner",
            registry=self.registry)

    def observe_call(self, call) -> None:
        """Feed from UsageService.log_call (LLMCall)."""
        try:
            self.llm_calls.labels(call.provider or "unknown",
                                  call.model or "unknown").inc()
            if call.prompt_tokens:
                self.llm_tokens.labels(call.model or "unknown",
                                       "prompt").inc(call.prompt_tokens)
            if call.completion_tokens:
                self.llm_tokens.labels(
                    call.model or "unknown",
                    "completion").inc(call.completion_tokens)
            if call.first_token_ms:
                self.ttft_ms.observe(call.first_token_ms)
        except Exception:
            pass

    def render(self) -> tuple:
        return generate_latest(self.registry), CONTENT_TYPE_LATEST
