"""Prometheus metrics (/metrics endpoint parity with
tests/test_http_server.py:32-34 of the reference)."""

from __future__ import annotations

from typing import Optional

from prometheus_client import REGISTRY, Counter, Gauge, Histogram


def _get_or_create(cls, name, doc, **kwargs):
    try:
        return cls(name, doc, **kwargs)
    except ValueError:
        # already registered (server restarted inside one process, e.g. tests)
        collector = REGISTRY._names_to_collectors.get(name)
        if collector is None:
            for full, c in REGISTRY._names_to_collectors.items():
                if full.startswith(name):
                    return c
        return collector


class EngineMetrics:
    def __init__(self, model_name: str):
        labels = {"model_name": model_name}
        self.labelnames = ["model_name"]
        self.request_success = _get_or_create(
            Counter, "tgis_amd:request_success", "Successfully finished requests",
            labelnames=self.labelnames,
        ).labels(**labels)
        self.prompt_tokens = _get_or_create(
            Counter, "tgis_amd:prompt_tokens", "Prefill tokens processed",
            labelnames=self.labelnames,
        ).labels(**labels)
        self.generation_tokens = _get_or_create(
            Counter, "tgis_amd:generation_tokens", "Generated tokens",
            labelnames=self.labelnames,
        ).labels(**labels)
        self.ttft = _get_or_create(
            Histogram, "tgis_amd:time_to_first_token_seconds", "TTFT",
            labelnames=self.labelnames,
            buckets=[0.001, 0.005, 0.01, 0.02, 0.04, 0.06, 0.08, 0.1, 0.25, 0.5,
                     0.75, 1.0, 2.5, 5.0, 7.5, 10.0],
        ).labels(**labels)
        self.time_per_output_token = _get_or_create(
            Histogram, "tgis_amd:time_per_output_token_seconds", "Per-token latency",
            labelnames=self.labelnames,
            buckets=[0.001, 0.0025, 0.005, 0.0075, 0.01, 0.015, 0.02, 0.03,
                     0.04, 0.05, 0.075, 0.1, 0.15, 0.2, 0.3, 0.5, 1.0],
        ).labels(**labels)
        self.num_running = _get_or_create(
            Gauge, "tgis_amd:num_requests_running", "Requests currently running",
            labelnames=self.labelnames,
        ).labels(**labels)
        self.num_waiting = _get_or_create(
            Gauge, "tgis_amd:num_requests_waiting", "Requests waiting",
            labelnames=self.labelnames,
        ).labels(**labels)
        self.kv_usage = _get_or_create(
            Gauge, "tgis_amd:gpu_cache_usage_perc", "KV cache usage fraction",
            labelnames=self.labelnames,
        ).labels(**labels)
