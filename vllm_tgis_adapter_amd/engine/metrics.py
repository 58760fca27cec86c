"""Prometheus metrics (/metrics endpoint parity with
tests/test_http_server.py:32-34 of the reference)."""

from __future__ import annotations


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

    # -- process-boundary support (mp_engine): ship snapshots parent-ward ---
    def snapshot(self) -> dict:
        """Raw values for mirroring into another process's registry."""
        return {
            "request_success": self.request_success._value.get(),
            "prompt_tokens": self.prompt_tokens._value.get(),
            "generation_tokens": self.generation_tokens._value.get(),
            "num_running": self.num_running._value.get(),
            "num_waiting": self.num_waiting._value.get(),
            "kv_usage": self.kv_usage._value.get(),
            "ttft_sum": self.ttft._sum.get(),
            "ttft_count": sum(b.get() for b in self.ttft._buckets),
            "tpot_sum": self.time_per_output_token._sum.get(),
            "tpot_count": sum(b.get() for b in self.time_per_output_token._buckets),
        }

    def apply_snapshot(self, snap: dict, prev: dict) -> None:
        """Mirror a child-process snapshot (counters by delta, gauges by set)."""
        self.request_success.inc(snap["request_success"] - prev.get("request_success", 0))
        self.prompt_tokens.inc(snap["prompt_tokens"] - prev.get("prompt_tokens", 0))
        self.generation_tokens.inc(snap["generation_tokens"] - prev.get("generation_tokens", 0))
        self.num_running.set(snap["num_running"])
        self.num_waiting.set(snap["num_waiting"])
        self.kv_usage.set(snap["kv_usage"])
        # histograms: reflect sum/count movement into the +Inf bucket so
        # rate() and averages stay correct across the process boundary
        d_ttft = snap["ttft_count"] - prev.get("ttft_count", 0)
        if d_ttft > 0:
            avg = (snap["ttft_sum"] - prev.get("ttft_sum", 0.0)) / d_ttft
            for _ in range(int(d_ttft)):
                self.ttft.observe(avg)
        d_tpot = snap["tpot_count"] - prev.get("tpot_count", 0)
        if d_tpot > 0:
            avg = (snap["tpot_sum"] - prev.get("tpot_sum", 0.0)) / d_tpot
            for _ in range(int(d_tpot)):
                self.time_per_output_token.observe(avg)
