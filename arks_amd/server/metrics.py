"""Prometheus metrics with vLLM-compatible names.

The reference's runtime ServiceMonitor relabels metrics matching
`(sglang|vllm|...)[:_].+` into a common `runtime` label space
(reference config/prometheus/monitor-runtime.yaml:14-37), so the engine
exports the vllm:* family it expects.
"""

from __future__ import annotations

from prometheus_client import (
    CollectorRegistry,
    Counter,
    Gauge,
    Histogram,
    generate_latest,
)

_TTFT_BUCKETS = (0.001, 0.005, 0.01, 0.02, 0.04, 0.06, 0.08, 0.1, 0.25, 0.5,
                 0.75, 1.0, 2.5, 5.0, 7.5, 10.0)
_TPOT_BUCKETS = (0.01, 0.025, 0.05, 0.075, 0.1, 0.15, 0.2, 0.3, 0.4, 0.5,
                 0.75, 1.0, 2.5)
_E2E_BUCKETS = (0.3, 0.5, 0.8, 1.0, 1.5, 2.0, 2.5, 5.0, 10.0, 15.0, 20.0,
                30.0, 40.0, 50.0, 60.0)
_LEN_BUCKETS = (1, 2, 5, 10, 20, 50, 100, 200, 500, 1000, 2000, 5000,
                10000, 20000, 50000)


class EngineMetrics:
    def __init__(self, model_name: str, registry: CollectorRegistry | None = None):
        self.registry = registry or CollectorRegistry()
        label = {"model_name": model_name}
        mk = lambda cls, name, doc, **kw: cls(
            name, doc, labelnames=["model_name"], registry=self.registry, **kw
        ).labels(**label)
        self.num_requests_running = mk(
            Gauge, "vllm:num_requests_running", "running requests"
        )
        self.num_requests_waiting = mk(
            Gauge, "vllm:num_requests_waiting", "waiting requests"
        )
        self.gpu_cache_usage_perc = mk(
            Gauge, "vllm:gpu_cache_usage_perc", "KV cache usage fraction"
        )
        self.prompt_tokens = mk(Counter, "vllm:prompt_tokens", "prompt tokens")
        self.generation_tokens = mk(
            Counter, "vllm:generation_tokens", "generated tokens"
        )
        self._model_name = model_name
        self._success = Counter(
            "vllm:request_success", "finished requests",
            labelnames=["model_name", "finished_reason"],
            registry=self.registry,
        )
        # back-compat handle (unlabeled reason) for existing callers
        self.request_success = self._success.labels(
            model_name=model_name, finished_reason="stop")
        self.request_prompt_tokens = mk(
            Histogram, "vllm:request_prompt_tokens",
            "per-request prompt length", buckets=_LEN_BUCKETS,
        )
        self.request_generation_tokens = mk(
            Histogram, "vllm:request_generation_tokens",
            "per-request generation length", buckets=_LEN_BUCKETS,
        )
        self.ttft = mk(
            Histogram, "vllm:time_to_first_token_seconds", "TTFT",
            buckets=_TTFT_BUCKETS,
        )
        self.tpot = mk(
            Histogram, "vllm:time_per_output_token_seconds", "TPOT",
            buckets=_TPOT_BUCKETS,
        )
        self.e2e = mk(
            Histogram, "vllm:e2e_request_latency_seconds", "E2E latency",
            buckets=_E2E_BUCKETS,
        )
        self.preemptions = mk(Counter, "vllm:num_preemptions", "preemptions")
        self.prefix_cache_queries = mk(
            Counter, "vllm:prefix_cache_queries",
            "prompt tokens queried against the prefix cache",
        )
        self.prefix_cache_hits = mk(
            Counter, "vllm:prefix_cache_hits",
            "prompt tokens served from the prefix cache",
        )
        self.spec_draft_tokens = mk(
            Counter, "vllm:spec_decode_num_draft_tokens",
            "speculative draft tokens proposed",
        )
        self.spec_accepted_tokens = mk(
            Counter, "vllm:spec_decode_num_accepted_tokens",
            "speculative draft tokens accepted",
        )

    def record_finished(self, reason: str | None, prompt_tokens: int,
                        output_tokens: int) -> None:
        """Per-request finish accounting: success counter by finish
        reason + prompt/generation length histograms (the reference
        dashboard's heatmap + finish-reason panels read these)."""
        self._success.labels(model_name=self._model_name,
                             finished_reason=reason or "stop").inc()
        self.request_prompt_tokens.observe(prompt_tokens)
        self.request_generation_tokens.observe(output_tokens)

    def render(self) -> bytes:
        return generate_latest(self.registry)
