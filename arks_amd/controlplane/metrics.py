"""Operator metrics (the controller-runtime metrics server equivalent —
reference cmd/main.go:82-98 exposes controller_runtime_* / workqueue_*
families on a guarded port; config/network-policy/allow-metrics-traffic.yaml
limits who may scrape it).

Families mirror controller-runtime's names so existing dashboards/alerts
keyed on them keep working:
  controller_runtime_reconcile_total{controller, result}
  controller_runtime_reconcile_errors_total{controller}
  controller_runtime_reconcile_time_seconds{controller}
  workqueue_depth{name}
"""

from __future__ import annotations

from prometheus_client import (
    CollectorRegistry,
    Counter,
    Gauge,
    Histogram,
    generate_latest,
)

registry = CollectorRegistry()

reconcile_total = Counter(
    "controller_runtime_reconcile_total",
    "Total number of reconciliations per controller",
    ["controller", "result"],
    registry=registry,
)
reconcile_errors = Counter(
    "controller_runtime_reconcile_errors_total",
    "Total number of reconciliation errors per controller",
    ["controller"],
    registry=registry,
)
reconcile_time = Histogram(
    "controller_runtime_reconcile_time_seconds",
    "Length of time per reconciliation per controller",
    ["controller"],
    registry=registry,
    buckets=(0.005, 0.01, 0.025, 0.05, 0.1, 0.25, 0.5, 1.0, 2.5, 5.0, 10.0),
)
workqueue_depth = Gauge(
    "workqueue_depth",
    "Current depth of the reconcile work queue",
    ["name"],
    registry=registry,
)


def render() -> bytes:
    """Prometheus text exposition of the operator registry."""
    return generate_latest(registry)


def make_metrics_server(port: int, host: str = "0.0.0.0"):
    """HTTP server exposing /metrics (the operator entrypoint mounts this on
    --metrics-port; deploy/network-policy.yaml restricts who may scrape)."""
    from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

    class Metrics(BaseHTTPRequestHandler):
        def do_GET(self):
            if self.path == "/metrics":
                body = render()
                self.send_response(200)
                self.send_header("Content-Type", "text/plain; version=0.0.4")
                self.end_headers()
                self.wfile.write(body)
            else:
                self.send_response(404)
                self.end_headers()

        def log_message(self, *a):
            pass

    return ThreadingHTTPServer((host, port), Metrics)
