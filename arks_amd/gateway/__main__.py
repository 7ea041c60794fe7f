"""Gateway entrypoint (the reference's gateway-plugins binary,
cmd/gateway/main.go — here the gateway IS the HTTP hop instead of an Envoy
ext_proc side-call).

    python -m arks_amd.gateway --port 8080 [--kube-api https://...]

Token/Quota/Endpoint CRs are read live from the cluster (KubeStore resync);
rate-limit and quota state is in-process by default (fixed-window counters
with the reference's redis key semantics — arks_amd/gateway/limiter.py), or
shared across gateway replicas via --redis-addr (RESP-backed stores).
"""

from __future__ import annotations

import argparse
import threading


def main(argv=None):
    ap = argparse.ArgumentParser(prog="arks_amd.gateway")
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=8080)
    ap.add_argument("--kube-api", default=None)
    ap.add_argument("--resync", type=float, default=5.0)
    ap.add_argument("--standalone", action="store_true",
                    help="in-memory store (no cluster; for local testing)")
    ap.add_argument("--ext-proc-port", type=int, default=0,
                    help="also serve the Envoy ext_proc gRPC protocol on "
                         "this port (reference default 50052; 0 = off)")
    ap.add_argument("--redis-addr", default=None, metavar="HOST:PORT",
                    help="Redis for shared rate-limit/quota state (multi-"
                         "replica gateways; reference cmd/gateway/main.go "
                         "redis flags). Default: in-process counters.")
    args = ap.parse_args(argv)

    import uvicorn

    from .app import create_gateway_app

    limiter = quota_service = None
    if args.redis_addr:
        from .limiter import RateLimiter, RedisCounterStore
        from .quota import QuotaService, RedisQuotaStore
        from .resp import RespClient

        host, _, port = args.redis_addr.partition(":")
        client = RespClient(host, int(port or 6379))
        if not client.ping():
            raise SystemExit(f"redis at {args.redis_addr} is unreachable")
        limiter = RateLimiter(store=RedisCounterStore(client))
        quota_service = QuotaService(store=RedisQuotaStore(client))

    if args.standalone:
        from ..controlplane.store import Store

        store = Store()
    else:
        from ..controlplane.kubestore import KubeStore

        store = KubeStore(api_base=args.kube_api)
        threading.Thread(target=store.run_resync, args=(args.resync,),
                         daemon=True).start()
    app = create_gateway_app(store, limiter=limiter,
                             quota_service=quota_service)
    if args.ext_proc_port:
        from .extproc import serve as serve_extproc

        serve_extproc(app.state.provider, app.state.limiter,
                      app.state.quota_service, port=args.ext_proc_port)
    uvicorn.run(app, host=args.host, port=args.port, log_level="info")


if __name__ == "__main__":
    main()
