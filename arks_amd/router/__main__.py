"""Router entrypoint. Flag surface mirrors what the control plane composes
(arks_amd/controlplane/commands.py router_command, modeled on the reference's
sglang-router invocation at arksdisaggregatedapplication_controller.go:1630-1670):

  python -m arks_amd.router --pd-disaggregation --service-discovery \
      --namespace ns --prefill-selector k=v k2=v2 --decode-selector k=v \
      --port 8080 --prometheus-port 9110 --policy cache_aware

or with static workers (no cluster):

  python -m arks_amd.router --prefill-urls http://p:8080 \
      --decode-urls http://d1:8080 http://d2:8080 --port 8000
"""

from __future__ import annotations

import argparse
import asyncio


def parse_args(argv=None):
    p = argparse.ArgumentParser(prog="arks_amd.router")
    p.add_argument("--pd-disaggregation", action="store_true")
    p.add_argument("--service-discovery", action="store_true")
    p.add_argument("--namespace", default="default")
    p.add_argument("--prefill-selector", nargs="*", default=[])
    p.add_argument("--decode-selector", nargs="*", default=[])
    p.add_argument("--prefill-urls", nargs="*", default=[])
    p.add_argument("--decode-urls", nargs="*", default=[])
    p.add_argument("--worker-port", type=int, default=8080)
    p.add_argument("--host", default="0.0.0.0")
    p.add_argument("--port", type=int, default=8080)
    p.add_argument("--prometheus-port", type=int, default=9110)
    p.add_argument("--policy", choices=["cache_aware", "round_robin"],
                   default="cache_aware")
    p.add_argument("--discovery-interval", type=float, default=5.0)
    return p.parse_args(argv)


def main(argv=None):
    args = parse_args(argv)
    import uvicorn

    from .app import RouterState, create_router_app

    state = RouterState(args.prefill_urls, args.decode_urls, policy=args.policy)
    app = create_router_app(state)

    if args.service_discovery:
        from .discovery import KubePodDiscovery

        disc = KubePodDiscovery(
            namespace=args.namespace,
            prefill_selector=",".join(args.prefill_selector),
            decode_selector=",".join(args.decode_selector),
            port=args.worker_port,
            interval_s=args.discovery_interval,
        )

        @app.on_event("startup")
        async def _start_discovery():
            asyncio.get_running_loop().create_task(disc.run(state))

        @app.on_event("shutdown")
        async def _stop_discovery():
            disc.stop()

    uvicorn.run(app, host=args.host, port=args.port, log_level="info")


if __name__ == "__main__":
    main()
