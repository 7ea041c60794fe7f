"""Operator entrypoint (the reference's /manager binary, cmd/main.go).

    python -m arks_amd.controlplane [--kube-api https://...] [--resync 5]

Runs the four reconcilers against the cluster through KubeStore, with a
poll-resync loop feeding the work queue and /healthz + /readyz probes on
--health-port (reference cmd/main.go:157-196).
"""

from __future__ import annotations

import argparse
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer


def main(argv=None):
    ap = argparse.ArgumentParser(prog="arks_amd.controlplane")
    ap.add_argument("--kube-api", default=None,
                    help="API server base URL (default: in-cluster)")
    ap.add_argument("--resync", type=float, default=5.0)
    ap.add_argument("--health-port", type=int, default=8081)
    ap.add_argument("--metrics-port", type=int, default=8443,
                    help="Prometheus metrics port (reference manager serves "
                         "controller metrics on :8443; 0 disables)")
    ap.add_argument("--leader-elect", action="store_true",
                    help="gate reconciling on a coordination.k8s.io Lease "
                         "(reference cmd/main.go:198-216)")
    ap.add_argument("--leader-elect-namespace", default="arks-system")
    args = ap.parse_args(argv)

    import os as _os

    from .kubestore import KubeStore
    from .operator import Operator

    store = KubeStore(api_base=args.kube_api)
    op = Operator(store)

    elector = None
    if args.leader_elect:
        from .leaderelect import LeaderElector

        identity = _os.environ.get("HOSTNAME") or f"pid-{_os.getpid()}"
        elector = LeaderElector(store, identity,
                                namespace=args.leader_elect_namespace)
        print(f"waiting for leader lease as {identity!r} ...", flush=True)
        elector.acquire()
        print("acquired leader lease", flush=True)

        def _lost():
            print("leader lease lost — exiting for restart", flush=True)
            _os._exit(1)  # pod restarts as a follower

        threading.Thread(target=elector.run_renew, args=(_lost,),
                         daemon=True).start()

    class Probe(BaseHTTPRequestHandler):
        def do_GET(self):
            if self.path in ("/healthz", "/readyz"):
                self.send_response(200)
                self.end_headers()
                self.wfile.write(b"ok")
            else:
                self.send_response(404)
                self.end_headers()

        def log_message(self, *a):
            pass

    health = ThreadingHTTPServer(("0.0.0.0", args.health_port), Probe)
    threading.Thread(target=health.serve_forever, daemon=True).start()

    metrics_srv = None
    if args.metrics_port:
        from .metrics import make_metrics_server

        metrics_srv = make_metrics_server(args.metrics_port)
        threading.Thread(target=metrics_srv.serve_forever,
                         daemon=True).start()
    # watch streams (informer path); a slow resync loop remains as the
    # safety net against missed events
    store.run_watch()
    threading.Thread(target=store.run_resync, args=(max(args.resync, 300.0),),
                     daemon=True).start()
    print(f"arks operator running (watch + {max(args.resync, 300.0):.0f}s "
          f"resync, health :{args.health_port})", flush=True)
    try:
        op.run()
    except KeyboardInterrupt:
        pass
    finally:
        if elector is not None:
            elector.release()
        store.stop()
        op.stop()
        health.shutdown()
        if metrics_srv is not None:
            metrics_srv.shutdown()


if __name__ == "__main__":
    main()
