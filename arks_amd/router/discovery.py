"""Kubernetes pod service-discovery for the router (the capability the
reference gets from sglang-router's --service-discovery flags plus the RBAC
it provisions at arksdisaggregatedapplication_controller.go:530-596).

Dependency-free: talks to the in-cluster API server over HTTPS with the
mounted serviceaccount token, polling pods by label selector. Static worker
lists (RouterState.set_workers) are used when not running in a cluster."""

from __future__ import annotations

import asyncio
import os

import httpx

SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"


class KubePodDiscovery:
    def __init__(self, namespace: str, prefill_selector: str,
                 decode_selector: str, port: int = 8080,
                 api_base: str | None = None, interval_s: float = 5.0,
                 transport=None):
        self.namespace = namespace
        self.prefill_selector = prefill_selector
        self.decode_selector = decode_selector
        self.port = port
        host = os.environ.get("KUBERNETES_SERVICE_HOST", "kubernetes.default.svc")
        kport = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
        self.api_base = api_base or f"https://{host}:{kport}"
        self.interval_s = interval_s
        self.transport = transport
        self._stop = asyncio.Event()

    def _auth(self) -> tuple[dict, str | bool]:
        token_path = os.path.join(SA_DIR, "token")
        ca_path = os.path.join(SA_DIR, "ca.crt")
        headers = {}
        verify: str | bool = False
        if os.path.exists(token_path):
            with open(token_path) as f:
                headers["Authorization"] = f"Bearer {f.read().strip()}"
        if os.path.exists(ca_path):
            verify = ca_path
        return headers, verify

    async def _list_ready_pods(self, client: httpx.AsyncClient,
                               selector: str) -> list[str]:
        r = await client.get(
            f"/api/v1/namespaces/{self.namespace}/pods",
            params={"labelSelector": selector},
        )
        r.raise_for_status()
        urls = []
        for pod in r.json().get("items", []):
            st = pod.get("status", {})
            ip = st.get("podIP")
            conds = {c["type"]: c["status"] for c in st.get("conditions", [])}
            if ip and conds.get("Ready") == "True":
                urls.append(f"http://{ip}:{self.port}")
        return sorted(urls)

    async def run(self, state) -> None:
        """Poll loop: keeps RouterState's worker lists in sync."""
        headers, verify = self._auth()
        async with httpx.AsyncClient(
            base_url=self.api_base, headers=headers, verify=verify,
            transport=self.transport, timeout=15.0,
        ) as client:
            while not self._stop.is_set():
                try:
                    prefill = await self._list_ready_pods(client, self.prefill_selector)
                    decode = await self._list_ready_pods(client, self.decode_selector)
                    state.set_workers(prefill_urls=prefill, decode_urls=decode)
                except Exception:
                    pass  # keep last-known-good workers on API hiccups
                try:
                    await asyncio.wait_for(self._stop.wait(), self.interval_s)
                except asyncio.TimeoutError:
                    continue

    def stop(self) -> None:
        self._stop.set()
