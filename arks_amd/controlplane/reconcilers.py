"""Controllers: phase machines over the Store.

Native re-implementations of the reference's four reconcilers
(internal/controller/*.go) with the same phases, conditions, finalizers,
cross-CR gating (app waits for model Ready) and readiness semantics.
Each reconcile() is idempotent and returns a requeue hint (seconds | None).
"""

from __future__ import annotations

from typing import Any

from ..crd.types import (
    COND_APP_READY,
    COND_LOADED,
    COND_MODEL_LOADED,
    COND_PRECHECK,
    COND_READY,
    COND_STORAGE_CREATED,
    RESERVED_MOUNT_PATH,
    RESERVED_VOLUME_NAME,
    SUPPORTED_RUNTIMES,
    ApplicationPhase,
    ArksApplication,
    ArksDisaggregatedApplication,
    ArksEndpoint,
    ArksModel,
    ModelPhase,
    get_condition,
    served_model_name,
    set_condition,
)
from . import manifests
from .store import Store


def _has_finalizer(meta, fin: str) -> bool:
    return fin in meta.finalizers


class ArksModelReconciler:
    """reference arksmodel_controller.go:143-367."""

    def __init__(self, store: Store):
        self.store = store

    def reconcile(self, namespace: str, name: str) -> float | None:
        model: ArksModel | None = self.store.get_opt("ArksModel", namespace, name)
        if model is None:
            return None
        meta = model.metadata
        if meta.deletion_timestamp:
            # cleanup: owned PVC + pod removed; drop finalizer
            self.store.delete("Pod", namespace, f"arks-worker-{name}")
            meta.finalizers = [f for f in meta.finalizers if f != manifests.FINALIZER_MODEL]
            self.store.finalize(model)
            return None
        if not _has_finalizer(meta, manifests.FINALIZER_MODEL):
            meta.finalizers.append(manifests.FINALIZER_MODEL)
            self.store.update(model)
            return 0

        conds = model.status.conditions
        # 1. storage
        if not (get_condition(conds, COND_STORAGE_CREATED) or Cond_false()).status == "True":
            pvc = manifests.generate_model_pvc(model)
            if self.store.get_opt("PersistentVolumeClaim", namespace,
                                  pvc["metadata"]["name"]) is None:
                self.store.create(pvc)
            model.status.phase = ModelPhase.STORAGE_CREATING
            set_condition(conds, COND_STORAGE_CREATED, "True", "Created")
            self.store.update(model)
            return 0

        # 2. download (only when a source is declared — reference :355-358)
        if not (get_condition(conds, COND_MODEL_LOADED) or Cond_false()).status == "True":
            if model.spec.source is None or model.spec.source.huggingface is None:
                set_condition(conds, COND_MODEL_LOADED, "True", "ExistingStorage")
                self.store.update(model)
                return 0
            pod_name = f"arks-worker-{name}"
            pod = self.store.get_opt("Pod", namespace, pod_name)
            if pod is None:
                self.store.create(manifests.generate_download_pod(model))
                model.status.phase = ModelPhase.MODEL_LOADING
                self.store.update(model)
                return 5
            phase = pod.get("status", {}).get("phase", "Pending")
            if phase == "Succeeded":
                set_condition(conds, COND_MODEL_LOADED, "True", "Downloaded")
                self.store.update(model)
                return 0
            if phase == "Failed":
                msg = pod.get("status", {}).get("message", "download failed")
                model.status.phase = ModelPhase.FAILED
                set_condition(conds, COND_MODEL_LOADED, "False", "DownloadFailed", msg)
                self.store.update(model)
                return None
            model.status.phase = ModelPhase.MODEL_LOADING
            self.store.update(model)
            return 5

        # 3. ready
        if model.status.phase is not ModelPhase.READY:
            model.status.phase = ModelPhase.READY
            set_condition(conds, COND_READY, "True", "Ready")
            self.store.update(model)
        return None


def Cond_false():
    from ..crd.types import Condition

    return Condition(type="_", status="False")


class ArksApplicationReconciler:
    """reference arksapplication_controller.go:206-506."""

    def __init__(self, store: Store, default_backend: str = "rbg"):
        self.store = store
        self.default_backend = default_backend

    # -- backend auto-detect (reference determineBackend :1192-1213):
    # keep LWS if an LWS object already exists for the app; else default rbg.
    def determine_backend(self, app: ArksApplication) -> str:
        if app.status.backend:
            return app.status.backend
        if app.spec.backend:
            return app.spec.backend
        if self.store.get_opt("LeaderWorkerSet", app.metadata.namespace,
                              app.metadata.name) is not None:
            return "lws"
        return self.default_backend

    def reconcile(self, namespace: str, name: str) -> float | None:
        app: ArksApplication | None = self.store.get_opt("ArksApplication", namespace, name)
        if app is None:
            return None
        meta = app.metadata
        if meta.deletion_timestamp:
            self.store.delete("LeaderWorkerSet", namespace, name)
            self.store.delete("RoleBasedGroupSet", namespace, name)
            self.store.delete("Service", namespace, manifests.app_service_name(name))
            meta.finalizers = [
                f for f in meta.finalizers if f != manifests.FINALIZER_APPLICATION
            ]
            self.store.finalize(app)
            return None
        if not _has_finalizer(meta, manifests.FINALIZER_APPLICATION):
            meta.finalizers.append(manifests.FINALIZER_APPLICATION)
            self.store.update(app)
            return 0

        conds = app.status.conditions
        # 1. precheck (reference :236-264)
        if not (get_condition(conds, COND_PRECHECK) or Cond_false()).status == "True":
            runtime = app.spec.runtime or "arks"
            err = None
            if runtime not in SUPPORTED_RUNTIMES:
                err = f"unsupported runtime {runtime!r}"
            inst = app.spec.instance_spec or {}
            for v in inst.get("volumes", []):
                if v.get("name") == RESERVED_VOLUME_NAME:
                    err = f"volume name {RESERVED_VOLUME_NAME!r} is reserved"
            for m in inst.get("volumeMounts", []):
                if m.get("mountPath") == RESERVED_MOUNT_PATH:
                    err = f"mount path {RESERVED_MOUNT_PATH!r} is reserved"
            if err:
                app.status.phase = ApplicationPhase.FAILED
                set_condition(conds, COND_PRECHECK, "False", "PrecheckFailed", err)
                self.store.update(app)
                return None
            app.status.phase = ApplicationPhase.CHECKING
            set_condition(conds, COND_PRECHECK, "True", "PrecheckPassed")
            self.store.update(app)
            return 0

        # 2. model gate (reference :266-296)
        model_name = app.spec.model.get("name", "")
        model: ArksModel | None = self.store.get_opt("ArksModel", namespace, model_name)
        if model is None or model.status.phase is not ModelPhase.READY:
            app.status.phase = ApplicationPhase.LOADING
            set_condition(conds, COND_LOADED, "False", "ModelNotReady")
            self.store.update(app)
            return 10
        if not (get_condition(conds, COND_LOADED) or Cond_false()).status == "True":
            set_condition(conds, COND_LOADED, "True", "ModelReady")
            self.store.update(app)

        # 3. workload (reference :303-373)
        backend = self.determine_backend(app)
        app.status.backend = backend
        if backend == "lws":
            desired = manifests.generate_lws(app, model)
            cur = self.store.get_opt("LeaderWorkerSet", namespace, name)
            if cur is None:
                self.store.create(desired)
            else:
                desired["metadata"]["resourceVersion"] = cur["metadata"].get(
                    "resourceVersion", 0
                )
                desired["status"] = cur.get("status", {})
                self.store.update(desired)
        else:
            desired = manifests.generate_rbgs(app, model)
            cur = self.store.get_opt("RoleBasedGroupSet", namespace, name)
            if cur is None:
                self.store.create(desired)
            else:
                desired["metadata"]["resourceVersion"] = cur["metadata"].get(
                    "resourceVersion", 0
                )
                desired["status"] = cur.get("status", {})
                self.store.update(desired)

        # 4. leader service (reference :376-415)
        svc = manifests.generate_leader_service(app)
        if self.store.get_opt("Service", namespace, svc["metadata"]["name"]) is None:
            self.store.create(svc)

        # 5. status sync from workload (reference :424-503)
        kind = "LeaderWorkerSet" if backend == "lws" else "RoleBasedGroupSet"
        wl = self.store.get_opt(kind, namespace, name)
        st = (wl or {}).get("status", {})
        app.status.replicas = st.get("replicas", 0)
        app.status.ready_replicas = st.get("readyReplicas", 0)
        app.status.updated_replicas = st.get("updatedReplicas", 0)
        if app.status.ready_replicas >= app.spec.replicas and app.spec.replicas > 0:
            app.status.phase = ApplicationPhase.RUNNING
            set_condition(conds, COND_APP_READY, "True", "Ready")
        else:
            app.status.phase = ApplicationPhase.CREATING
            set_condition(conds, COND_APP_READY, "False", "NotReady")
        self.store.update(app)
        return 10 if app.status.phase is not ApplicationPhase.RUNNING else None


class ArksEndpointReconciler:
    """reference arksendpoint_controller.go:258-417."""

    def __init__(self, store: Store):
        self.store = store

    def _ready_apps(self, namespace: str, endpoint_name: str) -> list[str]:
        ready = []
        for app in self.store.list("ArksApplication", namespace):
            if served_model_name(app) != endpoint_name:
                continue
            # fully ready only (reference :300)
            if (
                app.spec.replicas > 0
                and app.status.replicas == app.status.ready_replicas
                and app.status.ready_replicas >= app.spec.replicas
            ):
                ready.append(app.metadata.name)
        for dapp in self.store.list("ArksDisaggregatedApplication", namespace):
            if (dapp.spec.served_model_name or dapp.spec.model.get("name", "")) != endpoint_name:
                continue
            # router>0 & prefill/decode complete (reference :326-333)
            if (
                dapp.status.router.ready > 0
                and dapp.status.prefill.ready >= dapp.spec.prefill.replicas
                and dapp.status.decode.ready >= dapp.spec.decode.replicas
            ):
                ready.append(dapp.metadata.name)
        return ready

    def reconcile(self, namespace: str, name: str) -> float | None:
        ep: ArksEndpoint | None = self.store.get_opt("ArksEndpoint", namespace, name)
        if ep is None:
            return None
        meta = ep.metadata
        if meta.deletion_timestamp:
            self.store.delete("HTTPRoute", namespace, name)
            meta.finalizers = [
                f for f in meta.finalizers if f != manifests.FINALIZER_ENDPOINT
            ]
            self.store.finalize(ep)
            return None
        if not _has_finalizer(meta, manifests.FINALIZER_ENDPOINT):
            meta.finalizers.append(manifests.FINALIZER_ENDPOINT)
            self.store.update(ep)
            return 0

        ready = self._ready_apps(namespace, name)
        route = manifests.generate_http_route(
            ep, ready, namespace, ep.spec.default_weight, ep.spec.gateway_ref,
            ep.spec.match_configs, ep.spec.route_configs,
        )
        cur = self.store.get_opt("HTTPRoute", namespace, name)
        if cur is None:
            self.store.create(route)
        else:
            route["metadata"]["resourceVersion"] = cur["metadata"].get("resourceVersion", 0)
            self.store.update(route)
        ep.status.routes = route["spec"]["rules"][0]["backendRefs"] if route["spec"]["rules"] else []
        self.store.update(ep)
        return None


class ArksDisaggregatedApplicationReconciler:
    """reference arksdisaggregatedapplication_controller.go:182-500
    (RBG-unified backend: one RBGS with scheduler/prefill/decode roles)."""

    def __init__(self, store: Store):
        self.store = store

    def reconcile(self, namespace: str, name: str) -> float | None:
        dapp: ArksDisaggregatedApplication | None = self.store.get_opt(
            "ArksDisaggregatedApplication", namespace, name
        )
        if dapp is None:
            return None
        meta = dapp.metadata
        if meta.deletion_timestamp:
            for k, n in (
                ("LeaderWorkerSet", f"{name}-prefill"),
                ("LeaderWorkerSet", f"{name}-decode"),
                ("Deployment", f"{name}-router"),
                ("Service", manifests.app_service_name(name)),
            ):
                self.store.delete(k, namespace, n)
            meta.finalizers = [f for f in meta.finalizers if f != manifests.FINALIZER_DISAGG]
            self.store.finalize(dapp)
            return None
        if not _has_finalizer(meta, manifests.FINALIZER_DISAGG):
            meta.finalizers.append(manifests.FINALIZER_DISAGG)
            self.store.update(dapp)
            return 0

        conds = dapp.status.conditions
        runtime = dapp.spec.runtime or "arks"
        # reference :208-216 allows sglang only; ours allows arks or sglang.
        if runtime not in ("arks", "sglang"):
            dapp.status.phase = ApplicationPhase.FAILED
            set_condition(conds, COND_PRECHECK, "False", "PrecheckFailed",
                          f"disaggregated runtime must be arks or sglang, got {runtime!r}")
            self.store.update(dapp)
            return None
        set_condition(conds, COND_PRECHECK, "True", "PrecheckPassed")

        model_name = dapp.spec.model.get("name", "")
        model = self.store.get_opt("ArksModel", namespace, model_name)
        if model is None or model.status.phase is not ModelPhase.READY:
            dapp.status.phase = ApplicationPhase.LOADING
            set_condition(conds, COND_LOADED, "False", "ModelNotReady")
            self.store.update(dapp)
            return 10
        set_condition(conds, COND_LOADED, "True", "ModelReady")

        served = dapp.spec.served_model_name or model_name
        from ..crd.types import model_path as _mp

        mp = _mp(model)
        from . import commands

        # prefill/decode LWS + router Deployment + router Service
        for role, wl in (("prefill", dapp.spec.prefill), ("decode", dapp.spec.decode)):
            lws = self._disagg_lws(dapp, model, role, wl, mp, served)
            cur = self.store.get_opt("LeaderWorkerSet", namespace, f"{name}-{role}")
            if cur is None:
                self.store.create(lws)
            else:
                lws["metadata"]["resourceVersion"] = cur["metadata"].get("resourceVersion", 0)
                lws["status"] = cur.get("status", {})
                self.store.update(lws)
        router = self._router_deployment(dapp, served)
        cur = self.store.get_opt("Deployment", namespace, f"{name}-router")
        if cur is None:
            self.store.create(router)
        else:
            router["metadata"]["resourceVersion"] = cur["metadata"].get("resourceVersion", 0)
            router["status"] = cur.get("status", {})
            self.store.update(router)
        svc = self._router_service(dapp)
        if self.store.get_opt("Service", namespace, svc["metadata"]["name"]) is None:
            self.store.create(svc)

        # status sync
        for role, comp in (("prefill", dapp.status.prefill), ("decode", dapp.status.decode)):
            wl = self.store.get_opt("LeaderWorkerSet", namespace, f"{name}-{role}") or {}
            st = wl.get("status", {})
            comp.replicas = st.get("replicas", 0)
            comp.ready = st.get("readyReplicas", 0)
            comp.updated = st.get("updatedReplicas", 0)
        rst = (self.store.get_opt("Deployment", namespace, f"{name}-router") or {}).get(
            "status", {}
        )
        dapp.status.router.replicas = rst.get("replicas", 0)
        dapp.status.router.ready = rst.get("readyReplicas", 0)
        complete = (
            dapp.status.router.ready >= dapp.spec.router.replicas
            and dapp.status.prefill.ready >= dapp.spec.prefill.replicas
            and dapp.status.decode.ready >= dapp.spec.decode.replicas
        )
        dapp.status.phase = (
            ApplicationPhase.RUNNING if complete else ApplicationPhase.CREATING
        )
        set_condition(conds, COND_APP_READY, "True" if complete else "False",
                      "Ready" if complete else "NotReady")
        self.store.update(dapp)
        return None if complete else 10

    def _disagg_lws(self, dapp, model, role, wl, mp, served) -> dict[str, Any]:
        from ..crd.types import LABEL_DISAGG_ROLE
        from . import commands

        tp = 0
        leader_cmd = wl.leader_command_override or commands.disagg_worker_command(
            role, mp, served, tp, wl.runtime_common_args, True
        )
        worker_cmd = wl.worker_command_override or commands.disagg_worker_command(
            role, mp, served, tp, wl.runtime_common_args, False
        )
        # reuse the standard pod template via a pseudo-app
        app = ArksApplication.model_validate(
            {
                "metadata": {
                    "name": dapp.metadata.name,
                    "namespace": dapp.metadata.namespace,
                },
                "spec": {
                    "runtime": dapp.spec.runtime or "arks",
                    "runtimeImage": dapp.spec.runtime_image,
                    "model": dapp.spec.model,
                    "instanceSpec": wl.instance_spec,
                },
            }
        )
        leader_t = manifests._pod_template(app, model, "leader", leader_cmd)
        worker_t = manifests._pod_template(app, model, "worker", worker_cmd)
        for t in (leader_t, worker_t):
            t["metadata"]["labels"][LABEL_DISAGG_ROLE] = role
        return {
            "apiVersion": "leaderworkerset.x-k8s.io/v1",
            "kind": "LeaderWorkerSet",
            "metadata": {
                "name": f"{dapp.metadata.name}-{role}",
                "namespace": dapp.metadata.namespace,
                "labels": {LABEL_DISAGG_ROLE: role},
                "ownerReferences": [manifests._owner_ref(dapp)],
            },
            "spec": {
                "replicas": wl.replicas,
                "startupPolicy": "LeaderCreated",
                "leaderWorkerTemplate": {
                    "size": wl.size,
                    "restartPolicy": "RecreateGroupOnPodRestart",
                    "leaderTemplate": leader_t,
                    "workerTemplate": worker_t,
                },
            },
        }

    def _router_deployment(self, dapp, served) -> dict[str, Any]:
        from . import commands

        name = dapp.metadata.name
        cmd = dapp.spec.router.command_override or commands.router_command(
            served, dapp.metadata.namespace, name, dapp.spec.router.port,
            dapp.spec.router.metric_port, router_args=dapp.spec.router.router_args,
        )
        return {
            "apiVersion": "apps/v1",
            "kind": "Deployment",
            "metadata": {
                "name": f"{name}-router",
                "namespace": dapp.metadata.namespace,
                "labels": {"arks.ai/sglang-router": name},
                "ownerReferences": [manifests._owner_ref(dapp)],
            },
            "spec": {
                "replicas": dapp.spec.router.replicas,
                "selector": {"matchLabels": {"arks.ai/sglang-router": name}},
                "template": {
                    "metadata": {"labels": {"arks.ai/sglang-router": name}},
                    "spec": {
                        "serviceAccountName": "sglang-router",
                        "containers": [
                            {
                                "name": "router",
                                "image": dapp.spec.router_image
                                or commands.DEFAULT_ROUTER_IMAGE,
                                "command": cmd,
                                "ports": [
                                    {"containerPort": dapp.spec.router.port},
                                    {"containerPort": dapp.spec.router.metric_port},
                                ],
                            }
                        ],
                    },
                },
            },
        }

    def _router_service(self, dapp) -> dict[str, Any]:
        # Named arks-application-<name> so ArksEndpoint backendRefs resolve
        # uniformly (reference generateApplicationServiceName :1159).
        name = dapp.metadata.name
        return {
            "apiVersion": "v1",
            "kind": "Service",
            "metadata": {
                "name": manifests.app_service_name(name),
                "namespace": dapp.metadata.namespace,
                "labels": {"prometheus-discovery": "true", "managed-by": "arks"},
                "ownerReferences": [manifests._owner_ref(dapp)],
            },
            "spec": {
                "selector": {"arks.ai/sglang-router": name},
                "ports": [{"name": "http", "port": 8080,
                           "targetPort": dapp.spec.router.port}],
            },
        }
