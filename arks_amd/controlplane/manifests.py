"""Workload manifest builders.

Dict-shaped K8s manifests mirroring the reference's generated objects:
LeaderWorkerSet (generateLws, reference arksapplication_controller.go:509-699),
RoleBasedGroupSet (generateRBGS, :701-889), the leader Service (:376-415),
the model download pod (arksmodel_controller.go:218-359) and HTTPRoute
(arksendpoint_controller.go:258-417).
"""

from __future__ import annotations

import os
from typing import Any

from ..crd.types import (
    LABEL_APPLICATION,
    LABEL_MODEL,
    LABEL_WORKLOAD_ROLE,
    RESERVED_MOUNT_PATH,
    RESERVED_VOLUME_NAME,
    ArksApplication,
    ArksModel,
    model_path,
    served_model_name,
)
from . import commands

FINALIZER_APPLICATION = "application.arks.ai/controller"
FINALIZER_MODEL = "model.arks.ai/controller"
FINALIZER_ENDPOINT = "endpoint.arks.ai/controller"
FINALIZER_DISAGG = "disaggregatedapplication.arks.ai/controller"

SCRIPTS_IMAGE_ENV = "ARKS_SCRIPTS_IMAGE"
DEFAULT_SCRIPTS_IMAGE = "arks-amd/scripts:latest"


def app_service_name(app_name: str) -> str:
    return f"arks-application-{app_name}"


def _model_volume(model: ArksModel) -> tuple[dict, dict]:
    pvc_name = (
        model.spec.storage.pvc.name
        if model.spec.storage and model.spec.storage.pvc and model.spec.storage.pvc.name
        else model.metadata.name
    )
    vol = {
        "name": RESERVED_VOLUME_NAME,
        "persistentVolumeClaim": {"claimName": pvc_name, "readOnly": True},
    }
    mount = {"name": RESERVED_VOLUME_NAME, "mountPath": RESERVED_MOUNT_PATH,
             "readOnly": True}
    return vol, mount


def _base_labels(app: ArksApplication) -> dict[str, str]:
    return {
        LABEL_APPLICATION: app.metadata.name,
        LABEL_MODEL: app.spec.model.get("name", ""),
    }


def _pod_template(app: ArksApplication, model: ArksModel, role: str,
                  command: list[str]) -> dict[str, Any]:
    vol, mount = _model_volume(model)
    inst = app.spec.instance_spec or {}
    labels = {**_base_labels(app), LABEL_WORKLOAD_ROLE: role}
    container: dict[str, Any] = {
        "name": "runtime",
        "image": commands.runtime_image(
            app.spec.runtime or "arks", app.spec.runtime_image
        ),
        "command": command,
        "ports": [{"containerPort": 8080, "name": "http"}],
        "volumeMounts": [mount] + list(inst.get("volumeMounts", [])),
        "env": list(inst.get("env", [])),
        "resources": inst.get("resources", {}),
    }
    if role == "leader":
        container["readinessProbe"] = {
            "httpGet": {"path": "/health", "port": 8080},
            "initialDelaySeconds": 10,
            "periodSeconds": 5,
        }
    if app.spec.runtime == "sglang":
        # sglang needs LWS_WORKER_INDEX from the LWS pod label
        # (reference arksapplication_controller.go:560-569)
        container["env"].append(
            {
                "name": "LWS_WORKER_INDEX",
                "valueFrom": {
                    "fieldRef": {
                        "fieldPath": "metadata.labels['leaderworkerset.sigs.k8s.io/worker-index']"
                    }
                },
            }
        )
    spec: dict[str, Any] = {
        "containers": [container],
        "volumes": [vol] + list(inst.get("volumes", [])),
    }
    for f in ("nodeSelector", "affinity", "tolerations", "schedulerName",
              "priorityClassName", "imagePullSecrets", "initContainers",
              "terminationGracePeriodSeconds", "serviceAccountName"):
        if inst.get(f):
            spec[f] = inst[f]
    if app.spec.runtime_image_pull_secrets:
        spec.setdefault("imagePullSecrets", []).extend(
            app.spec.runtime_image_pull_secrets
        )
    return {"metadata": {"labels": labels}, "spec": spec}


def generate_lws(app: ArksApplication, model: ArksModel) -> dict[str, Any]:
    """LeaderWorkerSet manifest (reference generateLws :509-699)."""
    mp = model_path(model)
    served = served_model_name(app)
    runtime = app.spec.runtime or "arks"
    tp = app.spec.tensor_parallel_size
    leader_cmd = commands.leader_command(
        runtime, mp, served, tp, app.spec.runtime_common_args, app.spec.size
    )
    worker_cmd = commands.worker_command(
        runtime, mp, served, tp, app.spec.runtime_common_args
    )
    lws: dict[str, Any] = {
        "apiVersion": "leaderworkerset.x-k8s.io/v1",
        "kind": "LeaderWorkerSet",
        "metadata": {
            "name": app.metadata.name,
            "namespace": app.metadata.namespace,
            "labels": _base_labels(app),
            "ownerReferences": [_owner_ref(app)],
        },
        "spec": {
            "replicas": app.spec.replicas,
            "startupPolicy": "LeaderCreated",
            "leaderWorkerTemplate": {
                "size": app.spec.size,
                "restartPolicy": "RecreateGroupOnPodRestart",
                "leaderTemplate": _pod_template(app, model, "leader", leader_cmd),
                "workerTemplate": _pod_template(app, model, "worker", worker_cmd),
            },
        },
    }
    return lws


def generate_rbgs(app: ArksApplication, model: ArksModel) -> dict[str, Any]:
    """RoleBasedGroupSet manifest with one `inference` LWS role
    (reference generateRBGS :701-889)."""
    lws = generate_lws(app, model)
    rbgs: dict[str, Any] = {
        "apiVersion": "workloads.x-k8s.io/v1alpha1",
        "kind": "RoleBasedGroupSet",
        "metadata": {
            "name": app.metadata.name,
            "namespace": app.metadata.namespace,
            "labels": _base_labels(app),
            "ownerReferences": [_owner_ref(app)],
        },
        "spec": {
            "replicas": app.spec.replicas,
            "template": {
                "roles": [
                    {
                        "name": "inference",
                        "replicas": 1,
                        "workload": {
                            "apiVersion": "leaderworkerset.x-k8s.io/v1",
                            "kind": "LeaderWorkerSet",
                        },
                        "leaderWorkerSet": {
                            "size": app.spec.size,
                            "patchLeaderTemplate": lws["spec"]["leaderWorkerTemplate"][
                                "leaderTemplate"
                            ],
                            "patchWorkerTemplate": lws["spec"]["leaderWorkerTemplate"][
                                "workerTemplate"
                            ],
                        },
                        "rolloutStrategy": {
                            "rollingUpdate": {"maxUnavailable": 1, "maxSurge": 0}
                        },
                    }
                ]
            },
        },
    }
    if app.spec.pod_group_policy:
        rbgs["spec"]["podGroupPolicy"] = app.spec.pod_group_policy.model_dump(
            by_alias=True, exclude_none=True
        )
    return rbgs


def generate_leader_service(app: ArksApplication) -> dict[str, Any]:
    """Leader Service (reference :376-415): port 8080, selector app+leader,
    labeled for the runtime ServiceMonitor."""
    return {
        "apiVersion": "v1",
        "kind": "Service",
        "metadata": {
            "name": app_service_name(app.metadata.name),
            "namespace": app.metadata.namespace,
            "labels": {
                **_base_labels(app),
                "prometheus-discovery": "true",
                "managed-by": "arks",
            },
            "ownerReferences": [_owner_ref(app)],
        },
        "spec": {
            "selector": {
                LABEL_APPLICATION: app.metadata.name,
                LABEL_WORKLOAD_ROLE: "leader",
            },
            "ports": [{"name": "http", "port": 8080, "targetPort": 8080}],
        },
    }


def generate_model_pvc(model: ArksModel) -> dict[str, Any]:
    pvc_spec = (
        model.spec.storage.pvc.spec
        if model.spec.storage and model.spec.storage.pvc
        else {}
    )
    name = (
        model.spec.storage.pvc.name
        if model.spec.storage and model.spec.storage.pvc and model.spec.storage.pvc.name
        else model.metadata.name
    )
    return {
        "apiVersion": "v1",
        "kind": "PersistentVolumeClaim",
        "metadata": {
            "name": name,
            "namespace": model.metadata.namespace,
            "labels": {LABEL_MODEL: model.metadata.name},
            "ownerReferences": [_owner_ref(model)],
        },
        "spec": pvc_spec
        or {
            "accessModes": ["ReadWriteMany"],
            "resources": {"requests": {"storage": "100Gi"}},
        },
    }


def generate_download_pod(model: ArksModel) -> dict[str, Any]:
    """One-shot HF download pod (reference arksmodel_controller.go:218-359):
    runs scripts/download.py with MODEL_NAME/MODEL_PATH/HF_TOKEN."""
    vol, mount = _model_volume(model)
    mount = {**mount, "readOnly": False}
    vol = {**vol, "persistentVolumeClaim": {**vol["persistentVolumeClaim"], "readOnly": False}}
    env = [
        {"name": "MODEL_NAME", "value": model.spec.model},
        {"name": "MODEL_PATH", "value": model_path(model)},
    ]
    src = model.spec.source.huggingface if model.spec.source else None
    if src and src.token_secret_ref:
        env.append(
            {
                "name": "HF_TOKEN",
                "valueFrom": {
                    "secretKeyRef": {
                        "name": src.token_secret_ref.get("name", ""),
                        "key": "HF_TOKEN",
                    }
                },
            }
        )
    return {
        "apiVersion": "v1",
        "kind": "Pod",
        "metadata": {
            "name": f"arks-worker-{model.metadata.name}",
            "namespace": model.metadata.namespace,
            "labels": {LABEL_MODEL: model.metadata.name},
            "ownerReferences": [_owner_ref(model)],
        },
        "spec": {
            "restartPolicy": "Never",
            "containers": [
                {
                    "name": "download",
                    "image": os.environ.get(SCRIPTS_IMAGE_ENV, DEFAULT_SCRIPTS_IMAGE),
                    "command": ["python3", "-m", "arks_amd.loader.download"],
                    "env": env,
                    "volumeMounts": [mount],
                    "terminationMessagePolicy": "FallbackToLogsOnError",
                }
            ],
            "volumes": [vol],
            "imagePullSecrets": list(model.spec.image_pull_secrets),
        },
    }


def generate_http_route(endpoint, ready_apps: list[str], namespace: str,
                        default_weight: int, gateway_ref: dict,
                        match_configs: list[dict],
                        route_configs: list[dict]) -> dict[str, Any]:
    """HTTPRoute with weighted backends + namespace/model header matches
    (reference arksendpoint_controller.go:258-417). The `namespace` and
    `model` headers are injected by the gateway plugin
    (arks_amd/gateway — reference handle_request.go:208-231)."""
    backend_refs = list(route_configs)
    for app_name in sorted(ready_apps):
        backend_refs.append(
            {
                "name": app_service_name(app_name),
                "port": 8080,
                "weight": default_weight,
                "kind": "Service",
            }
        )
    matches = match_configs or [{"path": {"type": "PathPrefix", "value": "/"}}]
    rules = []
    header_match = [
        {"type": "Exact", "name": "namespace", "value": namespace},
        {"type": "Exact", "name": "model", "value": endpoint.metadata.name},
    ]
    for m in matches:
        mm = dict(m)
        mm["headers"] = list(mm.get("headers", [])) + header_match
        rules.append({"matches": [mm], "backendRefs": backend_refs})
    return {
        "apiVersion": "gateway.networking.k8s.io/v1",
        "kind": "HTTPRoute",
        "metadata": {
            "name": endpoint.metadata.name,
            "namespace": namespace,
            "ownerReferences": [_owner_ref(endpoint)],
        },
        "spec": {
            "parentRefs": [gateway_ref] if gateway_ref else [],
            "rules": rules,
        },
    }


def _owner_ref(obj) -> dict[str, Any]:
    return {
        "apiVersion": "arks.ai/v1",
        "kind": obj.kind,
        "name": obj.metadata.name,
        "controller": True,
    }
