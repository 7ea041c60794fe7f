"""The arks.ai/v1 API types, re-modeled natively (pydantic).

Field-for-field port of the reference CRDs (reference api/v1/*_types.go):
ArksModel, ArksApplication, ArksDisaggregatedApplication, ArksEndpoint,
ArksToken, ArksQuota — same phases, conditions, label keys and defaulting,
so the reference's sample YAMLs parse unchanged. The controllers in
arks_amd.controlplane consume these.
"""

from __future__ import annotations

import time
from enum import Enum
from typing import Any, Literal

from pydantic import BaseModel, ConfigDict, Field


def _camel(s: str) -> str:
    parts = s.split("_")
    return parts[0] + "".join(p.title() for p in parts[1:])


class K8sModel(BaseModel):
    model_config = ConfigDict(alias_generator=_camel, populate_by_name=True, extra="allow")


# --------------------------------------------------------------------------
# shared metadata / conditions
# --------------------------------------------------------------------------
class ObjectMeta(K8sModel):
    name: str = ""
    namespace: str = "default"
    labels: dict[str, str] = Field(default_factory=dict)
    annotations: dict[str, str] = Field(default_factory=dict)
    finalizers: list[str] = Field(default_factory=list)
    resource_version: int = 0
    deletion_timestamp: float | None = None


class Condition(K8sModel):
    type: str
    status: str  # "True"/"False"/"Unknown"
    reason: str = ""
    message: str = ""
    last_transition_time: float = Field(default_factory=time.time)


def set_condition(conds: list[Condition], type_: str, status: str,
                  reason: str = "", message: str = "") -> None:
    for c in conds:
        if c.type == type_:
            if c.status != status or c.reason != reason:
                c.status, c.reason, c.message = status, reason, message
                c.last_transition_time = time.time()
            return
    conds.append(Condition(type=type_, status=status, reason=reason, message=message))


def get_condition(conds: list[Condition], type_: str) -> Condition | None:
    return next((c for c in conds if c.type == type_), None)


# --------------------------------------------------------------------------
# label keys / constants (reference arksapplication_types.go:57-63)
# --------------------------------------------------------------------------
LABEL_APPLICATION = "arks.ai/application"
LABEL_MODEL = "arks.ai/model"
LABEL_TOKEN = "arks.ai/token"
LABEL_QUOTA = "arks.ai/quota"
LABEL_WORKLOAD_ROLE = "arks.ai/work-load-role"
LABEL_DISAGG_ROLE = "arks.ai/disaggregation-role"
LABEL_SGLANG_ROUTER = "arks.ai/sglang-router"

RUNTIME_ARKS = "arks"  # our first-party engine (the default)
RUNTIME_VLLM = "vllm"
RUNTIME_SGLANG = "sglang"
RUNTIME_DYNAMO = "dynamo"
SUPPORTED_RUNTIMES = (RUNTIME_ARKS, RUNTIME_VLLM, RUNTIME_SGLANG, RUNTIME_DYNAMO)

BACKEND_LWS = "lws"
BACKEND_RBG = "rbg"

RESERVED_VOLUME_NAME = "models"
RESERVED_MOUNT_PATH = "/models"


# --------------------------------------------------------------------------
# ArksModel (reference api/v1/arksmodel_types.go)
# --------------------------------------------------------------------------
class ModelPhase(str, Enum):
    EMPTY = ""
    PENDING = "Pending"
    STORAGE_CREATING = "StorageCreating"
    MODEL_LOADING = "ModelLoading"
    READY = "Ready"
    FAILED = "Failed"


COND_STORAGE_CREATED = "StorageCreated"
COND_MODEL_LOADED = "ModelLoaded"
COND_READY = "Ready"


class HuggingfaceSource(K8sModel):
    token_secret_ref: dict[str, Any] | None = None


class ModelSource(K8sModel):
    huggingface: HuggingfaceSource | None = None


class PVCSpec(K8sModel):
    name: str | None = None
    spec: dict[str, Any] = Field(default_factory=dict)


class ModelStorage(K8sModel):
    pvc: PVCSpec | None = None
    sub_path: str | None = None


class ArksModelSpec(K8sModel):
    model: str = ""  # HF repo id
    source: ModelSource | None = None
    storage: ModelStorage | None = None
    image_pull_secrets: list[dict[str, Any]] = Field(default_factory=list)
    instance_spec: dict[str, Any] = Field(default_factory=dict)


class ArksModelStatus(K8sModel):
    phase: ModelPhase = ModelPhase.EMPTY
    conditions: list[Condition] = Field(default_factory=list)


class ArksModel(K8sModel):
    api_version: str = "arks.ai/v1"
    kind: Literal["ArksModel"] = "ArksModel"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    spec: ArksModelSpec = Field(default_factory=ArksModelSpec)
    status: ArksModelStatus = Field(default_factory=ArksModelStatus)


def model_path(model: ArksModel) -> str:
    """Weights path contract (reference arksmodel_controller.go:377-382)."""
    if model.spec.storage and model.spec.storage.sub_path:
        return f"/models/{model.spec.storage.sub_path}"
    return f"/models/models/{model.metadata.namespace}/{model.metadata.name}"


# --------------------------------------------------------------------------
# ArksApplication (reference api/v1/arksapplication_types.go)
# --------------------------------------------------------------------------
class ApplicationPhase(str, Enum):
    EMPTY = ""
    PENDING = "Pending"
    CHECKING = "Checking"
    LOADING = "Loading"
    CREATING = "Creating"
    RUNNING = "Running"
    FAILED = "Failed"


COND_PRECHECK = "Precheck"
COND_LOADED = "Loaded"
COND_APP_READY = "Ready"


class PodGroupPolicy(K8sModel):
    kube_scheduling: dict[str, Any] | None = None
    volcano: dict[str, Any] | None = None


class ArksApplicationSpec(K8sModel):
    replicas: int = 1
    size: int = 1  # nodes per inference group (LWS group size)
    runtime: str = ""  # defaults to arks
    runtime_image: str = ""
    runtime_image_pull_secrets: list[dict[str, Any]] = Field(default_factory=list)
    model: dict[str, str] = Field(default_factory=dict)  # LocalObjectReference
    served_model_name: str = ""
    tensor_parallel_size: int = 0
    runtime_common_args: list[str] = Field(default_factory=list)
    instance_spec: dict[str, Any] = Field(default_factory=dict)
    pod_group_policy: PodGroupPolicy | None = None
    backend: str = ""  # lws | rbg ("" = auto-detect)


class ArksApplicationStatus(K8sModel):
    phase: ApplicationPhase = ApplicationPhase.EMPTY
    replicas: int = 0
    ready_replicas: int = 0
    updated_replicas: int = 0
    conditions: list[Condition] = Field(default_factory=list)
    backend: str = ""


class ArksApplication(K8sModel):
    api_version: str = "arks.ai/v1"
    kind: Literal["ArksApplication"] = "ArksApplication"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    spec: ArksApplicationSpec = Field(default_factory=ArksApplicationSpec)
    status: ArksApplicationStatus = Field(default_factory=ArksApplicationStatus)


def served_model_name(app: ArksApplication) -> str:
    return app.spec.served_model_name or app.spec.model.get("name", "")


# --------------------------------------------------------------------------
# ArksDisaggregatedApplication (reference arksdisaggregatedapplication_types.go)
# --------------------------------------------------------------------------
class DisaggRouter(K8sModel):
    replicas: int = 1
    command_override: list[str] = Field(default_factory=list)
    port: int = 8080
    metric_port: int = 9090
    router_args: list[str] = Field(default_factory=list)
    instance_spec: dict[str, Any] = Field(default_factory=dict)


class DisaggWorkload(K8sModel):
    replicas: int = 1
    size: int = 1
    leader_command_override: list[str] = Field(default_factory=list)
    worker_command_override: list[str] = Field(default_factory=list)
    runtime_common_args: list[str] = Field(default_factory=list)
    instance_spec: dict[str, Any] = Field(default_factory=dict)


class ArksDisaggregatedApplicationSpec(K8sModel):
    runtime: str = ""
    router_image: str = ""
    runtime_image: str = ""
    model: dict[str, str] = Field(default_factory=dict)
    served_model_name: str = ""
    router: DisaggRouter = Field(default_factory=DisaggRouter)
    prefill: DisaggWorkload = Field(default_factory=DisaggWorkload)
    decode: DisaggWorkload = Field(default_factory=DisaggWorkload)
    pod_group_policy: PodGroupPolicy | None = None


class DisaggComponentStatus(K8sModel):
    replicas: int = 0
    ready: int = 0
    updated: int = 0


class ArksDisaggregatedApplicationStatus(K8sModel):
    phase: ApplicationPhase = ApplicationPhase.EMPTY
    router: DisaggComponentStatus = Field(default_factory=DisaggComponentStatus)
    prefill: DisaggComponentStatus = Field(default_factory=DisaggComponentStatus)
    decode: DisaggComponentStatus = Field(default_factory=DisaggComponentStatus)
    conditions: list[Condition] = Field(default_factory=list)


class ArksDisaggregatedApplication(K8sModel):
    api_version: str = "arks.ai/v1"
    kind: Literal["ArksDisaggregatedApplication"] = "ArksDisaggregatedApplication"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    spec: ArksDisaggregatedApplicationSpec = Field(
        default_factory=ArksDisaggregatedApplicationSpec
    )
    status: ArksDisaggregatedApplicationStatus = Field(
        default_factory=ArksDisaggregatedApplicationStatus
    )


# --------------------------------------------------------------------------
# ArksEndpoint (reference api/v1/arksendpoint_types.go)
# --------------------------------------------------------------------------
class ArksEndpointSpec(K8sModel):
    default_weight: int = 1
    gateway_ref: dict[str, Any] = Field(default_factory=dict)
    match_configs: list[dict[str, Any]] = Field(default_factory=list)
    route_configs: list[dict[str, Any]] = Field(default_factory=list)


class ArksEndpointStatus(K8sModel):
    routes: list[dict[str, Any]] = Field(default_factory=list)


class ArksEndpoint(K8sModel):
    api_version: str = "arks.ai/v1"
    kind: Literal["ArksEndpoint"] = "ArksEndpoint"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    spec: ArksEndpointSpec = Field(default_factory=ArksEndpointSpec)
    status: ArksEndpointStatus = Field(default_factory=ArksEndpointStatus)


# --------------------------------------------------------------------------
# ArksToken / ArksQuota (reference arkstoken_types.go / arksquota_types.go)
# --------------------------------------------------------------------------
RATE_LIMIT_TYPES = ("rpm", "rpd", "tpm", "tpd")
QUOTA_TYPES = ("prompt", "response", "total")


class RateLimit(K8sModel):
    type: str  # rpm | rpd | tpm | tpd
    value: int


class TokenQos(K8sModel):
    # reference arkstoken_types.go: a LocalObjectReference ({"name": ...});
    # a bare string is accepted too.
    arks_endpoint: Any = ""
    rate_limits: list[RateLimit] = Field(default_factory=list)
    quota: dict[str, str] = Field(default_factory=dict)  # {"name": quota-name}

    @property
    def endpoint_name(self) -> str:
        if isinstance(self.arks_endpoint, dict):
            return self.arks_endpoint.get("name", "")
        return self.arks_endpoint or ""


class ArksTokenSpec(K8sModel):
    token: str = ""
    qos: list[TokenQos] = Field(default_factory=list)


class ArksToken(K8sModel):
    api_version: str = "arks.ai/v1"
    kind: Literal["ArksToken"] = "ArksToken"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    spec: ArksTokenSpec = Field(default_factory=ArksTokenSpec)
    status: dict[str, Any] = Field(default_factory=dict)


class QuotaEntry(K8sModel):
    type: str  # prompt | response | total
    value: int


class QuotaStatusEntry(K8sModel):
    type: str
    used: int = 0
    last_update_time: float = Field(default_factory=time.time)


class ArksQuotaSpec(K8sModel):
    quotas: list[QuotaEntry] = Field(default_factory=list)


class ArksQuotaStatus(K8sModel):
    quota_status: list[QuotaStatusEntry] = Field(default_factory=list)


class ArksQuota(K8sModel):
    api_version: str = "arks.ai/v1"
    kind: Literal["ArksQuota"] = "ArksQuota"
    metadata: ObjectMeta = Field(default_factory=ObjectMeta)
    spec: ArksQuotaSpec = Field(default_factory=ArksQuotaSpec)
    status: ArksQuotaStatus = Field(default_factory=ArksQuotaStatus)


KIND_MAP = {
    "ArksModel": ArksModel,
    "ArksApplication": ArksApplication,
    "ArksDisaggregatedApplication": ArksDisaggregatedApplication,
    "ArksEndpoint": ArksEndpoint,
    "ArksToken": ArksToken,
    "ArksQuota": ArksQuota,
}


def parse_manifest(doc: dict) -> Any:
    """Parse one YAML/JSON document into its typed CR."""
    kind = doc.get("kind")
    cls = KIND_MAP.get(kind)
    if cls is None:
        raise ValueError(f"unknown kind {kind!r}")
    return cls.model_validate(doc)
