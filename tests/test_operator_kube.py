"""Operator + KubeStore against the fake API server: the full reconcile
path (finalizers, PVC, download pod, phase transitions, LWS + Service
generation) through REST instead of the in-memory store."""

import httpx

from arks_amd.controlplane.kubestore import KubeStore
from arks_amd.controlplane.operator import Operator
from arks_amd.crd.types import (
    ArksApplication,
    ArksApplicationSpec,
    ArksModel,
    ArksModelSpec,
    ModelPhase,
    ObjectMeta,
)
from tests.test_kubestore import FakeKube


def mk():
    fake = FakeKube()
    store = KubeStore(api_base="https://fake", token="t", verify=False,
                      transport=httpx.MockTransport(fake.handler))
    return fake, store


def drain(store, op, rounds=6):
    for _ in range(rounds):
        store.resync_once()
        op.reconcile_until_stable()


def test_model_to_ready_and_app_to_running():
    fake, store = mk()
    op = Operator(store)
    store.create(ArksModel(
        metadata=ObjectMeta(name="m1", namespace="d"),
        spec=ArksModelSpec(model="org/repo", source={"huggingface": {}}),
    ))
    drain(store, op)
    # PVC + download pod exist; phase is ModelLoading
    assert store.get_opt("PersistentVolumeClaim", "d", "m1") is not None
    pod = store.get_opt("Pod", "d", "arks-worker-m1")
    assert pod is not None
    model = store.get("ArksModel", "d", "m1")
    assert model.status.phase == ModelPhase.MODEL_LOADING
    # flip the pod to Succeeded -> model becomes Ready
    pod["status"] = {"phase": "Succeeded"}
    fake.objects["pods/d/arks-worker-m1"] = pod
    drain(store, op)
    assert store.get("ArksModel", "d", "m1").status.phase == ModelPhase.READY

    # application gated on the model now proceeds to workload creation
    store.create(ArksApplication(
        metadata=ObjectMeta(name="app1", namespace="d"),
        spec=ArksApplicationSpec(
            replicas=1, size=1, runtime="arks", model={"name": "m1"},
            served_model_name="servedm",
        ),
    ))
    drain(store, op)
    assert store.get_opt("LeaderWorkerSet", "d", "app1") is not None or \
        store.get_opt("RoleBasedGroupSet", "d", "app1") is not None
    assert store.get_opt("Service", "d", "arks-application-app1") is not None
