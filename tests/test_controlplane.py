"""Control-plane tests: CRD parsing (the reference's own quickstart YAML),
reconciler phase machines, generated manifest shapes, endpoint readiness."""

import os

import pytest
import yaml

from arks_amd.controlplane import Operator, Store
from arks_amd.controlplane import manifests
from arks_amd.crd.types import (
    ApplicationPhase,
    ArksApplication,
    ArksModel,
    ModelPhase,
    parse_manifest,
    model_path,
)

QUICKSTART = "/root/reference/examples/quickstart/quickstart.yaml"


def load_quickstart():
    if not os.path.exists(QUICKSTART):
        pytest.skip("reference quickstart not available")
    with open(QUICKSTART) as f:
        return [parse_manifest(d) for d in yaml.safe_load_all(f) if d]


def test_parse_reference_quickstart():
    objs = load_quickstart()
    kinds = [o.kind for o in objs]
    assert kinds == ["ArksModel", "ArksApplication", "ArksEndpoint", "ArksQuota",
                     "ArksToken"]
    model, app, ep, quota, token = objs
    assert model.spec.model == "Qwen/Qwen2.5-7B-Instruct-1M"
    assert app.spec.tensor_parallel_size == 2 and app.spec.size == 2
    assert app.spec.runtime == "vllm"
    assert ep.spec.default_weight == 5
    assert token.spec.token == "sk-test123456"
    assert token.spec.qos[0].endpoint_name == "qwen-7b"
    assert {q.type: q.value for q in quota.spec.quotas} == {
        "prompt": 100000, "response": 500000, "total": 600000
    }
    assert model_path(model) == "/models/models/default/qwen-7b"


def mk_model(name="m1", with_source=True):
    doc = {
        "apiVersion": "arks.ai/v1",
        "kind": "ArksModel",
        "metadata": {"name": name, "namespace": "default"},
        "spec": {"model": "org/repo"},
    }
    if with_source:
        doc["spec"]["source"] = {"huggingface": {}}
    return parse_manifest(doc)


def mk_app(name="a1", model="m1", runtime="", tp=2, replicas=1):
    return parse_manifest(
        {
            "apiVersion": "arks.ai/v1",
            "kind": "ArksApplication",
            "metadata": {"name": name, "namespace": "default"},
            "spec": {
                "replicas": replicas,
                "size": 1,
                "runtime": runtime,
                "model": {"name": model},
                "tensorParallelSize": tp,
            },
        }
    )


def test_model_reconcile_download_flow():
    store = Store()
    op = Operator(store)
    store.create(mk_model())
    op.reconcile_until_stable()
    model = store.get("ArksModel", "default", "m1")
    assert model.status.phase is ModelPhase.MODEL_LOADING
    assert store.get_opt("PersistentVolumeClaim", "default", "m1") is not None
    pod = store.get("Pod", "default", "arks-worker-m1")
    env = {e["name"]: e.get("value") for e in pod["spec"]["containers"][0]["env"]}
    assert env["MODEL_NAME"] == "org/repo"
    assert env["MODEL_PATH"] == "/models/models/default/m1"
    # flip the pod to Succeeded -> model becomes Ready
    pod["status"] = {"phase": "Succeeded"}
    store.update(pod)
    op.reconcile_until_stable()
    model = store.get("ArksModel", "default", "m1")
    assert model.status.phase is ModelPhase.READY


def test_model_without_source_ready_immediately():
    store = Store()
    op = Operator(store)
    store.create(mk_model(with_source=False))
    op.reconcile_until_stable()
    assert store.get("ArksModel", "default", "m1").status.phase is ModelPhase.READY
    assert store.get_opt("Pod", "default", "arks-worker-m1") is None


def make_model_ready(store, op, name="m1"):
    store.create(mk_model(name))
    op.reconcile_until_stable()
    pod = store.get("Pod", "default", f"arks-worker-{name}")
    pod["status"] = {"phase": "Succeeded"}
    store.update(pod)
    op.reconcile_until_stable()


def test_app_gates_on_model_then_creates_rbgs():
    store = Store()
    op = Operator(store)
    store.create(mk_app())
    op.reconcile_until_stable()
    app = store.get("ArksApplication", "default", "a1")
    assert app.status.phase is ApplicationPhase.LOADING  # model not there yet

    make_model_ready(store, op)
    op.reconcile_until_stable()
    app = store.get("ArksApplication", "default", "a1")
    assert app.status.phase is ApplicationPhase.CREATING
    rbgs = store.get("RoleBasedGroupSet", "default", "a1")
    role = rbgs["spec"]["template"]["roles"][0]
    assert role["name"] == "inference"
    leader_cmd = role["leaderWorkerSet"]["patchLeaderTemplate"]["spec"]["containers"][0][
        "command"
    ]
    # default runtime is OUR engine with the operator flag contract
    joined = " ".join(leader_cmd)
    assert "arks_amd.server" in joined
    assert "--model /models/models/default/m1" in joined
    assert "--served-model-name m1" in joined
    assert "--tensor-parallel-size 2" in joined
    assert "--port 8080" in joined
    svc = store.get("Service", "default", "arks-application-a1")
    assert svc["spec"]["ports"][0]["port"] == 8080
    assert svc["metadata"]["labels"]["prometheus-discovery"] == "true"

    # flip workload ready -> app Running
    rbgs["status"] = {"replicas": 1, "readyReplicas": 1, "updatedReplicas": 1}
    store.update(rbgs)
    op.reconcile_until_stable()
    app = store.get("ArksApplication", "default", "a1")
    assert app.status.phase is ApplicationPhase.RUNNING


def test_app_precheck_rejects_bad_runtime_and_reserved_volume():
    store = Store()
    op = Operator(store)
    store.create(mk_app(name="bad", runtime="tgi"))
    op.reconcile_until_stable()
    assert store.get("ArksApplication", "default", "bad").status.phase is ApplicationPhase.FAILED

    app = mk_app(name="bad2")
    app.spec.instance_spec = {"volumes": [{"name": "models"}]}
    store.create(app)
    op.reconcile_until_stable()
    assert store.get("ArksApplication", "default", "bad2").status.phase is ApplicationPhase.FAILED


def test_vllm_command_compat():
    """The vllm runtime slot keeps the reference's command shape."""
    store = Store()
    op = Operator(store)
    make_model_ready(store, op)
    store.create(mk_app(name="v1app", runtime="vllm", tp=4))
    op.reconcile_until_stable()
    rbgs = store.get("RoleBasedGroupSet", "default", "v1app")
    cmd = " ".join(
        rbgs["spec"]["template"]["roles"][0]["leaderWorkerSet"]["patchLeaderTemplate"][
            "spec"
        ]["containers"][0]["command"]
    )
    assert "vllm.entrypoints.openai.api_server" in cmd
    assert "--tensor-parallel-size 4" in cmd
    assert "multi-node-serving.sh leader" in cmd


def test_endpoint_routes_only_fully_ready_apps():
    store = Store()
    op = Operator(store)
    make_model_ready(store, op)
    store.create(mk_app(name="a1"))
    store.create(mk_app(name="a2"))
    store.create(
        parse_manifest(
            {
                "apiVersion": "arks.ai/v1",
                "kind": "ArksEndpoint",
                "metadata": {"name": "m1", "namespace": "default"},
                "spec": {"defaultWeight": 3,
                         "gatewayRef": {"name": "arks-eg"}},
            }
        )
    )
    op.reconcile_until_stable()
    route = store.get("HTTPRoute", "default", "m1")
    assert route["spec"]["rules"][0]["backendRefs"] == []  # none ready

    rbgs = store.get("RoleBasedGroupSet", "default", "a1")
    rbgs["status"] = {"replicas": 1, "readyReplicas": 1, "updatedReplicas": 1}
    store.update(rbgs)
    op.reconcile_until_stable()
    route = store.get("HTTPRoute", "default", "m1")
    refs = route["spec"]["rules"][0]["backendRefs"]
    assert [r["name"] for r in refs] == ["arks-application-a1"]
    assert refs[0]["weight"] == 3
    # header matches injected for gateway routing
    headers = route["spec"]["rules"][0]["matches"][0]["headers"]
    assert {"type": "Exact", "name": "namespace", "value": "default"} in headers
    assert {"type": "Exact", "name": "model", "value": "m1"} in headers


def test_app_delete_cleans_up():
    store = Store()
    op = Operator(store)
    make_model_ready(store, op)
    store.create(mk_app(name="a1"))
    op.reconcile_until_stable()
    assert store.get_opt("RoleBasedGroupSet", "default", "a1") is not None
    store.mark_deleted("ArksApplication", "default", "a1")
    op.reconcile_until_stable()
    assert store.get_opt("ArksApplication", "default", "a1") is None
    assert store.get_opt("RoleBasedGroupSet", "default", "a1") is None
    assert store.get_opt("Service", "default", "arks-application-a1") is None


def test_disaggregated_app_flow():
    store = Store()
    op = Operator(store)
    make_model_ready(store, op)
    store.create(
        parse_manifest(
            {
                "apiVersion": "arks.ai/v1",
                "kind": "ArksDisaggregatedApplication",
                "metadata": {"name": "d1", "namespace": "default"},
                "spec": {
                    "runtime": "arks",
                    "model": {"name": "m1"},
                    "prefill": {"replicas": 1, "size": 1},
                    "decode": {"replicas": 2, "size": 1},
                    "router": {"replicas": 1},
                },
            }
        )
    )
    op.reconcile_until_stable()
    pre = store.get("LeaderWorkerSet", "default", "d1-prefill")
    dec = store.get("LeaderWorkerSet", "default", "d1-decode")
    cmd = " ".join(
        pre["spec"]["leaderWorkerTemplate"]["leaderTemplate"]["spec"]["containers"][0][
            "command"
        ]
    )
    assert "--disaggregation-mode prefill" in cmd
    assert dec["spec"]["replicas"] == 2
    router = store.get("Deployment", "default", "d1-router")
    rcmd = " ".join(router["spec"]["template"]["spec"]["containers"][0]["command"])
    assert "arks_amd.router" in rcmd and "--pd-disaggregation" in rcmd
    assert "arks.ai/disaggregation-role=prefill" in rcmd
    # readiness
    for lws, n in ((pre, 1), (dec, 2)):
        lws["status"] = {"replicas": n, "readyReplicas": n, "updatedReplicas": n}
        store.update(lws)
    router["status"] = {"replicas": 1, "readyReplicas": 1}
    store.update(router)
    op.reconcile_until_stable()
    dapp = store.get("ArksDisaggregatedApplication", "default", "d1")
    assert dapp.status.phase is ApplicationPhase.RUNNING


def test_app_replica_scale_propagates_to_workload():
    """kubectl scale (spec.replicas change) must propagate to the generated
    workload on the next reconcile — the HPA path (reference :400-457)."""
    store = Store()
    op = Operator(store)
    store.create(mk_app())
    make_model_ready(store, op)
    op.reconcile_until_stable()
    rbgs = store.get("RoleBasedGroupSet", "default", "a1")
    assert rbgs["spec"]["replicas"] == 1

    app = store.get("ArksApplication", "default", "a1")
    app.spec.replicas = 4
    store.update(app)
    op.reconcile_until_stable()
    rbgs = store.get("RoleBasedGroupSet", "default", "a1")
    assert rbgs["spec"]["replicas"] == 4
    # status no longer satisfies 4 replicas -> leaves Running
    appx = store.get("ArksApplication", "default", "a1")
    assert appx.status.phase is not None


def test_operator_run_concurrent_workers():
    """The threaded run() form reconciles events from multiple workers
    without double-running one key concurrently."""
    import threading
    import time

    from arks_amd.controlplane import Operator, Store
    from arks_amd.crd.types import parse_manifest

    store = Store()
    op = Operator(store)
    t = threading.Thread(target=op.run, kwargs={"poll_interval": 0.05,
                                                "workers": 4}, daemon=True)
    t.start()
    for i in range(6):
        store.apply(parse_manifest({
            "apiVersion": "arks.ai/v1", "kind": "ArksModel",
            "metadata": {"name": f"m{i}", "namespace": "default"},
            "spec": {"model": f"org/m{i}",
                     "storage": {"pvc": {"name": f"m{i}", "spec": {
                         "accessModes": ["ReadWriteOnce"],
                         "resources": {"requests": {"storage": "1Gi"}}}}}},
        }))
    deadline = time.time() + 10
    while time.time() < deadline:
        pvcs = store.list("PersistentVolumeClaim", "default")
        if len(pvcs) >= 6:
            break
        time.sleep(0.1)
    op.stop()
    assert len(store.list("PersistentVolumeClaim", "default")) >= 6


def test_operator_metrics_instrumented():
    """Reconciles record controller-runtime-style metric families
    (controller_runtime_reconcile_total etc. — reference manager metrics,
    cmd/main.go:82-98) and the registry renders as Prometheus text."""
    from arks_amd.controlplane import metrics as opmetrics
    from arks_amd.crd.types import parse_manifest

    before = opmetrics.reconcile_total.labels("ArksModel", "success")._value.get()
    store = Store()
    op = Operator(store)
    store.apply(parse_manifest({
        "apiVersion": "arks.ai/v1", "kind": "ArksModel",
        "metadata": {"name": "mm", "namespace": "default"},
        "spec": {"model": "org/mm",
                 "storage": {"pvc": {"name": "mm", "spec": {
                     "accessModes": ["ReadWriteOnce"],
                     "resources": {"requests": {"storage": "1Gi"}}}}}},
    }))
    op.reconcile_until_stable()
    after = opmetrics.reconcile_total.labels("ArksModel", "success")._value.get()
    assert after > before
    text = opmetrics.render().decode()
    assert "controller_runtime_reconcile_total" in text
    assert "controller_runtime_reconcile_time_seconds" in text
    assert "workqueue_depth" in text


def test_operator_metrics_http_endpoint():
    """`python -m arks_amd.controlplane` serves /metrics on --metrics-port
    (the same make_metrics_server the entrypoint mounts)."""
    import http.client
    import threading

    from arks_amd.controlplane.metrics import make_metrics_server

    srv = make_metrics_server(0, host="127.0.0.1")
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    try:
        conn = http.client.HTTPConnection("127.0.0.1", srv.server_port,
                                          timeout=5)
        conn.request("GET", "/metrics")
        resp = conn.getresponse()
        assert resp.status == 200
        assert b"controller_runtime_reconcile" in resp.read()
    finally:
        srv.shutdown()
