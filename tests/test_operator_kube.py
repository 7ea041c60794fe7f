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


def test_leader_election_lease():
    """Two candidates on one Lease: only one leads; the other takes over
    after expiry; renewal keeps leadership; release vacates."""
    from arks_amd.controlplane.leaderelect import LeaderElector

    fake, store = mk()
    t = [1000.0]
    clock = lambda: t[0]  # noqa: E731
    a = LeaderElector(store, "op-a", lease_s=15, renew_s=5, clock=clock)
    b = LeaderElector(store, "op-b", lease_s=15, renew_s=5, clock=clock)

    assert a.try_acquire() and a.is_leader
    assert not b.try_acquire() and not b.is_leader
    lease = store.get_opt("Lease", "arks-system", a.name)
    assert lease["spec"]["holderIdentity"] == "op-a"

    # renewal within the window keeps op-a leading and blocks op-b
    t[0] += 10
    assert a.try_acquire()
    t[0] += 10  # 10s since op-a's renewal < 15s lease
    assert not b.try_acquire()

    # op-a goes silent -> op-b steals after expiry with a transition bump
    t[0] += 20
    assert b.try_acquire() and b.is_leader
    lease = store.get_opt("Lease", "arks-system", a.name)
    assert lease["spec"]["holderIdentity"] == "op-b"
    assert lease["spec"]["leaseTransitions"] == 1
    # op-a notices it lost
    assert not a.try_acquire() and not a.is_leader

    # release vacates immediately; op-a re-acquires without waiting
    b.release()
    assert a.try_acquire()
    lease = store.get_opt("Lease", "arks-system", a.name)
    assert lease["spec"]["holderIdentity"] == "op-a"
    assert lease["spec"]["leaseTransitions"] == 2


def test_leader_election_renew_loop_and_loss_callback():
    """run_renew keeps renewing while leading and fires on_lost exactly
    once when another candidate has stolen the Lease."""
    import threading

    from arks_amd.controlplane.leaderelect import LeaderElector

    fake, store = mk()
    t = [0.0]
    clock = lambda: t[0]  # noqa: E731
    a = LeaderElector(store, "op-a", lease_s=15, renew_s=0.01, clock=clock)
    b = LeaderElector(store, "op-b", lease_s=15, renew_s=5, clock=clock)
    assert a.try_acquire()
    lost = threading.Event()
    th = threading.Thread(target=a.run_renew, args=(lost.set,), daemon=True)
    th.start()
    # renewals at a fixed clock keep op-a leading
    import time as _time

    _time.sleep(0.1)
    assert a.is_leader and not lost.is_set()
    # op-a's clock stalls 20s while op-b advances -> op-b steals; op-a's
    # next renewal must fail and report loss
    t[0] += 20.0
    assert b.try_acquire()
    lost.wait(timeout=5.0)
    assert lost.is_set() and not a.is_leader
    th.join(timeout=5.0)
    assert not th.is_alive()
