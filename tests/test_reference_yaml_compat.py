"""API compatibility: the REFERENCE's own sample YAMLs (read from
/root/reference at test time — public untrusted content used only as parse
fixtures) must load into our pydantic CRD types with their fields intact.
Skipped where the reference checkout is absent."""

import glob
import os

import pytest
import yaml

from arks_amd.crd import types as T

REF = "/root/reference"

KIND_MAP = {
    "ArksModel": T.ArksModel,
    "ArksApplication": T.ArksApplication,
    "ArksDisaggregatedApplication": T.ArksDisaggregatedApplication,
    "ArksEndpoint": T.ArksEndpoint,
    "ArksToken": T.ArksToken,
    "ArksQuota": T.ArksQuota,
}

pytestmark = pytest.mark.skipif(
    not os.path.isdir(REF), reason="reference checkout not present"
)


def _load_docs(path):
    with open(path) as f:
        return [d for d in yaml.safe_load_all(f) if d]


def test_reference_quickstart_parses():
    parsed = {}
    for doc in _load_docs(os.path.join(REF, "examples/quickstart/quickstart.yaml")):
        cls = KIND_MAP.get(doc.get("kind"))
        if cls is None:
            continue
        obj = cls.model_validate(doc)
        parsed[doc["kind"]] = obj
    assert {"ArksModel", "ArksApplication", "ArksEndpoint",
            "ArksToken", "ArksQuota"} <= parsed.keys()
    app = parsed["ArksApplication"]
    assert app.spec.replicas >= 1
    assert app.spec.model.get("name") == parsed["ArksModel"].metadata.name
    tok = parsed["ArksToken"]
    assert tok.spec.token
    assert tok.spec.qos and tok.spec.qos[0].rate_limits
    quota = parsed["ArksQuota"]
    assert quota.spec.quotas and quota.spec.quotas[0].type in (
        "prompt", "response", "total"
    )


def test_all_reference_samples_parse():
    n = 0
    for path in glob.glob(os.path.join(REF, "config/samples/arks_v1_*.yaml")):
        for doc in _load_docs(path):
            cls = KIND_MAP.get(doc.get("kind"))
            if cls is None:
                continue
            obj = cls.model_validate(doc)
            assert obj.metadata.name
            n += 1
    assert n >= 6, f"only parsed {n} sample CRs"


def test_reference_app_sample_drives_command_builder():
    """A reference vLLM sample must round-trip through our command builder
    with the documented flag contract."""
    path = os.path.join(REF, "config/samples/arks_v1_arksapplication_vllm.yaml")
    doc = _load_docs(path)[0]
    app = T.ArksApplication.model_validate(doc)
    from arks_amd.controlplane import commands, manifests

    cmd = commands.leader_command(
        app.spec.runtime or "vllm",
        manifests.model_path_for(app) if hasattr(manifests, "model_path_for")
        else "/models/models/default/m",
        T.served_model_name(app),
        app.spec.tensor_parallel_size or 0,
        list(app.spec.runtime_common_args or []),
        app.spec.size or 1,
    )
    joined = " ".join(cmd)
    assert "--port 8080" in joined
    assert "--served-model-name" in joined
