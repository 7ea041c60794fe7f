#!/usr/bin/env python3
"""Generate the arks.ai/v1 CRD manifests from the pydantic API types
(arks_amd/crd/types.py) — the controller-gen step of the reference build
(reference config/crd/bases/, Makefile `manifests` target).

Produces Kubernetes structural schemas: pydantic's $defs are inlined,
Optional[X] anyOf-collapses to X, open models become
x-kubernetes-preserve-unknown-fields, and pass-through pod-spec fields stay
schemaless. Output: deploy/crds/arks.ai_<plural>.yaml

Usage: python scripts/gen_crds.py [--out deploy/crds]
"""

from __future__ import annotations

import argparse
import copy
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import yaml

from arks_amd.crd import types as T

KINDS = [
    # (kind, plural, spec model, status model, printer columns)
    ("ArksModel", "arksmodels", T.ArksModelSpec, T.ArksModelStatus,
     [("Phase", ".status.phase"), ("Model", ".spec.model")]),
    ("ArksApplication", "arksapplications", T.ArksApplicationSpec,
     T.ArksApplicationStatus,
     [("Phase", ".status.phase"), ("Replicas", ".status.replicas"),
      ("Ready", ".status.readyReplicas")]),
    ("ArksDisaggregatedApplication", "arksdisaggregatedapplications",
     T.ArksDisaggregatedApplicationSpec, T.ArksDisaggregatedApplicationStatus,
     [("Phase", ".status.phase")]),
    ("ArksEndpoint", "arksendpoints", T.ArksEndpointSpec, T.ArksEndpointStatus,
     []),
    ("ArksToken", "arkstokens", T.ArksTokenSpec, None, []),
    ("ArksQuota", "arksquotas", T.ArksQuotaSpec, T.ArksQuotaStatus, []),
]


def _inline(schema: dict, defs: dict) -> dict:
    """Recursively inline $refs and normalize to a K8s structural schema."""
    if "$ref" in schema:
        name = schema["$ref"].split("/")[-1]
        return _inline(copy.deepcopy(defs[name]), defs)
    # Optional[X]: anyOf [X, {type: null}] -> X
    if "anyOf" in schema:
        non_null = [s for s in schema["anyOf"] if s.get("type") != "null"]
        if len(non_null) == 1:
            merged = {k: v for k, v in schema.items() if k != "anyOf"}
            inner = _inline(non_null[0], defs)
            inner.update({k: v for k, v in merged.items()
                          if k in ("description", "default") and v is not None})
            if inner.get("default") is None:
                inner.pop("default", None)
            return inner
        schema["anyOf"] = [_inline(s, defs) for s in non_null]
    out = dict(schema)
    out.pop("title", None)
    # pydantic extra="allow" / dict[str, Any] / Any
    if out.get("additionalProperties") is True or out == {}:
        out.pop("additionalProperties", None)
        out.setdefault("type", "object")
        out["x-kubernetes-preserve-unknown-fields"] = True
    if isinstance(out.get("additionalProperties"), dict):
        out["additionalProperties"] = _inline(out["additionalProperties"], defs)
    if "properties" in out:
        out["properties"] = {
            k: _inline(v, defs) for k, v in out["properties"].items()
        }
        out.setdefault("type", "object")
    if "items" in out:
        out["items"] = _inline(out["items"], defs)
    if "const" in out:
        out["enum"] = [out.pop("const")]
    if "default" in out and out["default"] is None:
        out.pop("default")
    # structural schemas reject unknown keywords pydantic may emit
    for bad in ("discriminator", "examples", "$defs"):
        out.pop(bad, None)
    return out


def model_schema(model) -> dict:
    raw = model.model_json_schema(by_alias=True)
    defs = raw.pop("$defs", {})
    return _inline(raw, defs)


def crd_for(kind, plural, spec_model, status_model, columns) -> dict:
    props = {
        "apiVersion": {"type": "string"},
        "kind": {"type": "string"},
        "metadata": {"type": "object"},
        "spec": model_schema(spec_model),
    }
    if status_model is not None:
        props["status"] = model_schema(status_model)
    version = {
        "name": "v1",
        "served": True,
        "storage": True,
        "schema": {
            "openAPIV3Schema": {
                "type": "object",
                "properties": props,
            }
        },
        "subresources": {"status": {}},
        "additionalPrinterColumns": [
            {"name": n, "type": "string", "jsonPath": p} for n, p in columns
        ] or None,
    }
    if version["additionalPrinterColumns"] is None:
        del version["additionalPrinterColumns"]
    if kind in ("ArksApplication", "ArksDisaggregatedApplication"):
        # HPA / kubectl scale compatibility (reference README.md:18)
        version["subresources"]["scale"] = {
            "specReplicasPath": ".spec.replicas",
            "statusReplicasPath": ".status.replicas",
        }
    return {
        "apiVersion": "apiextensions.k8s.io/v1",
        "kind": "CustomResourceDefinition",
        "metadata": {"name": f"{plural}.arks.ai"},
        "spec": {
            "group": "arks.ai",
            "names": {
                "kind": kind,
                "listKind": f"{kind}List",
                "plural": plural,
                "singular": kind.lower(),
            },
            "scope": "Namespaced",
            "versions": [version],
        },
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default="deploy/crds")
    args = ap.parse_args()
    os.makedirs(args.out, exist_ok=True)
    for kind, plural, spec, status, cols in KINDS:
        crd = crd_for(kind, plural, spec, status, cols)
        path = os.path.join(args.out, f"arks.ai_{plural}.yaml")
        with open(path, "w") as f:
            yaml.safe_dump(crd, f, sort_keys=False)
        print(f"wrote {path}")


if __name__ == "__main__":
    main()
