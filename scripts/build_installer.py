#!/usr/bin/env python3
"""Bundle deploy/ into single-apply installers (the reference's
`make build-installer` -> dist/operator.yaml + dist/gateway.yaml)."""

import glob
import os

os.makedirs("dist", exist_ok=True)


def bundle(out, paths):
    docs = []
    for p in paths:
        with open(p) as f:
            docs.append(f"# --- {p}\n" + f.read().strip())
    with open(out, "w") as f:
        f.write("\n---\n".join(docs) + "\n")
    print(f"wrote {out} ({sum(1 for _ in open(out))} lines)")


bundle("dist/operator.yaml",
       sorted(glob.glob("deploy/crds/*.yaml")) + ["deploy/operator.yaml",
                                                  "deploy/network-policy.yaml"])
bundle("dist/gateway.yaml", ["deploy/gateway.yaml"])
bundle("dist/gateway-envoy.yaml", ["deploy/gateway-envoy.yaml"])
