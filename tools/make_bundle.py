#!/usr/bin/env python3
"""Assemble the OLM bundle (operator-sdk `make bundle` analog, offline).

registry+v1 layout:
  bundle/manifests/  — the ClusterServiceVersion (CSV base with the
                       samples injected as alm-examples) + both CRDs
  bundle/metadata/annotations.yaml
"""
import json
import os
import shutil
import sys

import yaml

OUT = sys.argv[1] if len(sys.argv) > 1 else "bundle"
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def main() -> int:
    man = os.path.join(OUT, "manifests")
    meta = os.path.join(OUT, "metadata")
    shutil.rmtree(OUT, ignore_errors=True)
    os.makedirs(man)
    os.makedirs(meta)

    with open(os.path.join(REPO, "config/manifests/bases/cro-amd.clusterserviceversion.yaml")) as f:
        csv = yaml.safe_load(f)
    samples = []
    sdir = os.path.join(REPO, "config/samples")
    for fn in sorted(os.listdir(sdir)):
        if fn.endswith(".yaml") and fn != "kustomization.yaml":
            with open(os.path.join(sdir, fn)) as f:
                samples.append(yaml.safe_load(f))
    csv.setdefault("metadata", {}).setdefault("annotations", {})[
        "alm-examples"] = json.dumps(samples, indent=1)
    with open(os.path.join(man, "cro-amd.clusterserviceversion.yaml"), "w") as f:
        yaml.safe_dump(csv, f, sort_keys=False)

    for crd in os.listdir(os.path.join(REPO, "config/crd/bases")):
        shutil.copy(os.path.join(REPO, "config/crd/bases", crd), man)

    with open(os.path.join(meta, "annotations.yaml"), "w") as f:
        yaml.safe_dump({"annotations": {
            "operators.operatorframework.io.bundle.mediatype.v1": "registry+v1",
            "operators.operatorframework.io.bundle.manifests.v1": "manifests/",
            "operators.operatorframework.io.bundle.metadata.v1": "metadata/",
            "operators.operatorframework.io.bundle.package.v1": "cro-amd",
            "operators.operatorframework.io.bundle.channels.v1": "alpha",
        }}, f)
    print(f"bundle assembled at {OUT}/")
    return 0


if __name__ == "__main__":
    sys.exit(main())
