#!/usr/bin/env python3
"""Vendor the standard-Kubernetes OpenAPI subtrees embedded in the grove CRDs.

The byte-compatible CRD contract (BASELINE.json north star) embeds large schemas of
UPSTREAM Kubernetes API types — corev1.PodSpec, autoscaling/v2 MetricSpec,
metav1.Condition, resource.k8s.io ResourceClaim template spec — exactly as
controller-gen v0.17.3 renders them. These are mechanical codegen output of the
Kubernetes project's types (Apache-2.0), not Grove-specific design; re-deriving them by
hand would be transcription with typos. This tool extracts them once from the reference
CRD YAML (descriptions stripped — our renderer writes schemas without descriptions and
the parity test ignores descriptions) into grove_amd/api/_k8s_openapi.json, which
grove_amd/api/openapi.py composes with the HAND-DECLARED Grove API structure.

Regenerate with:  python scripts/vendor_k8s_schemas.py [reference_root]
"""
import json
import os
import sys

import yaml

REF = sys.argv[1] if len(sys.argv) > 1 else "/root/reference"
OUT = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                   "grove_amd", "api", "_k8s_openapi.json")


def strip_descriptions(node, is_prop_map=False):
    """Remove schema-metadata `description` strings WITHOUT touching fields that are
    literally named "description" (keys of a `properties` map are field names)."""
    if isinstance(node, dict):
        if is_prop_map:
            return {k: strip_descriptions(v) for k, v in node.items()}
        out = {}
        for k, v in node.items():
            if k == "description" and isinstance(v, str):
                continue
            out[k] = strip_descriptions(v, is_prop_map=(k == "properties"))
        return out
    if isinstance(node, list):
        return [strip_descriptions(x) for x in node]
    return node


def main() -> None:
    opdir = os.path.join(REF, "operator/api/core/v1alpha1/crds")
    pclq = yaml.safe_load(open(os.path.join(opdir, "grove.io_podcliques.yaml")))
    pcs = yaml.safe_load(open(os.path.join(opdir, "grove.io_podcliquesets.yaml")))
    s_pclq = strip_descriptions(pclq["spec"]["versions"][0]["schema"]["openAPIV3Schema"])
    s_pcs = strip_descriptions(pcs["spec"]["versions"][0]["schema"]["openAPIV3Schema"])
    spec = s_pclq["properties"]["spec"]["properties"]
    tmpl = s_pcs["properties"]["spec"]["properties"]["template"]["properties"]
    vendored = {
        "_provenance": "Kubernetes core-type OpenAPI schemas as rendered by "
                       "controller-gen v0.17.3 (extracted from the reference CRD "
                       "YAML by scripts/vendor_k8s_schemas.py; descriptions "
                       "stripped).",
        # corev1.PodSpec
        "podspec": spec["podSpec"],
        # []autoscalingv2.MetricSpec
        "hpa_metrics": spec["autoScalingConfig"]["properties"]["metrics"],
        # metav1.Condition (items schema)
        "metav1_condition":
            s_pclq["properties"]["status"]["properties"]["conditions"]["items"],
        # resource.k8s.io claim template {name, templateSpec}
        "resourceclaim_template": tmpl["resourceClaimTemplates"]["items"],
    }
    with open(OUT, "w") as f:
        json.dump(vendored, f, indent=1, sort_keys=True)
    print(f"wrote {OUT}: " +
          ", ".join(f"{k}={len(json.dumps(v))}B" for k, v in vendored.items()
                    if not k.startswith("_")))


if __name__ == "__main__":
    main()
