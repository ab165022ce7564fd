"""CRD surface parity: every field in the reference CRD schemas must be implemented
(declared in grove_amd/api/schema.py), passed through (embedded upstream types), or
listed in KNOWN_GAPS with a reason. Runs only where /root/reference is mounted."""
import functools
import glob

import pytest
import yaml

from grove_amd.api.schema import SCHEMAS, KNOWN_GAPS, declared_paths

pytestmark = pytest.mark.reference

CRD_GLOBS = [
    "/root/reference/operator/api/core/v1alpha1/crds/*.yaml",
    "/root/reference/scheduler/api/core/v1alpha1/crds/*.yaml",
]


def reference_paths(crd) -> set:
    out = set()

    def walk(schema, prefix, depth=0):
        if depth > 14:
            return
        for k, v in (schema.get("properties") or {}).items():
            path = f"{prefix}.{k}" if prefix else k
            out.add(path)
            if v.get("type") == "array" and "items" in v:
                walk(v["items"], path, depth + 1)
            else:
                walk(v, path, depth + 1)

    ver = crd["spec"]["versions"][0]
    walk(ver["schema"]["openAPIV3Schema"], "")
    return out


@functools.lru_cache(maxsize=1)
def _crds():
    files = []
    for g in CRD_GLOBS:
        files.extend(glob.glob(g))
    out = {}
    for f in files:
        crd = yaml.safe_load(open(f))
        out[crd["metadata"]["name"]] = crd
    return out


def test_all_reference_crds_covered():
    crds = _crds()
    assert set(crds) == set(SCHEMAS), "CRD inventory mismatch"


@pytest.mark.parametrize("crd_name", sorted(SCHEMAS))
def test_crd_fields_implemented(crd_name):
    crds = _crds()
    ref = reference_paths(crds[crd_name])
    ours = declared_paths(crd_name)
    passthrough_roots = [p[:-2] for p in ours if p.endswith(".*")]
    gaps = set(KNOWN_GAPS.get(crd_name, []))
    missing = []
    for path in sorted(ref):
        if path in ours or path in gaps:
            continue
        if any(path == root or path.startswith(root + ".")
               for root in passthrough_roots):
            continue
        missing.append(path)
    assert not missing, (
        f"{crd_name}: {len(missing)} reference fields not implemented/declared:\n  "
        + "\n  ".join(missing[:40]))


@pytest.mark.parametrize("crd_name", sorted(SCHEMAS))
def test_crd_versions_and_scope(crd_name):
    crds = _crds()
    crd = crds[crd_name]
    ver = crd["spec"]["versions"][0]
    assert ver["name"] == "v1alpha1"
    scope = crd["spec"]["scope"]
    if crd_name == "clustertopologybindings.grove.io":
        assert scope == "Cluster"
    else:
        assert scope == "Namespaced"


def test_committed_crds_match_renderer():
    """crds/*.yaml are generated artifacts — they must match api/crds.py exactly."""
    import os
    from grove_amd.api.crds import render_all
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    for crd in render_all():
        path = os.path.join(repo, "crds", f"{crd['metadata']['name']}.yaml")
        assert os.path.exists(path), f"missing {path} (run install-crds --output-dir crds)"
        on_disk = yaml.safe_load(open(path))
        assert on_disk == crd, f"{path} drifted — regenerate with install-crds"
